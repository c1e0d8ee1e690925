"""GPU numerics tests: every gfx950 HIP kernel vs the plain fp32 torch
oracle (computed on the same bf16-rounded inputs).

Asymmetric random operands throughout — transpose-detecting
(cdna_hip_programming.md §5.4 rule 16).
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def hip():
    from bflc_amd.ops import functional as fn
    return fn.hip_ops()


def bf(x):
    return x.to(DEV, torch.bfloat16).contiguous()



def assert_close(y, ref, rel=0.02):
    y = y.float().cpu()
    ref = ref.float().cpu()
    scale = ref.abs().max().clamp_min(1.0)
    torch.testing.assert_close(y, ref, rtol=rel, atol=float(scale) * rel)


class TestLinear:
    @pytest.mark.parametrize("m,k,n", [
        (100, 5, 2),        # reference logreg shape (main.py:120)
        (100, 3136, 128),   # FEMNIST fc1
        (100, 128, 62),     # FEMNIST head
        (37, 100, 62),      # odd sizes / predication
        (256, 512, 512),    # square-ish (256x256 double-buffered path)
        (512, 2304, 256),   # resnet50-mid shape class: 256-path + split-K
        (768, 320, 256),    # 256-path, K not a multiple of 256
        (2036, 5, 2),       # sponsor global test shape
    ])
    def test_fwd(self, m, k, n):
        torch.manual_seed(0)
        x = torch.randn(m, k)
        w = torch.randn(k, n)
        b = torch.randn(n)
        y = hip().linear_fwd(bf(x), bf(w), bf(b))
        ref = bf(x).float().cpu() @ bf(w).float().cpu() + bf(b).float().cpu()
        assert_close(y, ref)

    @pytest.mark.parametrize("m,k,n", [
        # thin VALU path (K<=32, N<=64, M>=64k, B [N][K]):
        (131072, 16, 32),   # FEMNIST conv1 GEMM shape
        (131072, 32, 16),   # CIFAR stem shape
    ])
    def test_thin_gemm(self, m, k, n):
        torch.manual_seed(3)
        a = torch.randn(m, k)
        bn = torch.randn(n, k)  # B in [N][K] layout (tb=True)
        y = hip().gemm_raw(bf(a), bf(bn), False, True)
        ref = bf(a).float().cpu() @ bf(bn).float().cpu().t()
        assert_close(y, ref)

    @pytest.mark.parametrize("m,k,n", [(100, 5, 2), (64, 3136, 128),
                                       (37, 100, 62), (128, 256, 192),
                                       # 256-path TB (dx) / TA (dw) layouts
                                       (256, 512, 512), (512, 2304, 256)])
    def test_bwd(self, m, k, n):
        torch.manual_seed(1)
        x, w, dy = torch.randn(m, k), torch.randn(k, n), torch.randn(m, n)
        dx, dw, db = hip().linear_bwd(bf(x), bf(w), bf(dy))
        xf, wf, dyf = (bf(t).float().cpu() for t in (x, w, dy))
        assert_close(dx, dyf @ wf.t())
        assert_close(dw, xf.t() @ dyf)
        assert_close(db, dyf.sum(0))


class TestSoftmaxCE:
    @pytest.mark.parametrize("m,c", [(100, 2), (256, 62), (64, 1000), (7, 10)])
    def test_fwd_bwd(self, m, c):
        torch.manual_seed(2)
        logits = torch.randn(m, c) * 3
        y = torch.randint(0, c, (m,))
        loss, probs = hip().softmax_ce_fwd(bf(logits), y.to(DEV))
        lf = bf(logits).float().cpu()
        ref_loss = torch.nn.functional.cross_entropy(lf, y)
        assert abs(float(loss) - float(ref_loss)) < 2e-3 * max(1, ref_loss.abs())
        assert_close(probs, torch.softmax(lf, dim=1), rel=0.01)

        g = torch.tensor(2.0)
        dl = hip().softmax_ce_bwd(probs, y.to(DEV), g.to(DEV))
        l2 = lf.clone().requires_grad_(True)
        (torch.nn.functional.cross_entropy(l2, y) * 2.0).backward()
        assert_close(dl, l2.grad, rel=0.03)


class TestConv:
    @pytest.mark.parametrize("n,c,h,k,r,stride,pad", [
        (4, 1, 28, 32, 3, 1, 1),   # FEMNIST conv1
        (4, 32, 14, 64, 3, 1, 1),  # FEMNIST conv2
        (2, 3, 32, 16, 3, 1, 1),   # resnet20 stem
        (2, 16, 32, 32, 3, 2, 1),  # strided downsample
        (2, 8, 9, 8, 3, 1, 0),     # odd, no pad
        (2, 8, 8, 16, 1, 1, 0),    # 1x1 conv
        (2, 3, 32, 16, 7, 2, 3),   # resnet50 stem 7x7/2
    ])
    def test_fwd_bwd(self, n, c, h, k, r, stride, pad):
        # NHWC x [N,H,W,C], w [Kout,R,S,C]; torch NCHW reference
        torch.manual_seed(3)
        x = torch.randn(n, h, h, c)
        w = torch.randn(k, r, r, c) * (1.0 / math.sqrt(c * r * r))
        b = torch.randn(k)
        y = hip().conv2d_fwd(bf(x), bf(w), bf(b), stride, pad)
        xf, wf, bfl = (bf(t).float().cpu() for t in (x, w, b))
        xn, wn = xf.permute(0, 3, 1, 2), wf.permute(0, 3, 1, 2)
        ref = torch.nn.functional.conv2d(xn, wn, bfl, stride=stride,
                                         padding=pad).permute(0, 2, 3, 1)
        assert y.shape == ref.shape
        assert_close(y, ref)

        dy = torch.randn_like(ref)
        dx, dw, db = hip().conv2d_bwd(bf(x), bf(w), bf(dy), stride, pad)
        x2 = xn.clone().requires_grad_(True)
        w2 = wn.clone().requires_grad_(True)
        b2 = bfl.clone().requires_grad_(True)
        out = torch.nn.functional.conv2d(x2, w2, b2, stride=stride,
                                         padding=pad)
        (out * bf(dy).float().cpu().permute(0, 3, 1, 2)).sum().backward()
        assert_close(dx, x2.grad.permute(0, 2, 3, 1), rel=0.03)
        assert_close(dw, w2.grad.permute(0, 2, 3, 1), rel=0.03)
        assert_close(db, b2.grad, rel=0.03)


class TestPoolReluAcc:
    def test_conv_fused_relu_matches_separate(self):
        from bflc_amd.ops import functional as O
        torch.manual_seed(7)
        dev = torch.device("cuda", 0)
        x = torch.randn(4, 14, 14, 32, device=dev,
                        dtype=torch.bfloat16).requires_grad_(True)
        w = torch.randn(64, 3, 3, 32, device=dev,
                        dtype=torch.bfloat16).requires_grad_(True)
        b = torch.randn(64, device=dev,
                        dtype=torch.bfloat16).requires_grad_(True)
        y = O.conv2d(x, w, b, 1, 1, relu=True)
        dy = torch.randn_like(y)
        (y.float() * dy.float()).sum().backward()
        g = [t.grad.clone() for t in (x, w, b)]
        for t in (x, w, b):
            t.grad = None
        y2 = O.relu(O.conv2d(x, w, b, 1, 1))
        assert torch.equal(y, y2)
        (y2.float() * dy.float()).sum().backward()
        for a, t in zip(g, (x, w, b)):
            assert torch.equal(a, t.grad)

    def test_maxpool(self):
        torch.manual_seed(4)
        x = torch.randn(3, 14, 14, 8)  # NHWC
        y, idx = hip().maxpool2d_fwd(bf(x), 2, 2)
        xn = bf(x).float().cpu().permute(0, 3, 1, 2)
        ref, ridx = torch.nn.functional.max_pool2d(
            xn, 2, 2, return_indices=True)
        assert_close(y, ref.permute(0, 2, 3, 1), rel=0.01)
        dy = torch.randn_like(ref)
        dx = hip().maxpool2d_bwd(bf(dy.permute(0, 2, 3, 1).contiguous()),
                                 idx, [3, 14, 14, 8], 2, 2)
        ref_dx = torch.nn.functional.max_unpool2d(
            bf(dy).float().cpu(), ridx, 2, 2, output_size=(14, 14))
        assert_close(dx, ref_dx.permute(0, 2, 3, 1), rel=0.01)

    def test_relu(self):
        x = torch.randn(1000)
        y = hip().relu_fwd(bf(x))
        assert torch.equal(y.cpu().float(), torch.relu(bf(x).cpu().float()))
        dy = torch.randn(1000)
        dx = hip().relu_bwd(y, bf(dy))
        ref = bf(dy).cpu().float() * (bf(x).cpu().float() > 0)
        assert_close(dx, ref, rel=0.01)

    def test_accuracy(self):
        torch.manual_seed(5)
        logits = torch.randn(1000, 10)
        y = torch.randint(0, 10, (1000,))
        a = hip().accuracy(bf(logits), y.to(DEV))
        ref = (bf(logits).float().cpu().argmax(1) == y).float().mean()
        assert a == pytest.approx(float(ref), abs=1e-6)


class TestFlatOps:
    def test_axpy_sgd_exact(self):
        y = torch.randn(100001, device=DEV)
        x = torch.randn(100001, device=DEV)
        y2 = y.clone()
        hip().axpy_(y, x, 2.5)
        # kernel uses fmaf (one rounding); torch mul+add rounds twice
        assert torch.allclose(y, y2 + 2.5 * x, atol=1e-5)

        p = torch.randn(12345, device=DEV)
        g = torch.randn(12345, device=DEV)
        p2 = p.clone()
        hip().sgd_step_(p, g, 0.1)
        ref = torch.tensor(0.1, device=DEV)  # fmaf(-lr, g, p)
        assert torch.allclose(p, p2 - 0.1 * g, atol=1e-7)

    def test_adam_matches_cpu(self):
        from bflc_amd.ops import functional as O
        torch.manual_seed(6)
        p = torch.randn(5000)
        g = torch.randn(5000)
        m = torch.zeros(5000)
        v = torch.zeros(5000)
        pg, gg, mg, vg = (t.to(DEV) for t in (p, g, m, v))
        for step in (1, 2, 3):
            O.adam_step_(p, g, m, v, step, 0.01)
            hip().adam_step_(pg, gg, mg, vg, step, 0.01, 0.9, 0.999, 1e-8)
        assert torch.allclose(pg.cpu(), p, atol=1e-5)

    def test_conv_bwd_skips_dx_for_first_layer(self):
        """want_dx=False (first-layer convs: the input is data) must
        skip the dgrad work, return an empty dx, and leave dw/db
        bitwise identical to the full backward."""
        torch.manual_seed(17)
        x = torch.randn(8, 14, 14, 16, device=DEV).bfloat16()
        w = torch.randn(32, 3, 3, 16, device=DEV).bfloat16() * 0.1
        dy = torch.randn(8, 14, 14, 32, device=DEV).bfloat16()
        dx1, dw1, db1 = hip().conv2d_bwd(x, w, dy, 1, 1)
        dx0, dw0, db0 = hip().conv2d_bwd(x, w, dy, 1, 1, want_dx=False)
        assert dx0.numel() == 0 and dx1.numel() == x.numel()
        assert torch.equal(dw0, dw1) and torch.equal(db0, db1)

    def test_score_load_bitwise_vs_old_chain(self):
        """score_load_ (one pass: shadow = bf16(global - lr*delta))
        must be BITWISE equal to the copy + axpy + cast chain it
        replaced — both round the same fmaf(-lr, d, g) fp32 value
        once."""
        torch.manual_seed(21)
        g = torch.randn(100003, device=DEV)
        d = torch.randn(100003, device=DEV)
        lr = 0.003
        shadow = torch.empty(100003, device=DEV, dtype=torch.bfloat16)
        hip().score_load_(shadow, g, d, lr)
        cand = g.clone()
        hip().axpy_(cand, d, -lr)
        ref = torch.empty_like(shadow)
        hip().refresh_shadow_(cand, ref)
        assert torch.equal(shadow, ref)
        # fp32 variant (fp32-compute models: cflat IS flat)
        s32 = torch.empty(100003, device=DEV)
        hip().score_load_(s32, g, d, lr)
        assert torch.equal(s32, cand)

    def test_delta_extract_matches_old_chain(self):
        """delta_extract_ (one pass: (global - w)/lr, IEEE fp32 divide)
        vs the clone + axpy(-1) + scalar-div chain it replaced: the
        subtraction is bitwise identical; torch's scalar div_ compiles
        to a reciprocal multiply, so the quotient may differ by 1 ulp —
        assert subtraction bitwise via lr=1, and <=1-ulp on the
        divide."""
        torch.manual_seed(22)
        g = torch.randn(54321, device=DEV)
        w = torch.randn(54321, device=DEV)
        out = torch.empty_like(g)
        hip().delta_extract_(out, g, w, 1.0)
        ref1 = g.clone()
        hip().axpy_(ref1, w, -1.0)
        assert torch.equal(out, ref1)  # subtraction path: bitwise

        lr = 0.01
        hip().delta_extract_(out, g, w, lr)
        ref = ref1.div_(lr)
        ulp = torch.abs(ref) * 2 ** -23 + 2 ** -126
        assert (torch.abs(out - ref) <= ulp).all(), \
            float(torch.abs(out - ref).max())

    def test_thin_conv_implicit_matches_col_path(self):
        """The grad-free conv forward (gemm_thin_conv_kernel: window
        gathered inside the GEMM, no im2col materialization) must be
        BITWISE equal to the col-backed path — same taps, same KT
        padding, same dot order."""
        torch.manual_seed(11)
        # M = N*OH*OW must clear the 65536 implicit gate (first four);
        # the small fifth shape checks the im2col fallback stays wired
        for (n, h, c, k, r, stride, pad) in [
            (96, 28, 1, 32, 3, 1, 1),    # FEMNIST conv1 (C=1, kp=16)
            (96, 32, 3, 16, 3, 1, 1),    # CIFAR stem (C=3, kp=32)
            (512, 28, 1, 32, 3, 2, 1),   # strided
            (128, 28, 1, 32, 3, 1, 0),   # no padding
            (16, 28, 1, 32, 3, 1, 1),    # below the gate: fallback
        ]:
            x = torch.randn(n, h, h, c, device=DEV).bfloat16()
            w = (torch.randn(k, r, r, c, device=DEV) * 0.1).bfloat16()
            b = torch.randn(k, device=DEV).bfloat16()
            for relu in (False, True):
                y_col, col = hip().conv2d_fwd_col(x, w, b, stride, pad,
                                                  relu, True)
                y_imp = hip().conv2d_fwd(x, w, b, stride, pad, relu)
                assert torch.equal(y_col, y_imp), \
                    (n, h, c, k, stride, pad, relu)

    def test_relu_bwd_colsum_matches_separate_ops(self):
        """Fused relu-backward+colsum == relu_bwd then the fp32 oracle
        colsum (bitwise dx; db vs fp32 oracle within bf16-out rounding)."""
        torch.manual_seed(12)
        for (m, n) in [(100000, 32), (524288, 64), (4096, 8)]:
            y = torch.randn(m, n, device=DEV).bfloat16()
            dy = torch.randn(m, n, device=DEV).bfloat16()
            dx_f, db_f = hip().relu_bwd_colsum(y, dy)
            dx_ref = hip().relu_bwd(y, dy)
            assert torch.equal(dx_f, dx_ref)
            db_oracle = dx_ref.float().sum(0)
            assert torch.allclose(db_f.float(), db_oracle,
                                  atol=max(1.0, 3e-2 * m ** 0.5),
                                  rtol=2e-2)

    def test_adam_master_matches_fp32_oracle(self):
        """Fused fp32-master/bf16-shadow Adam vs a plain fp32 torch
        oracle fed the bf16-rounded grads (same inputs the kernel sees).
        """
        torch.manual_seed(16)
        p = torch.randn(5000)
        m = torch.zeros(5000)
        v = torch.zeros(5000)
        pg, mg, vg = (t.to(DEV) for t in (p, m, v))
        sh = pg.bfloat16()
        b1, b2, eps, lr = 0.9, 0.999, 1e-8, 0.01
        for step in (1, 2, 3):
            g = torch.randn(5000, device=DEV).bfloat16()
            hip().adam_master_(pg, sh, g, mg, vg, step, lr, b1, b2, eps)
            gf = g.float().cpu()
            m = b1 * m + (1 - b1) * gf
            v = b2 * v + (1 - b2) * gf * gf
            p -= lr * (m / (1 - b1 ** step)) / \
                ((v / (1 - b2 ** step)).sqrt() + eps)
        assert torch.allclose(pg.cpu(), p, atol=2e-5)
        assert torch.equal(sh, pg.bfloat16())  # shadow = bf16(master)

    def test_adam_master_graph_matches_host_step(self):
        """The device-step variant (adam_tick + adam_master_dev) must
        reproduce the host-step adam_master_ sequence exactly — this is
        what makes the Adam train step hipGraph-capturable."""
        torch.manual_seed(17)
        p0 = torch.randn(4096, device=DEV)
        gs = [torch.randn(4096, device=DEV).bfloat16() for _ in range(4)]
        # host-step reference
        p_a = p0.clone(); sh_a = p_a.bfloat16()
        m_a = torch.zeros_like(p0); v_a = torch.zeros_like(p0)
        for step, g in enumerate(gs, 1):
            hip().adam_master_(p_a, sh_a, g, m_a, v_a, step, 0.01,
                               0.9, 0.999, 1e-8)
        # device-step pair
        p_b = p0.clone(); sh_b = p_b.bfloat16()
        m_b = torch.zeros_like(p0); v_b = torch.zeros_like(p0)
        step_t = torch.zeros(1, dtype=torch.int32, device=DEV)
        bc = torch.zeros(2, device=DEV)
        for g in gs:
            hip().adam_master_graph_(p_b, sh_b, g, m_b, v_b, step_t, bc,
                                     0.01, 0.9, 0.999, 1e-8)
        assert int(step_t.item()) == 4
        # identical math modulo device-vs-host powf in the corrections
        assert torch.allclose(p_a, p_b, atol=1e-6, rtol=1e-6)
        assert torch.equal(sh_b, p_b.bfloat16())

    def test_fedavg_matches_and_deterministic(self):
        torch.manual_seed(7)
        deltas = torch.randn(6, 200000, device=DEV)
        w = torch.tensor([305., 300., 310., 290., 305., 295.], device=DEV)
        a1 = hip().weighted_fedavg(deltas, w)
        a2 = hip().weighted_fedavg(deltas, w)
        assert torch.equal(a1, a2)  # bitwise determinism
        ref = torch.zeros(200000)
        dc = deltas.cpu()
        for k in range(6):
            ref += dc[k] * w[k].cpu()
        ref /= w.sum().cpu()
        assert torch.allclose(a1.cpu(), ref, atol=1e-4)


class TestEndToEndGPU:
    def test_femnist_round_and_determinism(self):
        import __graft_entry__ as ge
        ge.smoke()

    def test_logreg_fl_learns_on_gpu(self):
        from bflc_amd.config import FLConfig
        from bflc_amd.comm import Transport
        from bflc_amd.data import make_federated
        from bflc_amd.fl import FLEngine
        cfg = FLConfig()  # 20 clients, reference constants
        shards, test = make_federated(cfg)
        t = Transport(device=torch.device(DEV))
        eng = FLEngine(cfg, t, shards, test)
        stats = eng.run(8)
        acc = eng.evaluate_global()
        assert acc > 0.85
        assert stats[-1].global_loss < stats[0].global_loss


class TestEvalFastPaths:
    def test_maxpool_eval_skips_mask_and_bwd_raises_loudly(self):
        """want_idx=False must produce identical pooled values with an
        empty mask, and feeding that mask to the backward must raise a
        diagnosable error instead of a device fault (regression: grad
        mode is always off inside Function.forward, so the wrapper
        computes want_idx — a bug here once faulted at dispatch)."""
        import pytest as _pytest
        x = torch.randn(64, 28, 28, 32, device=DEV).bfloat16()
        y1, idx1 = hip().maxpool2d_fwd(x, 2, 2, True)
        y2, idx2 = hip().maxpool2d_fwd(x, 2, 2, False)
        assert torch.equal(y1, y2)
        assert idx1.numel() == y1.numel() and idx2.numel() == 0
        dy = torch.randn_like(y1)
        with _pytest.raises(RuntimeError, match="mask"):
            hip().maxpool2d_bwd(dy, idx2, list(x.shape), 2, 2)

    def test_train_conv_backward_works_after_eval_forwards(self):
        """The exact crash scenario: grad-free forwards (scoring) then
        a training forward+backward through the same model — the train
        pass must save real masks/cols even though eval passes do not."""
        from bflc_amd.config import FLConfig
        from bflc_amd.models import build_model
        cfg = FLConfig.for_world(1, model="femnist_cnn", n_class=62)
        m = build_model(cfg, DEV)
        xe = torch.randn(256, 28, 28, 1, device=DEV)
        ye = torch.randint(0, 62, (256,), device=DEV)
        m.accuracy_t(xe, ye)          # eval (no_grad): no masks saved
        loss = m.loss(xe, ye)         # train forward
        loss.backward()               # must not fault / raise
        torch.cuda.synchronize()
        assert m.grad_flat().abs().sum() > 0
