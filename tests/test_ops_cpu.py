"""CPU oracle tests for the op layer (the same oracles the GPU numerics
tests compare the HIP kernels against)."""
import pytest
import torch
import torch.nn.functional as F

from bflc_amd.ops import functional as O


def test_linear_matches_torch():
    x = torch.randn(32, 16, requires_grad=True)
    w = torch.randn(16, 8, requires_grad=True)
    b = torch.randn(8, requires_grad=True)
    y = O.linear(x, w, b)
    ref = x @ w + b
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    assert x.grad is not None and w.grad is not None and b.grad is not None


def test_softmax_ce_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(64, 10, requires_grad=True)
    y = torch.randint(0, 10, (64,))
    loss = O.softmax_cross_entropy(logits, y)
    ref = F.cross_entropy(logits, y)
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    logits2 = logits.detach().clone().requires_grad_(True)
    F.cross_entropy(logits2, y).backward()
    assert torch.allclose(logits.grad, logits2.grad, atol=1e-6)


def test_conv2d_matches_torch():
    # NHWC x [N,H,W,C], w [Kout,R,S,C] vs the torch NCHW reference
    torch.manual_seed(0)
    x = torch.randn(4, 14, 14, 3, requires_grad=True)
    w = torch.randn(8, 3, 3, 3, requires_grad=True)
    b = torch.randn(8, requires_grad=True)
    y = O.conv2d(x, w, b, stride=1, padding=1)
    ref = F.conv2d(x.permute(0, 3, 1, 2), w.permute(0, 3, 1, 2), b,
                   stride=1, padding=1).permute(0, 2, 3, 1)
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    (y * g).sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    (F.conv2d(x2.permute(0, 3, 1, 2), w2.permute(0, 3, 1, 2), b2,
              stride=1, padding=1).permute(0, 2, 3, 1) * g).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
    assert torch.allclose(b.grad, b2.grad, atol=1e-4)


def test_maxpool_matches_torch():
    x = torch.randn(2, 8, 8, 4, requires_grad=True)  # NHWC
    y = O.maxpool2d(x, 2)
    ref = F.max_pool2d(x.permute(0, 3, 1, 2), 2).permute(0, 2, 3, 1)
    assert torch.equal(y, ref)
    y.sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    F.max_pool2d(x2.permute(0, 3, 1, 2), 2).sum().backward()
    assert torch.equal(x.grad, x2.grad)


def test_accuracy():
    logits = torch.tensor([[1.0, 0.0], [0.0, 1.0], [2.0, 1.0]])
    y = torch.tensor([0, 1, 1])
    assert O.accuracy(logits, y) == pytest.approx(2 / 3)


def test_weighted_fedavg_fixed_order():
    deltas = torch.randn(6, 100)
    w = torch.tensor([100., 50., 25., 300., 10., 5.])
    avg = O.weighted_fedavg(deltas, w)
    ref = (deltas * w[:, None]).sum(0) / w.sum()
    assert torch.allclose(avg, ref, atol=1e-5)


def test_sgd_and_axpy():
    p = torch.ones(10)
    g = torch.full((10,), 2.0)
    O.sgd_step_(p, g, lr=0.5)
    assert torch.allclose(p, torch.zeros(10))
    y = torch.ones(10)
    O.axpy_(y, 3.0, torch.ones(10))
    assert torch.allclose(y, torch.full((10,), 4.0))


def test_score_load_and_delta_extract():
    """CPU fallbacks of the fused one-pass ops (the GPU kernels are
    oracle-tested in test_ops_gpu.py): score_load_ writes
    compute_dtype(global - lr*delta); delta_extract_ writes
    (global - w)/lr."""
    torch.manual_seed(3)
    g = torch.randn(1000)
    d = torch.randn(1000)
    lr = 0.01
    out32 = torch.empty(1000)
    O.score_load_(out32, g, d, lr)
    assert torch.allclose(out32, g - lr * d, atol=1e-7)
    out16 = torch.empty(1000, dtype=torch.bfloat16)
    O.score_load_(out16, g, d, lr)
    assert torch.equal(out16, (g - lr * d).bfloat16())

    w = torch.randn(1000)
    delta = torch.empty(1000)
    O.delta_extract_(delta, g, w, lr)
    # bitwise match with the clone+axpy+div chain it replaces
    ref = g.clone()
    O.axpy_(ref, -1.0, w)
    ref.div_(lr)
    assert torch.equal(delta, ref)


def test_adam_matches_torch_adam():
    torch.manual_seed(1)
    p = torch.randn(64)
    g = torch.randn(64)
    m = torch.zeros(64)
    v = torch.zeros(64)
    p_ours = p.clone()
    O.adam_step_(p_ours, g, m, v, step=1, lr=0.01)

    p_ref = p.clone().requires_grad_(True)
    opt = torch.optim.Adam([p_ref], lr=0.01, eps=1e-8)
    p_ref.grad = g.clone()
    opt.step()
    assert torch.allclose(p_ours, p_ref.detach(), atol=1e-6)


def test_gpu_path_requires_hip_ext():
    """On CUDA tensors the ops must NOT silently fall back to eager."""
    import bflc_amd.ops.functional as fn
    if fn.hip_available():
        pytest.skip("hip ext built")
    x = torch.randn(2, 2)

    class FakeCuda:
        pass
    # hip_ops() raises when the extension is missing
    with pytest.raises(RuntimeError, match="HIP"):
        fn.hip_ops()
