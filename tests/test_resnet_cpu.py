"""ResNet model family tests (CPU oracle path)."""
import pytest
import torch
import torch.nn.functional as F

from bflc_amd.config import FLConfig
from bflc_amd.models import build_model
from bflc_amd.ops import functional as O


class TestBatchNormOracle:
    def test_fwd_bwd_matches_torch(self):
        torch.manual_seed(0)
        x = torch.randn(4, 7, 7, 8, requires_grad=True)  # NHWC
        g = torch.randn(8, requires_grad=True)
        b = torch.randn(8, requires_grad=True)
        y = O.batchnorm2d(x, g, b)
        ref = F.batch_norm(x.permute(0, 3, 1, 2), None, None, g, b,
                           training=True, eps=1e-5).permute(0, 2, 3, 1)
        assert torch.allclose(y, ref, atol=1e-5)
        dy = torch.randn_like(y)
        (y * dy).sum().backward()
        x2 = x.detach().clone().requires_grad_(True)
        g2 = g.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        (F.batch_norm(x2.permute(0, 3, 1, 2), None, None, g2, b2,
                      training=True).permute(0, 2, 3, 1) * dy).sum() \
            .backward()
        assert torch.allclose(x.grad, x2.grad, atol=1e-4)
        assert torch.allclose(g.grad, g2.grad, atol=1e-4)
        assert torch.allclose(b.grad, b2.grad, atol=1e-4)

    def test_fused_bn_relu_matches_unfused(self):
        torch.manual_seed(2)
        x = torch.randn(4, 7, 7, 8, requires_grad=True)
        g = (torch.rand(8) + 0.5).requires_grad_(True)
        b = torch.randn(8, requires_grad=True)
        y = O.batchnorm2d(x, g, b, relu=True)
        dy = torch.randn_like(y)
        (y * dy).sum().backward()
        x2 = x.detach().clone().requires_grad_(True)
        g2 = g.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        y2 = torch.relu(O.batchnorm2d(x2, g2, b2))
        assert torch.allclose(y, y2)
        (y2 * dy).sum().backward()
        assert torch.allclose(x.grad, x2.grad, atol=1e-5)
        assert torch.allclose(g.grad, g2.grad, atol=1e-5)
        assert torch.allclose(b.grad, b2.grad, atol=1e-5)

    def test_fused_relu_epilogue_matches_unfused(self):
        torch.manual_seed(5)
        x = torch.randn(2, 7, 7, 3, requires_grad=True)
        w = torch.randn(8, 3, 3, 3, requires_grad=True)
        b = torch.randn(8, requires_grad=True)
        y = O.conv2d(x, w, b, 1, 1, relu=True)
        dy = torch.randn_like(y)
        (y * dy).sum().backward()
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        y2 = O.relu(O.conv2d(x2, w2, b2, 1, 1))
        assert torch.allclose(y, y2)
        (y2 * dy).sum().backward()
        assert torch.allclose(x.grad, x2.grad, atol=1e-5)
        assert torch.allclose(w.grad, w2.grad, atol=1e-5)
        assert torch.allclose(b.grad, b2.grad, atol=1e-5)

        a1 = torch.randn(16, 6, requires_grad=True)
        w1 = torch.randn(6, 4, requires_grad=True)
        b1 = torch.randn(4, requires_grad=True)
        z = O.linear(a1, w1, b1, relu=True)
        assert torch.allclose(z, torch.relu(a1 @ w1 + b1), atol=1e-5)
        z.sum().backward()
        mask = (a1 @ w1 + b1 > 0).float()
        assert torch.allclose(w1.grad, a1.t() @ mask, atol=1e-5)

    def test_gap_and_add_relu(self):
        x = torch.randn(2, 5, 5, 4, requires_grad=True)  # NHWC
        y = O.global_avgpool(x)
        assert torch.allclose(y, x.mean(dim=(1, 2)))
        y.sum().backward()
        assert torch.allclose(x.grad, torch.full_like(x, 1 / 25.0))

        a = torch.randn(100, requires_grad=True)
        b = torch.randn(100, requires_grad=True)
        z = O.add_relu(a, b)
        assert torch.allclose(z, torch.relu(a + b))
        z.sum().backward()
        mask = (a + b > 0).float()
        assert torch.allclose(a.grad, mask)
        assert torch.allclose(b.grad, mask)


class TestResNet20:
    def test_forward_backward(self):
        cfg = FLConfig(model="resnet20", n_class=10, client_num=1,
                       comm_count=1, needed_update_count=1,
                       aggregate_count=1)
        m = build_model(cfg, torch.device("cpu"))
        x = torch.randn(4, 32, 32, 3)
        y = torch.randint(0, 10, (4,))
        logits = m.forward(x)
        assert logits.shape == (4, 10)
        loss = m.loss(x, y)
        loss.backward()
        assert float(m.grad_flat().abs().sum()) > 0
        # param count sanity: resnet20 ~= 0.27M params
        assert 0.25e6 < m.numel < 0.30e6

    def test_loss_decreases_with_sgd(self):
        cfg = FLConfig(model="resnet20", n_class=4, client_num=1,
                       comm_count=1, needed_update_count=1,
                       aggregate_count=1, learning_rate=0.05)
        m = build_model(cfg, torch.device("cpu"))
        torch.manual_seed(0)
        x = torch.randn(16, 32, 32, 3)
        y = torch.randint(0, 4, (16,))
        losses = []
        for _ in range(6):
            m.zero_grad()
            loss = m.loss(x, y)
            loss.backward()
            m.sgd_step(0.05)
            losses.append(float(loss.detach()))
        assert losses[-1] < losses[0]


class TestResNet50:
    def test_forward_backward_small_input(self):
        cfg = FLConfig(model="resnet50", n_class=10, client_num=1,
                       comm_count=1, needed_update_count=1,
                       aggregate_count=1)
        m = build_model(cfg, torch.device("cpu"))
        # 64x64 input keeps the CPU test fast; the conv/pool stack and
        # global avgpool handle any spatial size
        x = torch.randn(2, 64, 64, 3)
        y = torch.randint(0, 10, (2,))
        loss = m.loss(x, y)
        loss.backward()
        assert float(m.grad_flat().abs().sum()) > 0
        # ~25.5M params at 1000 classes; ~23.5M at 10
        assert 20e6 < m.numel < 27e6


class TestMixedPrecisionCPU:
    """CPU oracle of the bf16-shadow mixed-precision path (the GPU path
    uses the fused sgd_master_/adam_master_ kernels; on CPU base.py
    implements the same math with torch ops)."""

    def _model(self, optimizer="sgd"):
        cfg = FLConfig(model="mlp", n_features=16, n_class=4, client_num=1,
                       comm_count=1, needed_update_count=1,
                       aggregate_count=1, optimizer=optimizer)
        return cfg, build_model(cfg, torch.device("cpu"),
                                compute_dtype=torch.bfloat16)

    def test_master_stays_fp32_and_learns(self):
        cfg, m = self._model()
        assert m.flat.dtype == torch.float32
        assert m.cflat.dtype == torch.bfloat16
        torch.manual_seed(0)
        x = torch.randn(64, 16)
        y = torch.randint(0, 4, (64,))
        losses = []
        for _ in range(30):
            m.zero_grad()
            loss = m.loss(x, y)
            loss.backward()
            m.sgd_step(0.1)
            losses.append(float(loss.detach()))
        assert losses[-1] < 0.75 * losses[0]
        # shadow tracks master (equal after bf16 rounding)
        assert torch.equal(m.cflat.data, m.flat.to(torch.bfloat16))

    def test_adam_master_step(self):
        cfg, m = self._model("adam")
        mom = torch.zeros(m.numel)
        vel = torch.zeros(m.numel)
        torch.manual_seed(0)
        x = torch.randn(64, 16)
        y = torch.randint(0, 4, (64,))
        losses = []
        for step in range(1, 21):
            m.zero_grad()
            loss = m.loss(x, y)
            loss.backward()
            m.adam_step(mom, vel, step, 0.01)
            losses.append(float(loss.detach()))
        assert losses[-1] < losses[0]
        assert torch.equal(m.cflat.data, m.flat.to(torch.bfloat16))
