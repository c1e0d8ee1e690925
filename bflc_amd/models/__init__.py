"""Model zoo: every family the BASELINE configs name.

- logreg: the reference's 5x2 logistic regression (main.py:113-120)
- mlp: 2-layer MLP (BASELINE config 1, MNIST-shaped plumbing)
- femnist_cnn: 4-layer CNN, the headline benchmark model (config 2)
- resnet20 / resnet50: CIFAR / ImageNet-shape (configs 3 and 5)
"""
from __future__ import annotations

from typing import Tuple

import torch

from bflc_amd.models.base import FlatModel
from bflc_amd.ops import functional as O


class LogReg(FlatModel):
    """pred = x @ W + b — reference main.py:113-120, zero-init like the
    on-chain initial global model (CommitteePrecompiled.h:30-34)."""

    def specs(self):
        c = self.cfg
        return [("w", (c.n_features, c.n_class), "zeros"),
                ("b", (c.n_class,), "zeros")]

    def forward(self, x):
        return O.linear(x, self.p("w"), self.p("b"))


class MLP(FlatModel):
    """2-layer MLP (BASELINE config 1). Hidden width 128."""

    HIDDEN = 128

    def specs(self):
        c = self.cfg
        return [("w1", (c.n_features, self.HIDDEN), "kaiming"),
                ("b1", (self.HIDDEN,), "zeros"),
                ("w2", (self.HIDDEN, c.n_class), "xavier"),
                ("b2", (c.n_class,), "zeros")]

    def forward(self, x):
        h = O.linear(x, self.p("w1"), self.p("b1"), relu=True)
        return O.linear(h, self.p("w2"), self.p("b2"))


class FemnistCNN(FlatModel):
    """4-layer FEMNIST CNN (BASELINE config 2): conv3x3(1->32) - pool -
    conv3x3(32->64) - pool - fc(3136->128) - fc(128->n_class), on 28x28
    grayscale. The classic FedAvg/LEAF FEMNIST architecture."""

    def specs(self):
        c = self.cfg
        # NHWC: conv weights [Kout, R, S, C]; x is [N, 28, 28, 1]
        return [("c1w", (32, 3, 3, 1), "kaiming"), ("c1b", (32,), "zeros"),
                ("c2w", (64, 3, 3, 32), "kaiming"), ("c2b", (64,), "zeros"),
                ("f1w", (64 * 7 * 7, 128), "kaiming"), ("f1b", (128,), "zeros"),
                ("f2w", (128, c.n_class), "xavier"), ("f2b", (c.n_class,), "zeros")]

    def forward(self, x):
        h = O.conv2d(x, self.p("c1w"), self.p("c1b"), 1, 1, relu=True)
        h = O.maxpool2d(h, 2)
        h = O.conv2d(h, self.p("c2w"), self.p("c2b"), 1, 1, relu=True)
        h = O.maxpool2d(h, 2)
        h = h.reshape(h.shape[0], -1)
        h = O.linear(h, self.p("f1w"), self.p("f1b"), relu=True)
        return O.linear(h, self.p("f2w"), self.p("f2b"))


def build_model(cfg, device, compute_dtype=None) -> FlatModel:
    name = cfg.model
    if name == "logreg":
        m = LogReg(cfg, device, compute_dtype)
    elif name == "mlp":
        m = MLP(cfg, device, compute_dtype)
    elif name == "femnist_cnn":
        m = FemnistCNN(cfg, device, compute_dtype)
    elif name in ("resnet20", "resnet50"):
        from bflc_amd.models.resnet import ResNet
        m = ResNet(cfg, device, compute_dtype)
    else:
        raise ValueError(f"unknown model {name}")
    m.init_params(cfg.seed)
    return m


__all__ = ["FlatModel", "LogReg", "MLP", "FemnistCNN", "build_model"]
