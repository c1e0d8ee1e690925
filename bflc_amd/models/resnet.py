"""ResNet-20 (CIFAR, BASELINE config 3) and ResNet-50 (ImageNet-shape,
BASELINE config 5) built from the gfx950 op set: NHWC MFMA conv (1x1 =
pure GEMM), batch-stats BatchNorm, fused add+relu, global avgpool,
linear head. All activations are channels-last [N, H, W, C].

Flat-parameter models (models/base.py): conv weights, BN gamma/beta and
the fc head all live in one fp32 master vector, so FL deltas/FedAvg stay
single-AXPY/single-reduce operations even at ResNet-50 size (~25.5 M
params -> ~102 MB fp32 deltas per update; 288 GB HBM holds hundreds).
"""
from __future__ import annotations

from typing import List, Tuple

import torch

from bflc_amd.models.base import FlatModel
from bflc_amd.ops import functional as O


class ResNet(FlatModel):
    """cfg.model selects the variant:
    - resnet20: CIFAR-style — 3x3 stem (16ch), 3 stages x 3 basic blocks
      (16/32/64), stride 2 between stages, global avgpool, fc.
    - resnet50: ImageNet-style — 7x7/2 stem (64ch) + 3x3/2 maxpool,
      bottleneck stages [3,4,6,3] x (64/128/256/512, expansion 4).
    """

    def __init__(self, cfg, device, compute_dtype=None):
        self.variant = cfg.model
        if self.variant == "resnet20":
            self.stem_ch, self.bottleneck = 16, False
            self.stage_blocks = [3, 3, 3]
            self.stage_ch = [16, 32, 64]
            self.expansion = 1
        elif self.variant == "resnet50":
            self.stem_ch, self.bottleneck = 64, True
            self.stage_blocks = [3, 4, 6, 3]
            self.stage_ch = [64, 128, 256, 512]
            self.expansion = 4
        else:
            raise ValueError(self.variant)
        super().__init__(cfg, device, compute_dtype)

    # ------------------------------------------------------------------
    def specs(self):
        c = self.cfg
        in_ch = 3
        s: List[Tuple[str, Tuple[int, ...], str]] = []

        def conv(name, cin, cout, k):  # NHWC: weights [Kout, R, S, C]
            s.append((f"{name}.w", (cout, k, k, cin), "kaiming"))

        def bn(name, ch):
            s.append((f"{name}.g", (ch,), "zeros"))  # filled with 1s below
            s.append((f"{name}.b", (ch,), "zeros"))

        conv("stem", in_ch, self.stem_ch, 7 if self.bottleneck else 3)
        bn("stem_bn", self.stem_ch)
        ch_in = self.stem_ch
        for si, (nb, ch) in enumerate(zip(self.stage_blocks, self.stage_ch)):
            out_ch = ch * self.expansion
            for bi in range(nb):
                p = f"s{si}b{bi}"
                if self.bottleneck:
                    conv(f"{p}.c1", ch_in, ch, 1); bn(f"{p}.bn1", ch)
                    conv(f"{p}.c2", ch, ch, 3); bn(f"{p}.bn2", ch)
                    conv(f"{p}.c3", ch, out_ch, 1); bn(f"{p}.bn3", out_ch)
                else:
                    conv(f"{p}.c1", ch_in, ch, 3); bn(f"{p}.bn1", ch)
                    conv(f"{p}.c2", ch, ch, 3); bn(f"{p}.bn2", ch)
                if bi == 0 and ch_in != out_ch:
                    conv(f"{p}.sc", ch_in, out_ch, 1)
                    bn(f"{p}.scbn", out_ch)
                ch_in = out_ch
        s.append(("fc.w", (ch_in, c.n_class), "xavier"))
        s.append(("fc.b", (c.n_class,), "zeros"))
        return s

    def init_params(self, seed: int) -> None:
        super().init_params(seed)
        # BN gammas start at 1
        with torch.no_grad():
            host = self.flat.detach().cpu()
            for name, shape, _ in self._specs:
                if name.endswith(".g"):
                    off, _ = self._offsets[name]
                    host[off:off + shape[0]].fill_(1.0)
            self.flat.copy_(host.to(self.device))

    # ------------------------------------------------------------------
    def _cbr(self, x, conv, bnname, stride=1, pad=1, relu=True):
        # fused conv+BN(+relu): epilogue stats, single norm pass
        return O.conv2d_bn(x, self.p(f"{conv}.w"), self.p(f"{bnname}.g"),
                           self.p(f"{bnname}.b"), stride, pad, relu=relu)

    def _shortcut(self, x, p, stride):
        if f"{p}.sc.w" in self._offsets:
            return O.conv2d_bn(x, self.p(f"{p}.sc.w"),
                               self.p(f"{p}.scbn.g"),
                               self.p(f"{p}.scbn.b"), stride, 0,
                               relu=False)
        return x

    def _basic_block(self, x, p, stride):
        h = self._cbr(x, f"{p}.c1", f"{p}.bn1", stride, 1, relu=True)
        sc = self._shortcut(x, p, stride)
        # final conv+BN fuses the residual add + relu as well
        return O.conv2d_bn(h, self.p(f"{p}.c2.w"), self.p(f"{p}.bn2.g"),
                           self.p(f"{p}.bn2.b"), 1, 1, relu=True,
                           residual=sc)

    def _bottleneck_block(self, x, p, stride):
        h = self._cbr(x, f"{p}.c1", f"{p}.bn1", 1, 0, relu=True)
        h = self._cbr(h, f"{p}.c2", f"{p}.bn2", stride, 1, relu=True)
        sc = self._shortcut(x, p, stride)
        return O.conv2d_bn(h, self.p(f"{p}.c3.w"), self.p(f"{p}.bn3.g"),
                           self.p(f"{p}.bn3.b"), 1, 0, relu=True,
                           residual=sc)

    def forward(self, x):
        if self.bottleneck:
            h = self._cbr(x, "stem", "stem_bn", 2, 3, relu=True)
            h = O.maxpool2d(h, 3, 2)
        else:
            h = self._cbr(x, "stem", "stem_bn", 1, 1, relu=True)
        for si, nb in enumerate(self.stage_blocks):
            for bi in range(nb):
                stride = 2 if (bi == 0 and si > 0) else 1
                p = f"s{si}b{bi}"
                if self.bottleneck:
                    h = self._bottleneck_block(h, p, stride)
                else:
                    h = self._basic_block(h, p, stride)
        h = O.global_avgpool(h)
        return O.linear(h, self.p("fc.w"), self.p("fc.b"))
