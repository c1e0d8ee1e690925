"""Synthetic datasets + client sharding (IID / non-IID / Byzantine).

Capability parity with the reference data pipeline
(reference python-sdk/main.py:32-53: UCI Occupancy CSV, 75/25 split,
one-hot labels, np.array_split into IID shards) — but generated
synthetically (no network for datasets, per BASELINE.json) and extended
with the non-IID Dirichlet partition and label-flip attackers that
BASELINE configs 2-4 require.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

import torch


@dataclass
class Shard:
    x: torch.Tensor          # [n, ...] features
    y: torch.Tensor          # [n] int64 class labels
    byzantine: bool = False  # label-flip attacker?

    @property
    def n(self) -> int:
        return self.x.shape[0]

    def to(self, device: torch.device, dtype: torch.dtype = torch.float32
           ) -> "Shard":
        return Shard(self.x.to(device=device, dtype=dtype),
                     self.y.to(device=device), self.byzantine)


def make_tabular(n: int, n_features: int, n_class: int,
                 gen: torch.Generator) -> Tuple[torch.Tensor, torch.Tensor]:
    """Occupancy-style separable tabular data: class-dependent Gaussian
    means so a linear model can reach high accuracy (the reference's
    logistic regression reaches 0.92 on the real CSV)."""
    means = torch.randn(n_class, n_features, generator=gen) * 2.0
    y = torch.randint(0, n_class, (n,), generator=gen)
    x = means[y] + torch.randn(n, n_features, generator=gen)
    return x, y


def make_images(n: int, channels: int, hw: int, n_class: int,
                gen: torch.Generator) -> Tuple[torch.Tensor, torch.Tensor]:
    """FEMNIST/CIFAR/ImageNet-shaped synthetic images (NHWC, the
    channels-last layout the gfx950 conv kernels use): class template +
    noise, so CNNs can learn and accuracy is a meaningful signal."""
    templates = torch.randn(n_class, hw, hw, channels, generator=gen)
    y = torch.randint(0, n_class, (n,), generator=gen)
    x = templates[y] * 0.5 + torch.randn(n, hw, hw, channels, generator=gen)
    return x, y


def _dataset_for(model: str, n: int, cfg, gen: torch.Generator):
    if model in ("logreg", "mlp"):
        return make_tabular(n, cfg.n_features, cfg.n_class, gen)
    if model == "femnist_cnn":
        return make_images(n, 1, 28, cfg.n_class, gen)
    if model == "resnet20":
        return make_images(n, 3, 32, cfg.n_class, gen)
    if model == "resnet50":
        return make_images(n, 3, 224, cfg.n_class, gen)
    raise ValueError(f"unknown model {model}")


def partition_iid(x: torch.Tensor, y: torch.Tensor, k: int,
                  gen: torch.Generator) -> List[Tuple[torch.Tensor, torch.Tensor]]:
    """np.array_split-style IID sharding (reference main.py:47-48)."""
    perm = torch.randperm(x.shape[0], generator=gen)
    xs = torch.tensor_split(x[perm], k)
    ys = torch.tensor_split(y[perm], k)
    return list(zip(xs, ys))


def partition_dirichlet(x: torch.Tensor, y: torch.Tensor, k: int,
                        alpha: float, n_class: int, gen: torch.Generator
                        ) -> List[Tuple[torch.Tensor, torch.Tensor]]:
    """Non-IID label-skew partition: per-class Dirichlet(alpha) over
    clients (the standard FedAvg non-IID benchmark protocol)."""
    idx_by_client: List[List[torch.Tensor]] = [[] for _ in range(k)]
    for c in range(n_class):
        idx = torch.nonzero(y == c, as_tuple=True)[0]
        idx = idx[torch.randperm(idx.numel(), generator=gen)]
        # sample proportions; torch has no Dirichlet w/ generator -> Gamma trick
        g = torch._standard_gamma(torch.full((k,), alpha), gen)
        p = g / g.sum().clamp_min(1e-9)
        # cumulative rounding: client i gets [round(cum[i-1]*n),
        # round(cum[i]*n)). Flooring each count and dumping the sum of
        # remainders on the last client concentrated ~one remainder
        # sample PER CLASS there — at 1000 classes x 2 samples/class
        # (ResNet-50 shards) client k-1 ended up with 2/3 of ALL data
        # (1375 of 2048) and its serial minibatch chain dominated every
        # protocol round.
        bounds = (torch.cumsum(p, 0) * idx.numel()).round().long()
        bounds[-1] = idx.numel()
        start = 0
        for i in range(k):
            end = int(bounds[i])
            idx_by_client[i].append(idx[start:end])
            start = end
    out = []
    for i in range(k):
        ids = torch.cat(idx_by_client[i]) if idx_by_client[i] else \
            torch.empty(0, dtype=torch.int64)
        if ids.numel() == 0:  # guarantee a non-empty shard
            ids = torch.randint(0, y.numel(), (1,), generator=gen)
        ids = ids[torch.randperm(ids.numel(), generator=gen)]
        out.append((x[ids], y[ids]))
    return out


def make_federated(cfg, seed_offset: int = 0
                   ) -> Tuple[List[Shard], Shard]:
    """Build (client shards, held-out test shard) for cfg.

    Byzantine clients (the last cfg.byzantine_clients) get their labels
    flipped label-flip-attack style: y -> (n_class - 1) - y (BASELINE
    config 4: committee scoring is the defense).
    """
    gen = torch.Generator().manual_seed(cfg.seed + seed_offset)
    total = cfg.samples_per_client * cfg.client_num
    x, y = _dataset_for(cfg.model, total + cfg.eval_samples, cfg, gen)
    x_test, y_test = x[total:], y[total:]
    x, y = x[:total], y[:total]

    if cfg.partition == "dirichlet":
        parts = partition_dirichlet(x, y, cfg.client_num,
                                    cfg.dirichlet_alpha, cfg.n_class, gen)
    else:
        parts = partition_iid(x, y, cfg.client_num, gen)

    shards = []
    for i, (sx, sy) in enumerate(parts):
        byz = i >= cfg.client_num - cfg.byzantine_clients
        if byz:
            sy = (cfg.n_class - 1) - sy
        shards.append(Shard(sx, sy, byzantine=byz))
    return shards, Shard(x_test, y_test)
