#!/usr/bin/env python3
"""World-partitioning invariance audit: the committee protocol's
trajectory must not depend on HOW the 8 clients are split over ranks —
admission order is global client order and every reduction is
fixed-order, so the final global model must be BITWISE identical at
world size 1, 2, 4 (and 8). Rank 0 prints a sha256 of the final flat
model; compare across launches:

    python benchmarks/world_invariance.py --rounds 100
    torchrun --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 \
        benchmarks/world_invariance.py --rounds 100
"""
import argparse
import hashlib
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bflc_amd.comm import Transport          # noqa: E402
from bflc_amd.config import FLConfig         # noqa: E402
from bflc_amd.data import make_federated     # noqa: E402
from bflc_amd.fl import FLEngine             # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=100)
    ap.add_argument("--model", default="mlp")
    ap.add_argument("--byzantine", type=int, default=0)
    args = ap.parse_args()
    cfg = FLConfig.for_world(8, model=args.model, n_features=32,
                             n_class=8, samples_per_client=128,
                             batch_size=64, eval_samples=128,
                             partition="dirichlet",
                             byzantine_clients=args.byzantine,
                             max_epoch=args.rounds + 10)
    t = Transport()
    shards, test = make_federated(cfg)
    eng = FLEngine(cfg, t, shards, test)
    eng.run(args.rounds)
    h = hashlib.sha256(
        eng.global_flat.cpu().numpy().tobytes()).hexdigest()
    if t.rank == 0:
        sel = sorted({o for o, _ in eng.last_decision.selected})
        print(f"world={t.world_size} rounds={args.rounds} "
              f"byz={args.byzantine} model_sha256={h} "
              f"acc={eng.evaluate_global():.4f} last_selected={sel}",
              flush=True)
    t.close()


if __name__ == "__main__":
    main()
