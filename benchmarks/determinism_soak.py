#!/usr/bin/env python3
"""Long-horizon determinism audit: two independent engine instances
step the SAME protocol side by side for N rounds; every round must
produce bitwise-identical global models and identical committee
decisions, with a checkpoint/restore of one instance midway (the
restore must not perturb the trajectory).

This is the long-soak form of tests/test_engine_cpu.py::
test_two_runs_bitwise_identical — run it when touching anything on the
replicated-state path (ledger, FedAvg order, delta math):

    python benchmarks/determinism_soak.py --rounds 2000
"""
import argparse
import os
import sys
import tempfile

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bflc_amd.comm import Transport          # noqa: E402
from bflc_amd.config import FLConfig         # noqa: E402
from bflc_amd.data import make_federated     # noqa: E402
from bflc_amd.fl import FLEngine             # noqa: E402


def build(cfg):
    shards, test = make_federated(cfg)
    return FLEngine(cfg, Transport(), shards, test)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=2000)
    ap.add_argument("--model", default="mlp",
                    choices=["logreg", "mlp", "femnist_cnn"])
    ap.add_argument("--optimizer", default="sgd", choices=["sgd", "adam"])
    ap.add_argument("--restore-at", type=int, default=None,
                    help="checkpoint/restore engine B at this round "
                         "(default rounds//2)")
    args = ap.parse_args()
    restore_at = args.restore_at or args.rounds // 2

    cfg = FLConfig.for_world(8, model=args.model, n_features=32,
                             n_class=8, samples_per_client=128,
                             batch_size=64, eval_samples=128,
                             partition="dirichlet",
                             optimizer=args.optimizer,
                             max_epoch=args.rounds + 10)
    a, b = build(cfg), build(cfg)
    ck = os.path.join(tempfile.gettempdir(), "det_soak_ck.pt")
    for r in range(args.rounds):
        sa = a.run_round()
        sb = b.run_round()
        assert torch.equal(a.global_flat, b.global_flat), \
            f"round {r}: global models diverged"
        assert [o for o, _ in a.last_decision.selected] == \
               [o for o, _ in b.last_decision.selected], \
            f"round {r}: decisions diverged"
        assert sa.n_updates == sb.n_updates
        if r == restore_at:  # restore must not perturb the trajectory
            b.save(ck)
            b = build(cfg)
            b.load(ck)
        if (r + 1) % 500 == 0:
            print(f"  {r + 1}/{args.rounds} rounds bitwise-identical "
                  f"(epoch {a.ledger.epoch})", flush=True)
    acc = a.evaluate_global()
    print(f"DETERMINISM SOAK PASS: {args.rounds} rounds, two replicas "
          f"bitwise-identical throughout (restore at {restore_at}); "
          f"final acc {acc:.4f}")


if __name__ == "__main__":
    main()
