"""Single source of truth for protocol + training configuration.

The reference scatters its constants across a C++ header
(reference CommitteePrecompiled.h:6-19: n_features=5, n_class=2,
COMM_COUNT=4, AGGREGATE_COUNT=6, NEEDED_UPDATE_COUNT=10, CLIENT_NUM=20,
learning_rate=0.001) and module-level Python constants
(reference python-sdk/main.py:52-69,87-88) with the learning rate
duplicated in both languages.  Here there is exactly one config object,
shared by the ledger, the engine, and the kernels, loadable from JSON
and overridable from the CLI.
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass
from typing import Any, Dict


@dataclass
class FLConfig:
    """Committee-consensus FL protocol + run configuration."""

    # --- protocol constants (reference CommitteePrecompiled.h:6-19) ---
    client_num: int = 20
    comm_count: int = 4
    needed_update_count: int = 10
    aggregate_count: int = 6
    learning_rate: float = 1e-3
    max_epoch: int = 1000          # reference main.py:65 (50 * CLIENT_NUM)

    # --- model / data ---
    model: str = "logreg"          # logreg | mlp | femnist_cnn | resnet20 | resnet50
    n_features: int = 5            # logreg/mlp input dim (reference main.py:68)
    n_class: int = 2
    batch_size: int = 100          # reference main.py:87
    local_epochs: int = 1          # passes over the local shard per round
    samples_per_client: int = 305  # synthetic shard size (reference ~8143*0.75/20)
    eval_samples: int = 2036       # held-out sponsor set (reference 8143*0.25)
    seed: int = 42
    dtype: str = "bf16"            # compute dtype on GPU; fp32 on CPU tests
    optimizer: str = "sgd"         # sgd | adam (reference main.py:126-127)

    # --- sharding ---
    partition: str = "iid"         # iid | dirichlet (non-IID)
    dirichlet_alpha: float = 0.3
    byzantine_clients: int = 0     # label-flip attackers (BASELINE config 4)

    # --- execution ---
    use_graphs: bool = True        # hipGraph-captured train step (GPU+SGD;
                                   # eager fallback elsewhere, fl/graphs.py)

    def __post_init__(self) -> None:
        if self.comm_count < 1 or self.client_num < 1:
            raise ValueError("comm_count and client_num must be >= 1")
        if self.aggregate_count > self.needed_update_count:
            raise ValueError("aggregate_count > needed_update_count")
        if self.comm_count >= self.client_num and self.client_num > 1:
            raise ValueError("comm_count must leave at least one trainer")
        if self.client_num > 1 and self.needed_update_count < self.comm_count:
            # rotation draws the next committee from scored trainers
            # (reference .cpp:443-455); fewer scored trainers than
            # comm_count would shrink the committee and deadlock scoring
            raise ValueError("needed_update_count must be >= comm_count")
        if self.client_num > 1 and \
                self.needed_update_count > self.client_num - self.comm_count:
            raise ValueError("needed_update_count exceeds trainer count")

    # ------------------------------------------------------------------
    @classmethod
    def reference_defaults(cls) -> "FLConfig":
        """The exact configuration of the reference demo."""
        return cls()

    @classmethod
    def for_world(cls, n_nodes: int, **overrides: Any) -> "FLConfig":
        """Scale the protocol sanely to n_nodes FL clients (1/2/4/8 GPUs).

        Keeps the committee/quota structure of the reference while making
        every world size well-formed:
          - committee = min(4, floor(n/2)) but at least 1 when n >= 2;
          - n == 1 degenerates to plain local SGD with self-scoring
            (committee == trainer on alternating epochs is meaningless at
            n == 1, so the single client both trains and scores).
        """
        n = int(n_nodes)
        if n < 1:
            raise ValueError("n_nodes >= 1 required")
        if n == 1:
            cfg = dict(client_num=1, comm_count=1, needed_update_count=1,
                       aggregate_count=1, self_score=True)
        else:
            comm = max(1, min(4, n // 2))
            trainers = n - comm
            # barrier-driven: every trainer submits each round; the quota
            # must be >= comm_count so rotation can always fill the
            # committee from scored trainers.
            needed = max(trainers, 1)
            agg = max(1, min(needed, -(-needed * 6 // 10)))  # ceil(0.6*needed)
            cfg = dict(client_num=n, comm_count=comm,
                       needed_update_count=needed, aggregate_count=agg)
        cfg.pop("self_score", None)
        cfg.update(overrides)
        return cls(**cfg)

    # n==1 special case: the lone node may both train and score.
    @property
    def self_scoring(self) -> bool:
        return self.client_num == 1

    # ------------------------------------------------------------------
    def to_dict(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), sort_keys=True)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "FLConfig":
        names = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in names})

    @classmethod
    def from_json_file(cls, path: str) -> "FLConfig":
        with open(path) as f:
            return cls.from_dict(json.load(f))

    def ledger_config(self):
        from bflc_amd._ledger import LedgerConfig  # lazy: built ext
        lc = LedgerConfig()
        lc.client_num = self.client_num
        lc.comm_count = self.comm_count
        lc.needed_update_count = self.needed_update_count
        lc.aggregate_count = self.aggregate_count
        lc.learning_rate = self.learning_rate
        lc.max_epoch = self.max_epoch
        return lc
