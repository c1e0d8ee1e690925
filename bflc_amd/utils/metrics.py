"""Structured observability: per-round JSONL metrics + phase timers.

Replaces the reference's print-based observability (sponsor line
main.py:327-328, contract clog counters CommitteePrecompiled.cpp:240-
293,422-425) with structured records suitable for the 1/2/4/8-GPU
scaling curve.
"""
from __future__ import annotations

import json
import time
from dataclasses import asdict, is_dataclass
from typing import Any, Dict, IO, Optional


class JsonlLogger:
    def __init__(self, path: Optional[str], rank: int = 0,
                 only_rank0: bool = True) -> None:
        self._f: Optional[IO] = None
        self.rank = rank
        if path and (rank == 0 or not only_rank0):
            self._f = open(path, "a")

    def log(self, kind: str, payload: Any = None, **fields: Any) -> None:
        if self._f is None:
            return
        rec: Dict[str, Any] = {"t": time.time(), "kind": kind,
                               "rank": self.rank}
        if payload is not None:
            rec.update(asdict(payload) if is_dataclass(payload) else payload)
        rec.update(fields)
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()

    def close(self) -> None:
        if self._f:
            self._f.close()
            self._f = None
