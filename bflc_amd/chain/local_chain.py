"""LocalChain: the on-chain coordinator's full behavior, in-process.

This is the drop-in replacement for what FISCO-BCOS provided the
reference: a totally-ordered, admission-guarded record store + a
deterministic aggregator behind the 6-function contract ABI
(CommitteePrecompiled.cpp:132-311). The C++ CommitteeLedger supplies the
state machine; this class adds (a) the JSON wire formats, (b) the
aggregation *math* for the JSON model path (exact fp32 arithmetic like
the contract's float loops, .cpp:373-414), (c) thread-safe serialization
of concurrent client calls (the reference got ordering from PBFT block
order; here a mutex provides the total order for in-process clients),
and (d) an append-only JSONL record log = the persistence/checkpoint the
chain's replicated table provided.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Any, Optional, Tuple

import numpy as np

from bflc_amd._ledger import Admit, CommitteeLedger
from bflc_amd.chain import records
from bflc_amd.config import FLConfig


class LocalChain:
    def __init__(self, cfg: FLConfig, log_path: Optional[str] = None,
                 keys=None) -> None:
        """`keys`: optional chain.identity.KeyTable. When set, update and
        score transactions must carry a valid HMAC tag for their origin
        (the reference chain verifies each tx's ECDSA signature against
        the sender's account, README.md:283-299); unsigned/forged
        submissions are rejected."""
        self.cfg = cfg
        self.keys = keys
        self.ledger = CommitteeLedger(cfg.ledger_config())
        self.ledger.set_global_model(
            records.zero_model(cfg.n_features, cfg.n_class).encode())
        self._lock = threading.RLock()
        self._log_path = log_path
        self._log_f = open(log_path, "a") if log_path else None
        self._event_seq = 0

    # ------------------------------------------------------------------
    def _log(self, kind: str, **fields: Any) -> None:
        if self._log_f is None:
            return
        rec = {"seq": self._event_seq, "t": time.time(), "kind": kind}
        rec.update(fields)
        self._event_seq += 1
        self._log_f.write(json.dumps(rec) + "\n")
        self._log_f.flush()

    # --- the 6 ABI functions ------------------------------------------
    def register_node(self, origin: str) -> None:
        with self._lock:
            if self.ledger.register_node(origin):
                self._log("register", origin=origin, epoch=self.ledger.epoch)

    def query_state(self, origin: str) -> Tuple[str, int]:
        with self._lock:
            return self.ledger.query_state(origin)

    def query_global_model(self) -> Tuple[str, int]:
        with self._lock:
            blob, epoch = self.ledger.query_global_model()
            return blob.decode(), epoch

    def _verify(self, kind: str, origin: str, epoch: int, payload: str,
                tag) -> bool:
        if self.keys is None:
            return True
        return self.keys.verify(kind, origin, int(epoch), payload.encode(),
                                tag)

    def upload_local_update(self, origin: str, update: str,
                            epoch: int, tag: Optional[bytes] = None) -> bool:
        up = records.parse_update(update)
        meta = up["meta"]
        if not self._verify("update", origin, epoch, update, tag):
            self._log("update_rejected", origin=origin, epoch=epoch,
                      code=str(Admit.BAD_SIGNATURE))
            return False
        with self._lock:
            code = self.ledger.upload_local_update(
                origin, update.encode(), int(epoch),
                int(meta["n_samples"]), float(meta["avg_cost"]))
            if code == Admit.ACCEPTED:
                # "the update of local model is collected" (.cpp:255-257)
                self._log("update_accepted", origin=origin, epoch=epoch,
                          n_samples=meta["n_samples"])
                return True
            self._log("update_rejected", origin=origin, epoch=epoch,
                      code=str(code))
            return False

    def query_all_updates(self) -> str:
        """Returns "" until the quota is reached, then the dict of
        trainer -> update-json (nested strings, like the chain state)."""
        with self._lock:
            ups = self.ledger.query_all_updates()
            if not ups:
                return ""
            return json.dumps({k: v.decode() for k, v in ups})

    def upload_scores(self, origin: str, epoch: int, scores: str,
                      tag: Optional[bytes] = None) -> bool:
        if not self._verify("scores", origin, epoch, scores, tag):
            self._log("scores_rejected", origin=origin, epoch=epoch,
                      code=str(Admit.BAD_SIGNATURE))
            return False
        smap = {k: float(v) for k, v in json.loads(scores).items()}
        with self._lock:
            dec = self.ledger.upload_scores(origin, int(epoch), smap)
            self._log("scores", origin=origin, epoch=epoch,
                      count=self.ledger.score_count)
            if dec is not None:
                self._aggregate(dec)
            return True

    # ------------------------------------------------------------------
    def _aggregate(self, dec) -> None:
        """Weighted FedAvg over the selected updates + global update —
        the contract's Aggregate math (.cpp:373-414) in fp32."""
        cfg = self.cfg
        W, b = records.parse_model(self.query_global_model_unlocked())
        W = np.asarray(W, dtype=np.float32)
        b = np.asarray(b, dtype=np.float32)
        accW = np.zeros_like(W)
        accB = np.zeros_like(b)
        total_n = np.float32(0)
        for origin, n in dec.selected:  # fixed decision order
            up = records.parse_update(
                self.ledger.update_blob(origin).decode())
            dW = np.asarray(up["delta_model"]["ser_W"], dtype=np.float32)
            dB = np.asarray(up["delta_model"]["ser_b"], dtype=np.float32)
            accW += dW * np.float32(n)
            accB += dB * np.float32(n)
            total_n += np.float32(n)
        accW /= total_n
        accB /= total_n
        lr = np.float32(cfg.learning_rate)
        W -= lr * accW
        b -= lr * accB
        new_blob = records.model_record(W.tolist(), b.tolist())
        self.ledger.commit_aggregate(new_blob.encode())
        # "the E epoch , global loss : L" (.cpp:422-425)
        self._log("aggregate", epoch=dec.epoch,
                  selected=[o for o, _ in dec.selected],
                  global_loss=dec.avg_cost,
                  next_committee=list(dec.next_committee))

    def query_global_model_unlocked(self) -> str:
        blob, _ = self.ledger.query_global_model()
        return blob.decode()

    # --- checkpoint/resume --------------------------------------------
    def save(self, path: str) -> None:
        with self._lock, open(path, "w") as f:
            snap = self.ledger.snapshot()
            snap = _bytes_to_str(snap)
            json.dump({"config": self.cfg.to_dict(), "ledger": snap}, f)

    def load(self, path: str) -> None:
        with self._lock, open(path) as f:
            d = json.load(f)
        self.ledger.restore(_str_to_bytes(d["ledger"]))

    def close(self) -> None:
        if self._log_f:
            self._log_f.close()
            self._log_f = None


def _bytes_to_str(obj):
    if isinstance(obj, bytes):
        return obj.decode()
    if isinstance(obj, dict):
        return {k: _bytes_to_str(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_bytes_to_str(v) for v in obj]
    return obj


def _str_to_bytes(obj, keys=("global_model", "blob")):
    if isinstance(obj, dict):
        return {k: (v.encode() if isinstance(v, str) and k in keys
                    else _str_to_bytes(v, keys)) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_str_to_bytes(v, keys) for v in obj]
    return obj
