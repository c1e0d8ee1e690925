"""Transport: torch.distributed over RCCL (xGMI) on GPU, gloo on CPU.

Replaces the reference's FISCO-BCOS Channel/PBFT substrate
(reference README.md:145-183, python-sdk/main.py:13-17): the total order
the chain provided is obtained here from rank-ordered all-gathers — every
rank receives the same submissions in the same (rank, slot) order and
feeds them to its own deterministic ledger replica.

xGMI note: MI355X links are point-to-point (7 links x ~153 GB/s per
GPU); RCCL's all-gather uses them pairwise, so gathering one large flat
delta tensor per rank per round (instead of many small messages) is the
right shape for this fabric.
"""
from __future__ import annotations

import datetime
import os
import pickle
from typing import Any, List, Optional

import torch
import torch.distributed as dist


class Transport:
    """Thin wrapper over torch.distributed with a world_size==1 fast path."""

    def __init__(self, backend: Optional[str] = None,
                 device: Optional[torch.device] = None,
                 timeout_s: Optional[float] = None) -> None:
        self.rank = int(os.environ.get("RANK", "0"))
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self._initialized_here = False
        # per-collective timeout: a dead/slow peer turns an infinite hang
        # into a raised error the engine converts to a clean abort with a
        # checkpoint (the reference tolerated crashed trainers by 10-of-16
        # over-provisioning, CommitteePrecompiled.h:15 — a barrier-driven
        # engine needs an explicit failure path instead)
        if timeout_s is None:
            timeout_s = float(os.environ.get("BFLC_COLL_TIMEOUT_S", "300"))
        self.timeout_s = timeout_s

        if device is not None:
            self.device = device
        elif torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", self.rank))
            self.device = torch.device("cuda", local % torch.cuda.device_count())
            torch.cuda.set_device(self.device)
        else:
            self.device = torch.device("cpu")

        if self.world_size > 1 and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if self.device.type == "cuda" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29531")
            dist.init_process_group(
                backend=backend,
                rank=self.rank,
                world_size=self.world_size,
                timeout=datetime.timedelta(seconds=self.timeout_s),
            )
            self._initialized_here = True
        self.backend = (dist.get_backend() if dist.is_initialized()
                        else (backend or "local"))

    # ------------------------------------------------------------------
    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1 and dist.is_initialized()

    def barrier(self) -> None:
        if self.is_distributed:
            if self.device.type == "cuda":
                dist.barrier(device_ids=[self.device.index])
            else:
                dist.barrier()

    def close(self) -> None:
        if self._initialized_here and dist.is_initialized():
            dist.destroy_process_group()

    # ------------------------------------------------------------------
    def all_gather_blobs(self, blob: bytes) -> List[bytes]:
        """Variable-length byte-blob all-gather (two-phase: sizes, then
        max-padded payload). Returns blobs in rank order on every rank —
        this IS the total order the chain used to provide."""
        if not self.is_distributed:
            return [blob]
        dev = self.device if self.backend == "nccl" else torch.device("cpu")
        n = torch.tensor([len(blob)], dtype=torch.int64, device=dev)
        sizes = [torch.zeros(1, dtype=torch.int64, device=dev)
                 for _ in range(self.world_size)]
        dist.all_gather(sizes, n)
        sizes = [int(s.item()) for s in sizes]
        maxlen = max(max(sizes), 1)
        payload = torch.zeros(maxlen, dtype=torch.uint8, device=dev)
        if blob:
            payload[: len(blob)] = torch.frombuffer(
                bytearray(blob), dtype=torch.uint8).to(dev)
        outs = [torch.zeros(maxlen, dtype=torch.uint8, device=dev)
                for _ in range(self.world_size)]
        dist.all_gather(outs, payload)
        return [bytes(outs[r][: sizes[r]].cpu().numpy().tobytes())
                for r in range(self.world_size)]

    def all_gather_objects(self, obj: Any) -> List[Any]:
        """Pickle-based object all-gather in rank order (control plane)."""
        return [pickle.loads(b) for b in self.all_gather_blobs(
            pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL))]

    def all_gather_tensor(self, t: torch.Tensor) -> List[torch.Tensor]:
        """All-gather equal-shaped tensors (data plane: flat delta stacks).
        The input must have the same shape on every rank; returns the
        rank-ordered list. On GPU this is one RCCL all-gather over xGMI."""
        if not self.is_distributed:
            return [t]
        t = t.contiguous()
        if self.backend == "nccl" and not t.is_cuda:
            t = t.to(self.device, non_blocking=True)
        if self.backend == "gloo" and t.is_cuda:
            t = t.cpu()
        outs = [torch.empty_like(t) for _ in range(self.world_size)]
        dist.all_gather(outs, t)
        return outs

    def broadcast_tensor(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if not self.is_distributed:
            return t
        if self.backend == "gloo" and t.is_cuda:
            cpu = t.cpu()
            dist.broadcast(cpu, src=src)
            return cpu.to(t.device)
        dist.broadcast(t, src=src)
        return t
