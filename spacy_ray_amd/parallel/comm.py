"""Comm backends: the collective layer under the ZeRO-1 engine.

Replaces the reference's Ray-actor RPC transport wholesale (SURVEY.md §2.4
C1-C7 mapping): gradient pushes become bucketed reduce-scatter, parameter
broadcasts become all-gather, the Evaluator actor becomes a rank-0 object
broadcast.  Three implementations of one interface:

  * DistComm  — torch.distributed; backend "nccl" IS RCCL on ROCm (xGMI
    inside a node).  gloo works for CPU multi-process tests (BASELINE
    config #1) — reduce_scatter/all_gather_into_tensor are emulated there
    (gloo lacks the fused collectives).
  * LocalComm — world_size 1, no process group (also the FakeComm seam for
    single-process tests).
"""
from __future__ import annotations

import datetime
import os
from typing import Any, Optional

import torch
import torch.distributed as dist


class Comm:
    rank: int = 0
    world: int = 1

    def reduce_scatter_flat(self, flat: torch.Tensor, out_shard: torch.Tensor, async_op: bool = False):
        raise NotImplementedError

    def all_gather_flat(self, flat_out: torch.Tensor, shard: torch.Tensor, async_op: bool = False):
        raise NotImplementedError

    def all_reduce_(self, t: torch.Tensor):
        raise NotImplementedError

    def broadcast_obj(self, obj: Any, src: int = 0) -> Any:
        raise NotImplementedError

    def all_gather_obj(self, obj: Any) -> list:
        """Every rank's object, in rank order (sharded-eval merge)."""
        return [obj]

    def barrier(self) -> None:
        pass


class LocalComm(Comm):
    """world=1: reduce-scatter/all-gather collapse to slice copies."""

    def reduce_scatter_flat(self, flat, out_shard, async_op=False):
        out_shard.copy_(flat[: out_shard.numel()])
        return None

    def all_gather_flat(self, flat_out, shard, async_op=False):
        flat_out[: shard.numel()].copy_(shard)
        return None

    def all_reduce_(self, t):
        return None

    def broadcast_obj(self, obj, src: int = 0):
        return obj


class DistComm(Comm):
    def __init__(self, backend: Optional[str] = None, device: Optional[torch.device] = None,
                 timeout_s: int = 600):
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        self.backend = backend
        if not dist.is_initialized():
            dist.init_process_group(
                backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
            )
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = device
        self._fused = backend == "nccl"
        # SRX_COLLECTIVE_CHECK=1: record every collective's (op, numel) and
        # assert the SEQUENCE is identical across ranks at each barrier —
        # NCCL deadlocks are ordering bugs, and the one found in round 1
        # (rank-dependent warmup counts) was exactly a sequence divergence.
        self._oplog: Optional[list] = (
            [] if os.environ.get("SRX_COLLECTIVE_CHECK") == "1" else None
        )

    def _log(self, op: str, numel: int) -> None:
        if self._oplog is not None:
            self._oplog.append((op, int(numel)))

    def reduce_scatter_flat(self, flat, out_shard, async_op=False):
        self._log("rs", flat.numel())
        if self._fused:
            work = dist.reduce_scatter_tensor(out_shard, flat, op=dist.ReduceOp.AVG,
                                              async_op=async_op)
            return work
        # gloo emulation: allreduce then take own slice
        dist.all_reduce(flat, op=dist.ReduceOp.SUM)
        flat.div_(self.world)
        per = out_shard.numel()
        out_shard.copy_(flat[self.rank * per : (self.rank + 1) * per])
        return None

    def all_gather_flat(self, flat_out, shard, async_op=False):
        self._log("ag", flat_out.numel())
        if self._fused:
            return dist.all_gather_into_tensor(flat_out, shard, async_op=async_op)
        per = shard.numel()
        chunks = list(flat_out.split(per))
        dist.all_gather(chunks, shard.contiguous())
        return None

    def all_reduce_(self, t):
        self._log("ar", t.numel())
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return None

    def broadcast_obj(self, obj, src: int = 0):
        holder = [obj if self.rank == src else None]
        dist.broadcast_object_list(holder, src=src)
        return holder[0]

    def all_gather_obj(self, obj):
        out = [None] * self.world
        dist.all_gather_object(out, obj)
        return out

    def barrier(self) -> None:
        if self._oplog is not None:
            self._check_collective_order()
        dist.barrier()

    def _check_collective_order(self) -> None:
        """Assert every rank issued the same collective sequence since the
        last check (hash of the oplog, compared via all_gather_object)."""
        import hashlib

        h = hashlib.sha256(repr(self._oplog).encode()).hexdigest()
        payload = [None] * self.world
        dist.all_gather_object(payload, (h, len(self._oplog)))
        if any(p != payload[0] for p in payload):
            raise RuntimeError(
                f"collective-sequence divergence across ranks: {payload}; "
                f"rank {self.rank} issued {self._oplog[-8:]}"
            )
        self._oplog.clear()


def init_comm_from_env(device: Optional[torch.device] = None) -> Comm:
    """RANK/WORLD_SIZE env (torchrun contract) -> DistComm; else LocalComm."""
    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        return DistComm(device=device)
    return LocalComm()
