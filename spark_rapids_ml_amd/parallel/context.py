"""Distributed runtime: one process per GPU, collectives over RCCL/xGMI.

This replaces the reference's Spark-barrier + NCCL-uid-bootstrap + RAFT-handle
injection machinery (reference common/cuml_context.py:75-156) with the
MI355X-idiomatic layout: the framework runs SPMD — every rank is a Python
process pinned to one GPU, rendezvous and collectives go through
torch.distributed whose "nccl" backend IS RCCL on ROCm, riding the node's
7-link xGMI mesh. On CPU-only machines (unit tests) the gloo backend provides
identical semantics.

Design notes vs the reference:
- reference bootstraps an NCCL uniqueId over Spark `BarrierTaskContext.allGather`
  (cuml_context.py:75-81); here torch.distributed's TCP store does the
  rendezvous (MASTER_ADDR/MASTER_PORT env, torchrun-compatible).
- reference injects NCCL+UCX into a RAFT handle consumed by cuML C++
  (cuml_context.py:116-156); here the handle equivalent is (torch device,
  HIP stream, process group) and p2p (kNN merge) is rcclSend/Recv via
  torch.distributed send/recv — no UCX needed intra-node.
- abort-on-exception semantics (cuml_context.py:162-167) map to
  destroy_process_group in shutdown_comm.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Any, List, Optional

import torch
import torch.distributed as dist

_COMM: Optional["Comm"] = None


class Comm:
    """Rank/world handle + collective helpers.

    world_size==1 with no env rendezvous works without initializing
    torch.distributed at all (single-process mode: every collective is a
    no-op/identity)."""

    def __init__(self, rank: int, world_size: int, device: torch.device, backend: str):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.backend = backend

    # -- collectives ------------------------------------------------------
    @property
    def initialized(self) -> bool:
        return dist.is_available() and dist.is_initialized()

    def barrier(self) -> None:
        if self.initialized:
            dist.barrier()

    def allreduce(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        """In-place all-reduce. The hot call: centroid sums (k×d), gradient
        (C×(d+1)), Gram partials (d×d) — MB-sized latency-sensitive messages
        over the xGMI mesh (SURVEY.md §5 comm backend)."""
        if self.initialized:
            ops = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX, "min": dist.ReduceOp.MIN}
            if t.is_contiguous():
                dist.all_reduce(t, op=ops[op])
            else:
                # collectives ship the raw buffer; a strided view would be
                # reduced element-order-mismatched across ranks
                ct = t.contiguous()
                dist.all_reduce(ct, op=ops[op])
                t.copy_(ct)
        return t

    def allreduce_t(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        """All-reduce returning the reduced tensor ON t's device. Needed
        because `allreduce(to_coll(t))` reduces a COPY when the collective
        device differs from t's (gloo collectives with GPU tensors) and the
        original would silently stay unreduced."""
        ct = self.to_coll(t)
        self.allreduce(ct, op)
        if ct is t:
            return t
        return ct.to(t.device)

    def allreduce_scalar(self, value: float, op: str = "sum") -> float:
        if not self.initialized:
            return float(value)
        t = torch.tensor([value], dtype=torch.float64, device=self._coll_device())
        self.allreduce(t, op)
        return float(t.item())

    def allgather(self, t: torch.Tensor) -> List[torch.Tensor]:
        if not self.initialized:
            return [t]
        out = [torch.empty_like(t) for _ in range(self.world_size)]
        dist.all_gather(out, t.contiguous())
        return out

    def allgather_obj(self, obj: Any) -> List[Any]:
        """Python-object allGather (JSON-serializable small control data),
        mirroring Spark BarrierTaskContext.allGather of JSON strings
        (reference utils.py:345-351)."""
        if not self.initialized:
            return [obj]
        out: List[Any] = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def allgather_rows(self, t: torch.Tensor) -> List[torch.Tensor]:
        """Variable-row-count tensor all-gather: pads local rows to the max,
        runs ONE tensor all-gather (RCCL over xGMI on GPU — large payloads
        like DBSCAN's replicated dataset or kNN's query/partial buffers must
        not ride the pickled-object path), and trims per-rank results."""
        if not self.initialized:
            return [t]
        n_local = t.shape[0]
        counts = self.allgather_obj(int(n_local))
        n_max = max(counts)
        shape = (n_max,) + tuple(t.shape[1:])
        padded = torch.zeros(shape, dtype=t.dtype, device=self._coll_device())
        if n_local > 0:
            padded[:n_local] = self.to_coll(t)
        out = [torch.empty_like(padded) for _ in range(self.world_size)]
        dist.all_gather(out, padded.contiguous())
        return [o[: counts[r]].to(t.device) for r, o in enumerate(out)]

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        """In-place broadcast, staged on the backend's device (gloo wants
        CPU buffers even when t lives on the GPU, same as allreduce_t)."""
        if not self.initialized:
            return t
        ct = self.to_coll(t).contiguous()
        dist.broadcast(ct, src=src)
        if ct.data_ptr() != t.data_ptr():
            t.copy_(ct.to(t.device))
        return t

    def broadcast_obj(self, obj: Any, src: int = 0) -> Any:
        if not self.initialized:
            return obj
        box = [obj]
        dist.broadcast_object_list(box, src=src)
        return box[0]

    def send(self, t: torch.Tensor, dst: int) -> None:
        dist.send(t.contiguous(), dst=dst)

    def recv(self, t: torch.Tensor, src: int) -> torch.Tensor:
        dist.recv(t, src=src)
        return t

    def _coll_device(self) -> torch.device:
        # nccl collectives need device tensors; gloo wants CPU tensors.
        return self.device if self.backend == "nccl" else torch.device("cpu")

    def to_coll(self, t: torch.Tensor) -> torch.Tensor:
        return t.to(self._coll_device())


def init_comm(backend: Optional[str] = None, timeout_s: int = 300) -> Comm:
    """Initialize (or return) the process-wide communicator.

    Reads torchrun env (RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT).
    Without env: single-process mode, using cuda:0 when a GPU is present.
    SRML_COMM_TIMEOUT_S bounds how long a collective blocks on a dead peer
    (the failure-detection analog of the reference's barrier-stage
    all-or-nothing semantics + NCCL abort, cuml_context.py:162-167).
    """
    global _COMM
    if _COMM is not None:
        return _COMM
    env_to = os.environ.get("SRML_COMM_TIMEOUT_S")
    if env_to:
        timeout_s = int(env_to)

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        device = torch.device(f"cuda:{local_rank % max(1, torch.cuda.device_count())}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    if world > 1:
        if backend is None:
            backend = os.environ.get("SRML_BACKEND") or ("nccl" if use_cuda else "gloo")
        if not dist.is_initialized():
            dist.init_process_group(
                backend=backend,
                rank=rank,
                world_size=world,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
    else:
        backend = backend or ("nccl" if use_cuda else "gloo")

    _COMM = Comm(rank=rank, world_size=world, device=device, backend=backend)
    return _COMM


def get_comm() -> Comm:
    """The process communicator, initializing single-process mode on demand."""
    global _COMM
    if _COMM is None:
        return init_comm()
    return _COMM


def shutdown_comm(abort: bool = False) -> None:
    """Tear down the process group (reference cuml_context.py:158-175 destroys
    on clean exit and aborts on exception to unblock surviving ranks).
    torch.distributed exposes one teardown — destroy_process_group — which on
    an exception path has abort semantics for peers (their next collective
    fails fast instead of blocking on a dead rank); the flag is accepted for
    reference-API parity."""
    global _COMM
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()
    _COMM = None


@dataclass
class PartitionDescriptor:
    """Global partition metadata assembled by allGather, mirroring
    reference utils.py:300-355: per-rank row counts (and nnz for sparse),
    global m (rows) and n (cols)."""

    parts_rank_size: List[tuple]  # [(rank, rows), ...] one per rank
    m: int
    n: int
    rank: int
    total_nnz: Optional[int] = None

    @classmethod
    def build(cls, comm: Comm, n_local_rows: int, n_cols: int, nnz: Optional[int] = None) -> "PartitionDescriptor":
        payload = {"rank": comm.rank, "rows": int(n_local_rows), "cols": int(n_cols)}
        if nnz is not None:
            payload["nnz"] = int(nnz)
        gathered = comm.allgather_obj(payload)
        parts = [(g["rank"], g["rows"]) for g in gathered]
        m = sum(r for _, r in parts)
        # max over ranks: an EMPTY shard of a variable-width list parquet
        # column cannot know the feature width locally (it reads as (0,0));
        # sizing fused all-reduce buffers from a rank-local 0 would deadlock
        # the collective (buffer shape mismatch across ranks)
        n_global = max(g.get("cols", 0) for g in gathered)
        total_nnz = sum(g.get("nnz", 0) for g in gathered) if nnz is not None else None
        return cls(parts_rank_size=parts, m=m, n=int(n_global), rank=comm.rank, total_nnz=total_nnz)

    def row_offset(self) -> int:
        """Global row offset of this rank's first local row."""
        off = 0
        for r, n in sorted(self.parts_rank_size):
            if r == self.rank:
                return off
            off += n
        return off
