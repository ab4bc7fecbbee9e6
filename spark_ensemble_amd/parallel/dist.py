"""RCCL/xGMI process-group layer — the replacement for Spark's treeAggregate.

The reference's only cross-process machinery is Spark's driver-coordinated
``treeReduce``/``treeAggregate`` (full call-site inventory: SURVEY.md
section 2.6; e.g. reference BoostingRegressor.scala:208,234,244-249,
GBMClassifier.scala:344-355).  On MI355X that becomes: one process per GPU,
``torch.distributed`` with the nccl backend (= RCCL over xGMI) and

  * scalar / short-vector reductions (boosting errors, hessian sums,
    line-search loss+gradient partials) -> latency-bound all-reduce; callers
    batch per-round scalars into one fused buffer where it matters,
  * per-tree-level histogram all-reduce (multi-MB, bandwidth-bound; new in
    this rebuild because distributed tree building lived inside MLlib).

On CPU test rigs the same code runs over the gloo backend; single-process
runs short-circuit every collective (world_size == 1).
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


class _NoopWork:
    def wait(self):
        return True


class Comm:
    """Thin communicator: no-op at world_size 1, collective otherwise.

    Tracing: set ``SEA_COMM_TRACE=<path>`` to append one line per
    collective — ``op dtype numel`` — so tests can assert that every rank
    issues the IDENTICAL collective sequence (the property that keeps
    RCCL deadlock-free) and that payload sizes match the documented
    per-level histogram math (docs/distributed.md)."""

    def __init__(self, rank: int = 0, world_size: int = 1, device=None):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        base = os.environ.get("SEA_COMM_TRACE")
        self._trace_path = f"{base}.r{rank}" if base else None

    def _trace(self, op: str, tensor: torch.Tensor):
        if self._trace_path:
            with open(self._trace_path, "a") as f:
                f.write(
                    f"{op} {str(tensor.dtype).replace('torch.', '')} "
                    f"{tensor.numel()}\n"
                )

    # -- factory ----------------------------------------------------------
    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    # -- collectives ------------------------------------------------------
    _OPS = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
            "min": dist.ReduceOp.MIN} if dist.is_available() else {}

    def all_reduce_(self, tensor: torch.Tensor, op: str = "sum") -> torch.Tensor:
        """In-place all-reduce. ``op`` in {sum, max, min}."""
        if not self.is_distributed:
            return tensor
        self._trace(f"all_reduce_{op}", tensor)
        dist.all_reduce(tensor, op=self._OPS[op])
        return tensor

    def all_reduce_async(self, tensor: torch.Tensor, op: str = "sum"):
        """Enqueue an all-reduce and return a work handle; ``.wait()``
        orders the CURRENT stream after the collective (host does not
        block on NCCL/RCCL).  Lets callers pipeline chunked histogram
        reduces against split-search compute (tree_grower)."""
        if not self.is_distributed:
            return _NoopWork()
        self._trace(f"all_reduce_async_{op}", tensor)
        return dist.all_reduce(tensor, op=self._OPS[op], async_op=True)

    def all_reduce_scalar(self, value: float, op: str = "sum") -> float:
        if not self.is_distributed:
            return float(value)
        dev = self.device if self.device is not None else torch.device("cpu")
        t = torch.tensor([value], dtype=torch.float64, device=dev)
        self.all_reduce_(t, op)
        return float(t.item())

    def all_gather_object(self, obj):
        if not self.is_distributed:
            return [obj]
        out = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def broadcast_(self, tensor: torch.Tensor, src: int = 0) -> torch.Tensor:
        if not self.is_distributed:
            return tensor
        self._trace("broadcast", tensor)
        dist.broadcast(tensor, src=src)
        return tensor

    def barrier(self):
        if self.is_distributed:
            dist.barrier()

    def __repr__(self):
        return f"Comm(rank={self.rank}, world={self.world_size}, dev={self.device})"


_GLOBAL_COMM: Optional[Comm] = None


def get_comm() -> Comm:
    """The process-global communicator (auto-initialized from env if the
    process was launched by torch.distributed.run / torchrun)."""
    global _GLOBAL_COMM
    if _GLOBAL_COMM is None:
        if dist.is_available() and dist.is_initialized():
            rank = dist.get_rank()
            ws = dist.get_world_size()
            dev = _local_device()
            _GLOBAL_COMM = Comm(rank, ws, dev)
        elif "RANK" in os.environ and "WORLD_SIZE" in os.environ and int(
            os.environ["WORLD_SIZE"]
        ) > 1:
            _GLOBAL_COMM = init_from_env()
        else:
            _GLOBAL_COMM = Comm()
    return _GLOBAL_COMM


def set_comm(comm: Optional[Comm]):
    global _GLOBAL_COMM
    _GLOBAL_COMM = comm


def _local_device():
    if torch.cuda.is_available():
        local = int(os.environ.get("LOCAL_RANK", 0))
        return torch.device("cuda", local)
    return torch.device("cpu")


def init_from_env(backend: Optional[str] = None, timeout_s: int = 1800) -> Comm:
    """Initialize torch.distributed from torchrun env vars.

    backend: 'nccl' (RCCL on ROCm) when a GPU is visible, else 'gloo'.
    """
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        dist.init_process_group(
            backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
        )
    comm = Comm(dist.get_rank(), dist.get_world_size(), _local_device())
    set_comm(comm)
    return comm
