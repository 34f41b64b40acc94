"""Federation communication layer — RCCL over xGMI (GPU) / gloo (CPU).

This replaces the reference's in-process dict passing + `torch.save` disk
round-trips (ref:experiment.py:189-202,233-241; ref:modules/server.py:46-57)
with real collectives.  Execution model: ONE PROCESS PER GPU, launched by
`torch.distributed.run`; each rank hosts a shard of the simulated edge
clients and a replica of the server object.  On MI355X the backend string
"nccl" is RCCL, whose ring/direct collectives run over the node's
all-to-all xGMI mesh (7 p2p links × ≈153 GB/s per GPU).

Aggregation strategies (see SURVEY.md §2.9 C1–C4):
  C1 weighted average  -> pre-scale by k_c/Σk locally, all_reduce(SUM)
  C2/C3 state gather   -> flat-buffer all_gather_into_tensor (tensor codec)
  C4 personalized mix  -> all_gather uploads, mix locally per destination
The generic `all_gather_object` path is the correctness fallback for
arbitrary python state (Fisher dicts, exemplar sets, ...).

Stale-upload semantics preserved: the server replica keeps every client's
LAST upload and averages over all registered clients, not just this round's
online set (ref:methods/fedavg.py:388-390) — offline clients contribute their
cached state because their entry in `server.clients` is simply not replaced.
"""

from __future__ import annotations

import datetime
import os
from typing import Any, Dict, List, Optional

import torch
import torch.distributed as dist

from flreid_amd.tools.logger import Logger

_log = Logger("parallel.comm")
_CTX: Optional["FedContext"] = None


class FedContext:
    """Process-group wrapper: rank/world/device + federation collectives."""

    def __init__(self, rank: int = 0, world_size: int = 1, device: str = "cpu",
                 backend: Optional[str] = None):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.backend = backend

    # ------------------------------------------------------------ properties
    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def is_rank0(self) -> bool:
        return self.rank == 0

    def owner_of(self, client_index: int) -> int:
        """Static client→rank sharding: round robin."""
        return client_index % self.world_size

    # ----------------------------------------------------------- collectives
    def barrier(self) -> None:
        if self.is_distributed:
            if self.backend == "nccl":
                dist.barrier(device_ids=[torch.cuda.current_device()])
            else:
                dist.barrier()

    def all_gather_object(self, obj: Any) -> List[Any]:
        if not self.is_distributed:
            return [obj]
        out: List[Any] = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def broadcast_object(self, obj: Any, src: int = 0) -> Any:
        if not self.is_distributed:
            return obj
        box = [obj if self.rank == src else None]
        dist.broadcast_object_list(box, src=src)
        return box[0]

    def all_reduce_(self, tensor: torch.Tensor, op: str = "sum") -> torch.Tensor:
        if self.is_distributed:
            dist.all_reduce(tensor, op=dist.ReduceOp.SUM if op == "sum" else dist.ReduceOp.MAX)
        return tensor

    def all_reduce_scalar(self, value: float, op: str = "sum") -> float:
        if not self.is_distributed:
            return value
        t = torch.tensor([value], dtype=torch.float64, device=self._comm_device())
        self.all_reduce_(t, op)
        return float(t.item())

    def _comm_device(self) -> str:
        return self.device if self.backend == "nccl" else "cpu"

    # ------------------------------------------- tensor-codec state gather
    def all_gather_flat(self, flat: torch.Tensor) -> torch.Tensor:
        """All-gather one equally-sized flat buffer per rank -> [W, numel].

        This is the xGMI hot path for C2/C3 (per-layer / full-state gather):
        one contiguous buffer, one RCCL all-gather, no pickling.
        """
        if not self.is_distributed:
            return flat.unsqueeze(0)
        flat = flat.contiguous().to(self._comm_device())
        out = torch.empty(self.world_size * flat.numel(), dtype=flat.dtype,
                          device=flat.device)
        dist.all_gather_into_tensor(out, flat)
        return out.view(self.world_size, flat.numel())

    def weighted_allreduce(self, tensors: Dict[str, torch.Tensor],
                           weight: float) -> Dict[str, torch.Tensor]:
        """C1: Σ_ranks weight_r · tensors_r via pre-scale + all_reduce(SUM).

        Caller supplies its local weighted PARTIAL SUM weight (e.g. the sum of
        k_c/Σk over its local clients); tensors must share names/shapes across
        ranks.  Bucketed into one flat buffer per dtype for per-link-bound
        xGMI rings (fewer, larger collectives).
        """
        scaled = {n: t.detach().to(torch.float32) * weight for n, t in tensors.items()}
        if not self.is_distributed:
            return scaled
        names = sorted(scaled.keys())
        flat = torch.cat([scaled[n].reshape(-1) for n in names]).to(self._comm_device())
        self.all_reduce_(flat)
        out: Dict[str, torch.Tensor] = {}
        offset = 0
        for n in names:
            numel = scaled[n].numel()
            out[n] = flat[offset:offset + numel].view_as(scaled[n]).to(scaled[n].device)
            offset += numel
        return out


def init_context(device: Optional[str] = None,
                 timeout_s: int = 600) -> FedContext:
    """Initialise from torchrun env (RANK/WORLD_SIZE/MASTER_*) or fall back to
    a single-process context (the reference-simulator-compatible mode)."""
    global _CTX
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    if device is None:
        if torch.cuda.is_available():
            device = f"cuda:{local_rank % max(1, torch.cuda.device_count())}"
        else:
            device = "cpu"

    if world_size <= 1:
        _CTX = FedContext(0, 1, device)
        return _CTX

    backend = "nccl" if device.startswith("cuda") else "gloo"
    if device.startswith("cuda"):
        torch.cuda.set_device(device)
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    _CTX = FedContext(rank, world_size, device, backend)
    _log.info(f"initialised rank {rank}/{world_size} on {device} ({backend})")
    return _CTX


def get_context() -> FedContext:
    global _CTX
    if _CTX is None:
        _CTX = init_context()
    return _CTX


def destroy_context() -> None:
    global _CTX
    if dist.is_initialized():
        dist.destroy_process_group()
    _CTX = None
    from flreid_amd.parallel import codec
    codec.reset_schema_cache()
