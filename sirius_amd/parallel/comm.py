from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


class Comm:
    """Thin communicator facade; no-op in serial runs."""

    def __init__(self, group=None):
        self.group = group
        if dist.is_available() and dist.is_initialized():
            self.rank = dist.get_rank(group)
            self.size = dist.get_world_size(group)
        else:
            self.rank = 0
            self.size = 1

    @property
    def active(self) -> bool:
        return self.size > 1

    def allreduce_(self, t: torch.Tensor):
        if self.active:
            dist.all_reduce(t, group=self.group)
        return t

    def allreduce_scalar(self, x: float) -> float:
        if not self.active:
            return x
        t = torch.tensor([x], dtype=torch.float64)
        dist.all_reduce(t, group=self.group)
        return float(t[0])

    def allgather_object(self, obj):
        if not self.active:
            return [obj]
        out = [None] * self.size
        dist.all_gather_object(out, obj, group=self.group)
        return out

    def broadcast_(self, t: torch.Tensor, src: int = 0):
        if self.active:
            dist.broadcast(t, src=src, group=self.group)
        return t

    def barrier(self):
        if self.active:
            dist.barrier(group=self.group)

    def allgather_rows(self, t: torch.Tensor, counts: list[int]) -> torch.Tensor:
        """Concatenate row-blocks from all ranks: local t [n_r, m] ->
        [sum(counts), m].  Pads to max(counts) for gloo/nccl equal-shape
        all_gather, then trims."""
        if not self.active:
            return t
        nmax = max(counts)
        m = t.shape[1] if t.dim() > 1 else 0
        pad = torch.zeros(nmax, m, dtype=t.dtype, device=t.device)
        pad[:t.shape[0]] = t
        outs = [torch.empty_like(pad) for _ in range(self.size)]
        dist.all_gather(outs, pad.contiguous(), group=self.group)
        return torch.cat([outs[r][:counts[r]] for r in range(self.size)], dim=0)


def make_band_comm(npb: int):
    """Split the world into k-groups of `npb` band ranks each
    (reference Simulation_context::init_comm, simulation_context.cpp:1301:
    world -> comm_k x comm_band).  Rank r: band_rank = r % npb,
    k_color = r // npb.  Returns (band_comm, k_color, num_kgroups)."""
    world = get_comm()
    if not world.active or npb <= 1:
        return Comm(), 0, world.size if world.active else 1
    if world.size % npb:
        raise ValueError(f"world size {world.size} not divisible by "
                         f"band-group size {npb}")
    nk_groups = world.size // npb
    my_group = None
    for color in range(nk_groups):
        ranks = list(range(color * npb, (color + 1) * npb))
        g = dist.new_group(ranks=ranks)
        if world.rank in ranks:
            my_group = g
    return Comm(my_group), world.rank // npb, nk_groups


_comm: Optional[Comm] = None


def init_distributed(backend: str | None = None) -> Comm:
    """Initialize torch.distributed from torchrun env vars if present.

    backend defaults to nccl (=RCCL on ROCm) when a GPU is visible,
    gloo otherwise. MASTER_ADDR should be 127.0.0.1 in this environment.
    """
    global _comm
    if "RANK" in os.environ and not (dist.is_available() and dist.is_initialized()):
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        dist.init_process_group(backend=backend)
    _comm = Comm()
    return _comm


def get_comm() -> Comm:
    global _comm
    if _comm is None:
        _comm = Comm()
    return _comm
