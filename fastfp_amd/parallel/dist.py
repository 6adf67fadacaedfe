"""Data-parallel sharding over torch.distributed (RCCL on ROCm).

The reference has no distributed layer at all; this is
the MI355X-native scaling design: one process per GPU, NM-Fp draws (or
plain-Fp frequencies) block-sharded across ranks, one RCCL all-gather of
the Fp spectrum shards over xGMI at the end of the sweep.  Payloads are
O(F*D*8 bytes) so the collective is latency-bound and a single flat
all-gather suffices — no overlap machinery is needed.

On CPU CI the same code runs with the gloo backend (world_size >= 1).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str = None, device: torch.device = None):
    """Initialize torch.distributed from the torchrun environment.

    Returns (rank, world_size, device).  With no RANK in the env this is
    a single-process run: returns (0, 1, device) without init.
    Backend default: "nccl" (= RCCL on ROCm) when CUDA devices are
    visible, else "gloo".

    A torchrun launch with WORLD_SIZE=1 still initializes the process
    group: the RCCL communicator setup and every collective call site
    then run exactly as in the multi-GPU case (one-rank collectives are
    cheap), so a 1-GPU box exercises the full distributed path.
    """
    if "RANK" not in os.environ:
        if device is None:
            device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
        return 0, 1, device

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if device is None:
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
            device = torch.device(f"cuda:{local_rank}")
        else:
            device = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=600),
        )
    return rank, world, device


def shard_slice(n: int, rank: int, world: int) -> slice:
    """Contiguous block shard of range(n) for this rank (first ranks get
    the remainder)."""
    base = n // world
    rem = n % world
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return slice(lo, hi)


def all_gather_concat(local: torch.Tensor, world: int, dim: int = 0) -> torch.Tensor:
    """All-gather variable-length shards along ``dim`` and concatenate.

    Uses all_gather with per-rank padded buffers (shard sizes may differ
    by 1 from block sharding).  With an initialized 1-rank group the
    collectives still run (exercising the RCCL path); without a group
    this is the identity.
    """
    if not dist.is_initialized():
        return local
    local = local.contiguous()
    n_local = torch.tensor([local.shape[dim]], dtype=torch.int64, device=local.device)
    sizes = [torch.zeros_like(n_local) for _ in range(world)]
    dist.all_gather(sizes, n_local)
    sizes = [int(s.item()) for s in sizes]
    maxn = max(sizes)
    shape = list(local.shape)
    if shape[dim] < maxn:
        shape[dim] = maxn
        padded = torch.zeros(shape, dtype=local.dtype, device=local.device)
        sl = [slice(None)] * local.dim()
        sl[dim] = slice(0, local.shape[dim])
        padded[tuple(sl)] = local
        local = padded
    bufs = [torch.empty_like(local) for _ in range(world)]
    dist.all_gather(bufs, local)
    outs = []
    for b, n in zip(bufs, sizes):
        sl = [slice(None)] * b.dim()
        sl[dim] = slice(0, n)
        outs.append(b[tuple(sl)])
    return torch.cat(outs, dim=dim)


def cleanup():
    if dist.is_initialized():
        dist.destroy_process_group()
