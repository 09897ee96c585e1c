"""Distribution context helpers.

One process per GPU, torch.distributed over RCCL (backend "nccl" IS RCCL
on ROCm) for device tensors, gloo for CPU tests.  Bucket ownership is
round-robin: rank r owns buckets {b : b % world == r} — the build-time
all-to-all exchanges rows to their owners; the accelerated join then runs
with ZERO cross-GPU communication because both sides of a bucket live on
the same rank.
"""

from __future__ import annotations

from typing import List

import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def collective_tensor(values):
    """Small int/float tensor placed correctly for the active backend:
    NCCL/RCCL collectives need device tensors (gloo-only CPU tests never
    catch a CPU tensor here — it fails on real multi-GPU runs)."""
    import torch
    t = torch.tensor(values)
    if is_distributed() and dist.get_backend() == "nccl" and \
            torch.cuda.is_available():
        t = t.cuda()
    return t


def bucket_owner(bucket: int, world: int) -> int:
    return bucket % world


def owned_buckets(num_buckets: int, rank: int, world: int) -> List[int]:
    return [b for b in range(num_buckets) if b % world == rank]
