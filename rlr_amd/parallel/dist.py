"""Distributed runtime: one process per GPU over RCCL/xGMI.

The reference is single-process (no torch.distributed anywhere; its
"network" is a python dict, federated.py:67-72).  Here:
  * ranks are launched by torch.distributed.run (one per GPU); backend is
    "nccl" (= RCCL on ROCm) on GPU, "gloo" for the CPU test path.
  * the sampled agents of a round are partitioned into contiguous chunks
    across ranks; each rank trains its chunk sequentially and the per-agent
    fp64 update vectors are ALL-GATHERED (BASELINE.json: per-agent update
    all-gather over xGMI) into the identical (S, n_params) matrix on every
    rank; aggregation then runs redundantly — deterministic, no broadcast
    needed.
"""

import os

import torch
import torch.distributed as dist


def setup(args=None):
    """Initialize from torchrun env vars; no-op single-process otherwise.
    Returns (rank, world_size)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    # any torchrun launch (RANK/WORLD_SIZE in env) initializes the group —
    # including world_size 1, so a single-GPU lease exercises the real
    # RCCL init + collectives rather than the uninitialized shortcut
    if 'RANK' in os.environ and 'WORLD_SIZE' in os.environ:
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get('LOCAL_RANK', 0)))
        dist.init_process_group(backend=backend)
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def teardown():
    if dist.is_initialized():
        dist.destroy_process_group()


def rank_world():
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def is_main():
    return rank_world()[0] == 0


def barrier():
    if dist.is_initialized():
        dist.barrier()


def chunk_bounds(n_items, world, rank):
    """Contiguous chunking: every rank owns ceil(n/w) slots (the gather
    layout pads the tail).  Returns (lo, hi) of the rank's valid items."""
    import math
    c = math.ceil(n_items / world)
    lo = min(rank * c, n_items)
    hi = min(lo + c, n_items)
    return lo, hi, c


def all_gather_updates(local: torch.Tensor, n_valid_per_rank, chunk: int):
    """local: (chunk, n) — rank's updates in its slots (tail rows zero).
    Returns the (S, n) stacked matrix in global sampled order, S = sum of
    valid counts.  Single-process: returns local's valid rows."""
    rank, world = rank_world()
    if world == 1:
        if not dist.is_initialized():
            return local[:n_valid_per_rank[0]]
        # ws=1 under an initialized group still issues the real collective
        # (a self-copy) so single-GPU runs exercise the RCCL path end-to-end
    out = torch.empty(world * chunk, local.shape[1], dtype=local.dtype,
                      device=local.device)
    dist.all_gather_into_tensor(out, local)
    rows = []
    for r in range(world):
        rows.append(out[r * chunk: r * chunk + n_valid_per_rank[r]])
    return torch.cat(rows, dim=0)


def all_reduce_(t: torch.Tensor):
    if dist.is_initialized():
        dist.all_reduce(t)
    return t
