"""Per-purpose deterministic RNG streams.

The reference draws from three global RNGs (np.random for client sampling,
federated.py:68; random.sample for poison-index selection, utils.py:166;
torch's global RNG for init/dropout), which makes runs irreproducible across
process counts.  Here every stream is derived from (master seed, purpose,
key...) with a stable hash, so any rank — and any world size — derives the
identical stream (SURVEY.md §7 hard part 3).
"""

import hashlib
import struct

import numpy as np
import torch

_PURPOSES = (
    'init',        # model weight init
    'sampling',    # per-round client sampling
    'poison',      # per-agent poison index selection
    'shuffle',     # per-(agent, round, epoch) batch order
    'dropout',     # per-(agent, round, step) dropout masks
    'noise',       # per-round server DP noise
    'data',        # synthetic dataset generation
)


def derive_seed(master_seed: int, purpose: str, *keys) -> int:
    """64-bit stable seed from (master, purpose, keys) via blake2b."""
    h = hashlib.blake2b(digest_size=8)
    h.update(struct.pack('<q', master_seed))
    h.update(purpose.encode())
    for k in keys:
        h.update(struct.pack('<q', int(k)))
    return int.from_bytes(h.digest(), 'little') & 0x7FFFFFFFFFFFFFFF


def np_rng(master_seed: int, purpose: str, *keys) -> np.random.Generator:
    return np.random.Generator(np.random.PCG64(derive_seed(master_seed, purpose, *keys)))


def torch_gen(master_seed: int, purpose: str, *keys, device='cpu') -> torch.Generator:
    g = torch.Generator(device=device)
    g.manual_seed(derive_seed(master_seed, purpose, *keys))
    return g


def sample_agents(master_seed: int, rnd: int, num_agents: int, agent_frac: float):
    """Per-round client sampling — same draw on every rank.

    Mirrors np.random.choice(K, floor(K*frac), replace=False)
    (reference federated.py:68).  Returns a python list of agent ids in the
    drawn order.
    """
    import math
    n = math.floor(num_agents * agent_frac)
    rng = np_rng(master_seed, 'sampling', rnd)
    return rng.choice(num_agents, n, replace=False).tolist()
