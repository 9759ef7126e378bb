"""RCCL cold-path proofing on a single GPU.

The multi-GPU protocol (parallel/dist.py) is bitwise-tested on CPU/gloo at
world sizes 1-3, but an 8-GPU driver run initializes NCCL(=RCCL) cold.
These tests run the REAL driver under an initialized world_size=1 NCCL
group in a spawned process — real RCCL init, real fp64/fp32 collectives
over the group (all_gather_updates no longer short-circuits when a group
exists) — and assert bitwise equality with the uninitialized single-process
path.  What this cannot cover on one GPU: cross-rank xGMI transport.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

SIZES = (2000, 400)


def _nccl_worker(out_q):
    os.environ['RANK'] = '0'
    os.environ['LOCAL_RANK'] = '0'
    os.environ['WORLD_SIZE'] = '1'
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = '29711'
    import torch
    import torch.distributed as dist
    import rlr_amd.data.datasets as D
    D.DEFAULT_SIZES['fmnist'] = SIZES
    from rlr_amd.federated import run
    from rlr_amd.options import default_args
    from rlr_amd.parallel import dist as pdist

    args = default_args(num_agents=4, rounds=2, snap=2, local_ep=1, bs=64,
                        synthetic=True, no_tb=True, data='fmnist',
                        num_corrupt=1, poison_frac=0.5, robustLR_threshold=3,
                        device='cuda:0')
    try:
        h = run(args)
        assert dist.is_initialized() and dist.get_backend() == 'nccl'

        # direct collective coverage on the group: fp64 all-reduce,
        # fp64 all-gather (the update transport dtype), fp32 broadcast
        x = torch.arange(1024, dtype=torch.float64, device='cuda:0') * 0.5
        ref = x.clone()
        dist.all_reduce(x)
        assert torch.equal(x, ref)
        out = torch.empty(1, 1024, dtype=torch.float64, device='cuda:0')
        dist.all_gather_into_tensor(out, x.unsqueeze(0))
        assert torch.equal(out[0], x)
        y = torch.randn(257, device='cuda:0')
        ref = y.clone()
        dist.broadcast(y, src=0)
        assert torch.equal(y, ref)

        out_q.put(h['final_params'].numpy().copy())
    finally:
        pdist.teardown()


@pytest.mark.timeout(600)
def test_ws1_nccl_bitwise_matches_uninitialized():
    """Full driver under a ws=1 NCCL group == plain single-process run."""
    ctx = mp.get_context('spawn')
    q = ctx.Queue()
    p = ctx.Process(target=_nccl_worker, args=(q,))
    p.start()
    nccl_params = torch.from_numpy(q.get(timeout=420))
    p.join(timeout=120)
    assert p.exitcode == 0

    import rlr_amd.data.datasets as D
    old = D.DEFAULT_SIZES['fmnist']
    D.DEFAULT_SIZES['fmnist'] = SIZES
    try:
        from rlr_amd.federated import run
        from rlr_amd.options import default_args
        args = default_args(num_agents=4, rounds=2, snap=2, local_ep=1,
                            bs=64, synthetic=True, no_tb=True, data='fmnist',
                            num_corrupt=1, poison_frac=0.5,
                            robustLR_threshold=3, device='cuda:0')
        h = run(args)
    finally:
        D.DEFAULT_SIZES['fmnist'] = old
    assert torch.equal(h['final_params'], nccl_params)
