"""World-size invariance: the same seed must produce the
bit-identical global model at world_size 1 and 2.  Runs the real driver in
spawned processes over gloo on CPU (the GPU path swaps in RCCL with the
same protocol)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

SIZES = (2000, 400)


def _worker(rank, world_size, port, out_q):
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world_size)
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    import rlr_amd.data.datasets as D
    D.DEFAULT_SIZES['fmnist'] = SIZES
    from rlr_amd.federated import run
    from rlr_amd.options import default_args
    from rlr_amd.parallel import dist as pdist
    args = default_args(num_agents=4, rounds=2, snap=2, local_ep=1, bs=64,
                        synthetic=True, no_tb=True, data='fmnist',
                        num_corrupt=1, poison_frac=0.5,
                        robustLR_threshold=3, device='cpu')
    try:
        h = run(args)
        if rank == 0:
            # numpy copy: pickled by value (a torch tensor would ship a
            # shared-memory fd that dies with this process)
            out_q.put((h['final_params'].numpy().copy(),
                       list(h['val_acc']), list(h['poison_acc'])))
    finally:
        pdist.teardown()


def _run_world(world_size):
    ctx = mp.get_context('spawn')
    q = ctx.Queue()
    port = 29531 + world_size
    procs = [ctx.Process(target=_worker, args=(r, world_size, port, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    params, val_acc, poison_acc = q.get(timeout=300)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    return torch.from_numpy(params), val_acc, poison_acc


@pytest.mark.timeout(600)
def test_world_size_invariance():
    p1, v1, pa1 = _run_world(1)
    p2, v2, pa2 = _run_world(2)
    assert torch.equal(p1, p2), (p1 - p2).abs().max()
    # sharded eval (strided batches + all-reduced counts) must report the
    # exact same accuracies as the single-process full pass
    assert v1 == v2 and pa1 == pa2


@pytest.mark.timeout(600)
def test_three_ranks_uneven_chunks():
    """4 sampled agents over 3 ranks (2/2/0 after ceil-chunking — one rank
    idle) must still match the single-process result."""
    p1, v1, _ = _run_world(1)
    p3, v3, _ = _run_world(3)
    assert torch.equal(p1, p3)
    assert v1 == v3
