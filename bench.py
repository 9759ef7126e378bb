#!/usr/bin/env python3
"""Flagship benchmark: FL rounds/sec on the BASELINE.json headline config —
FMNIST CNN, 10-agent RLR (num_corrupt=1, poison_frac=0.5,
robustLR_threshold=4), synthetic data, random-init weights.

One "step" = one full FL communication round: every sampled agent runs
local_ep epochs of clipped SGD on its shard, per-agent fp64 updates are
(all-)gathered, the RLR-modulated FedAvg aggregate is applied.

Scaling is WEAK: each GPU hosts 10 agents with 6000 samples each
(num_agents = 10*N, num_corrupt = N, train set = 60000*N), so rounds/sec
should stay flat as N grows.  Compute dtype is fp32 — the reference's own
precision (PyTorch-1.9-era fp32 training; nothing is reduced).

Contract: --gpus N --steps K --warmup W; W untimed rounds, then
barrier+synchronize, K timed rounds, barrier+synchronize; elapsed is the
MAX over ranks; rank 0 prints one JSON line."""

import argparse
import json
import os
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=20)
    ap.add_argument('--warmup', type=int, default=5)
    ap.add_argument('--agents_per_gpu', type=int, default=10)
    ap.add_argument('--data', type=str, default='fmnist')
    ap.add_argument('--model', type=str, default=None,
                    help="e.g. resnet18 (BASELINE configs 3-4: CIFAR10 "
                         "ResNet18); default: the dataset's reference CNN")
    ap.add_argument('--agents_per_stream_override', type=int, default=0)
    ap.add_argument('--dtype', type=str, default='fp32',
                    choices=['fp32', 'bf16'],
                    help="client compute dtype; the headline FMNIST config "
                         "is fp32 (the reference's own precision)")
    a = ap.parse_args()

    from rlr_amd.federated import build_world, run_round
    from rlr_amd.options import default_args
    from rlr_amd.parallel import dist as pdist
    import rlr_amd.data.datasets as D

    n = int(os.environ.get('WORLD_SIZE', a.gpus))
    rank, world = pdist.setup()
    assert world == n or n == 1, (world, n)
    n = world

    use_cuda = torch.cuda.is_available()
    device = f'cuda:{int(os.environ.get("LOCAL_RANK", 0))}' if use_cuda else 'cpu'

    num_agents = a.agents_per_gpu * n
    samples_per_agent = 6000 if a.data == 'fmnist' else 1250
    D.DEFAULT_SIZES[a.data] = (samples_per_agent * num_agents, 10000)
    args = default_args(
        data=a.data, model=a.model, dtype=a.dtype, num_agents=num_agents,
        num_corrupt=1 * n, poison_frac=0.5, robustLR_threshold=4,
        aggr='avg', local_ep=2, bs=256, agent_frac=1.0, synthetic=True,
        no_tb=True, snap=10 ** 9, device=device,
        agents_per_stream=a.agents_per_stream_override)

    world_state = build_world(args)

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        pdist.barrier()

    for rnd in range(1, a.warmup + 1):
        run_round(args, world_state, rnd, rank, n)

    sync()
    t0 = time.perf_counter()
    for rnd in range(a.warmup + 1, a.warmup + a.steps + 1):
        run_round(args, world_state, rnd, rank, n)
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if use_cuda else 'cpu')
    if torch.distributed.is_initialized():
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed_max = float(t.item())

    if rank == 0:
        value = a.steps / elapsed_max
        out = {
            "metric": "fl_rounds_per_sec",
            "value": value,
            "unit": "rounds/s",
            "n_gpus": n,
            "steps": a.steps,
            "warmup": a.warmup,
            "ms_per_step": elapsed_max / a.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no throughput numbers
            "dtype": a.dtype,
            "data": "synthetic",
            "config": {
                "model": a.model or ("CNN_MNIST" if a.data != 'cifar10'
                                     else "CNN_CIFAR"),
                "dataset": f"{a.data} (synthetic, "
                           f"{samples_per_agent} samples/agent)",
                "num_agents": num_agents,
                "agents_per_gpu": a.agents_per_gpu,
                "num_corrupt": 1 * n,
                "poison_frac": 0.5,
                "robustLR_threshold": 4,
                "aggr": "avg",
                "local_ep": 2,
                "global_batch": 256,
                "parallelism": f"dp{n} (agent-sharded, RCCL all-gather)",
            },
        }
        print(json.dumps(out))
    pdist.teardown()


if __name__ == '__main__':
    main()
