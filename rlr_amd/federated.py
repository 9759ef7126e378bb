"""FL driver — the orchestrator (reference src/federated.py:21-95), rebuilt
for one-process-per-GPU execution.

Per round (reference federated.py:65-92):
  1. every rank derives the SAME sampled agent list from the shared
     'sampling' stream (replaces np.random.choice, federated.py:68);
  2. the sampled list is split into contiguous chunks across ranks; each
     rank trains its agents back-to-back on its GPU (agents restore the
     global replica themselves — the "broadcast" of federated.py:72 is a
     flat-buffer copy);
  3. per-agent fp64 updates are all-gathered over RCCL/xGMI into the same
     (S, n_params) matrix everywhere; every rank aggregates redundantly
     (deterministic — bitwise-identical global models at any world size);
  4. on snap rounds, clean + poisoned validation metrics are computed and
     logged with the reference's TensorBoard scalar names
     (federated.py:78-92), and a checkpoint is written if --ckpt_dir.
"""

import copy
import os
import time

import torch

from .agent import Agent
from .aggregation import Aggregation
from .data import get_datasets, distribute_data, poison_dataset
from .flatmodel import FlatParamModel
from .models import get_model
from .options import args_parser
from .parallel import dist as pdist
from .utils import (build_writer, derive_seed, get_loss_n_accuracy,
                    load_checkpoint, materialize_eval_set,
                    print_exp_details, sample_agents, save_checkpoint)


def build_world(args):
    """Datasets, agents, model, aggregator — identical on every rank."""
    device = torch.device(args.device)

    if args.data == 'fedemnist':
        user_datasets, val_dataset = get_datasets(args.data, args)
        agents, agent_data_sizes = [], {}
        for _id in range(args.num_agents):
            a = Agent(_id, args, user_dataset=user_datasets[_id])
            agent_data_sizes[_id] = a.n_data
            agents.append(a)
        X_val, Y_val = materialize_eval_set(val_dataset, device=device)
        # poisoned val: base-class samples, full pattern (federated.py:42-45)
        pv = copy.deepcopy(val_dataset)
        idxs = (pv.targets == args.base_class).nonzero().flatten().tolist()
        poison_dataset(pv, args, idxs, poison_all=True, agent_idx=-1)
        X_pv, Y_pv = materialize_eval_set(pv, idxs=idxs, device=device)
    else:
        train_dataset, val_dataset = get_datasets(args.data, args)
        if device.type == 'cuda':
            # raw uint8 storage moves to HBM BEFORE poisoning: trojan
            # injection runs on-device and poisoned batches never leave HBM
            # (BASELINE.json; ops/csrc/poison.hip)
            train_dataset.data = train_dataset.data.to(device)
            train_dataset.targets = train_dataset.targets.to(device)
            val_dataset.data = val_dataset.data.to(device)
            val_dataset.targets = val_dataset.targets.to(device)
        user_groups = distribute_data(train_dataset, args)
        agents, agent_data_sizes = [], {}
        for _id in range(args.num_agents):
            a = Agent(_id, args, train_dataset=train_dataset,
                      data_idxs=user_groups[_id])
            agent_data_sizes[_id] = a.n_data
            agents.append(a)
        X_val, Y_val = materialize_eval_set(val_dataset, device=device)
        pv = copy.deepcopy(val_dataset)
        idxs = (pv.targets == args.base_class).nonzero().flatten().tolist()
        poison_dataset(pv, args, idxs, poison_all=True, agent_idx=-1)
        X_pv, Y_pv = materialize_eval_set(pv, idxs=idxs, device=device)

    # deterministic init on every rank (no broadcast needed)
    torch.manual_seed(derive_seed(args.seed, 'init'))
    model = get_model(args.data, getattr(args, 'model', None))
    if getattr(args, 'dtype', 'fp32') == 'bf16' and device.type == 'cuda':
        model.set_compute_dtype(torch.bfloat16)
    global_model = FlatParamModel(model, device)

    aggregator = Aggregation(agent_data_sizes, global_model.n_params,
                             (X_pv, Y_pv), args, None)
    return dict(agents=agents, agent_data_sizes=agent_data_sizes,
                global_model=global_model, aggregator=aggregator,
                X_val=X_val, Y_val=Y_val, X_pv=X_pv, Y_pv=Y_pv)


def _replica_pool(world, args, gm):
    """Model replicas on separate HIP streams: up to R sampled agents train
    CONCURRENTLY per GPU (the reference trains agents strictly sequentially,
    federated.py:68-72; the tiny CNN's kernels underfill 256 CUs, so
    stream-level concurrency is the chip-filling lever)."""
    if 'replicas' not in world:
        # sweep on MI355X: 2 replicas > 4/6/8 — the big conv
        # kernels already fill the chip; 2 streams cover the small-kernel
        # tails without cache thrash
        R = args.agents_per_stream if args.agents_per_stream > 0 else 2
        reps = []
        for _ in range(max(1, R)):
            m = get_model(args.data, getattr(args, 'model', None))
            if getattr(args, 'dtype', 'fp32') == 'bf16':
                m.set_compute_dtype(torch.bfloat16)
            rep = FlatParamModel(m, gm.device)
            reps.append((rep, torch.cuda.Stream(device=gm.device)))
        world['replicas'] = reps
    return world['replicas']


def run_round(args, world, rnd, rank, world_size, phase_times=None):
    """One FL round; returns the stacked update matrix's agent id list.

    phase_times (optional dict): per-phase wall times (train/gather/
    aggregate) are measured with a device sync at each boundary and
    accumulated into it — only passed on snap rounds, so the syncs never
    touch steady-state throughput."""
    gm = world['global_model']
    agents = world['agents']

    def _mark(name, t0):
        if phase_times is None:
            return None
        if gm.device.type == 'cuda':
            torch.cuda.synchronize(gm.device)
        t1 = time.perf_counter()
        phase_times[name] = phase_times.get(name, 0.0) + (t1 - t0)
        return t1

    t0 = time.perf_counter() if phase_times is not None else None

    sampled = sample_agents(args.seed, rnd, args.num_agents, args.agent_frac)
    lo, hi, chunk = pdist.chunk_bounds(len(sampled), world_size, rank)
    mine = sampled[lo:hi]

    local = torch.zeros(chunk, gm.n_params, dtype=torch.float64,
                        device=gm.device)
    local_buf = (torch.zeros(chunk, gm.n_buffers, device=gm.device)
                 if gm.n_buffers else None)

    from .utils.tracing import trace_range
    use_streams = (gm.device.type == 'cuda' and len(mine) > 1
                   and args.agents_per_stream != 1)
    if use_streams:
        reps = _replica_pool(world, args, gm)
        R = len(reps)
        main = torch.cuda.current_stream(gm.device)
        for _, st in reps:
            st.wait_stream(main)
        for slot, agent_id in enumerate(mine):
            rep, st = reps[slot % R]
            with torch.cuda.stream(st):
                rep.flat_params.copy_(gm.flat_params)
                if rep.n_buffers:
                    rep.flat_buffers.copy_(gm.flat_buffers)
                update = agents[agent_id].local_train(rep, rnd=rnd)
                local[slot].copy_(update)
                if local_buf is not None:
                    local_buf[slot].copy_(agents[agent_id].buffer_delta)
        for _, st in reps:
            main.wait_stream(st)
    else:
        for slot, agent_id in enumerate(mine):
            update = agents[agent_id].local_train(gm, rnd=rnd)
            local[slot].copy_(update)
            if local_buf is not None:
                local_buf[slot].copy_(agents[agent_id].buffer_delta)

    t0 = _mark('train', t0)

    n_valid = []
    for r in range(world_size):
        rlo, rhi, _ = pdist.chunk_bounds(len(sampled), world_size, r)
        n_valid.append(rhi - rlo)
    with trace_range('gather_updates'):
        stacked = pdist.all_gather_updates(local, n_valid, chunk)
    t0 = _mark('gather', t0)
    with trace_range('aggregate'):
        world['aggregator'].aggregate_updates(gm, stacked, rnd,
                                              agent_ids=sampled)
    if local_buf is not None:
        buf_stacked = pdist.all_gather_updates(local_buf, n_valid, chunk)
        world['aggregator'].aggregate_buffers(gm, buf_stacked, sampled)
    _mark('aggregate', t0)
    return sampled


def run(args, writer=None, progress=False):
    """Full training loop; returns a metrics history dict."""
    rank, world_size = pdist.setup(args)
    if rank == 0:
        print_exp_details(args)
        if writer is None:
            writer = build_writer(args)

    world = build_world(args)
    gm = world['global_model']
    aggregator = world['aggregator']

    start_round = 0
    cum_poison_acc_mean = 0.0
    if args.resume:
        state = load_checkpoint(args.resume, gm, expect_args=args)
        start_round = state['round']
        cum_poison_acc_mean = state['cum_poison_acc_mean']

    history = {'round': [], 'val_acc': [], 'val_loss': [], 'poison_acc': [],
               'poison_loss': [], 'rounds_per_sec': []}

    rounds_iter = range(start_round + 1, args.rounds + 1)
    if progress and rank == 0:
        try:
            from tqdm import tqdm
            rounds_iter = tqdm(rounds_iter)
        except ImportError:
            pass

    shard = (rank, world_size) if world_size > 1 else None
    t_prev = time.perf_counter()
    for rnd in rounds_iter:
        phases = {} if rnd % args.snap == 0 else None
        run_round(args, world, rnd, rank, world_size, phase_times=phases)

        if rnd % args.snap == 0:
            t_now = time.perf_counter()
            rps = args.snap / (t_now - t_prev)
            with torch.no_grad():
                val_loss, (val_acc, val_pc) = get_loss_n_accuracy(
                    gm, world['X_val'], world['Y_val'], args, shard=shard)
                poison_loss, (poison_acc, _) = get_loss_n_accuracy(
                    gm, world['X_pv'], world['Y_pv'], args, shard=shard)
            phases['eval'] = time.perf_counter() - t_now
            t_prev = time.perf_counter()
            cum_poison_acc_mean += poison_acc
            history['round'].append(rnd)
            history['val_acc'].append(val_acc)
            history['val_loss'].append(val_loss)
            history['poison_acc'].append(poison_acc)
            history['poison_loss'].append(poison_loss)
            history['rounds_per_sec'].append(rps)
            if rank == 0:
                if writer:
                    writer.add_scalar('Validation/Loss', val_loss, rnd)
                    writer.add_scalar('Validation/Accuracy', val_acc, rnd)
                    writer.add_scalar('Poison/Base_Class_Accuracy',
                                      val_pc[args.base_class], rnd)
                    writer.add_scalar('Poison/Poison_Accuracy', poison_acc, rnd)
                    writer.add_scalar('Poison/Poison_Loss', poison_loss, rnd)
                    writer.add_scalar('Poison/Cumulative_Poison_Accuracy_Mean',
                                      cum_poison_acc_mean / rnd, rnd)
                    writer.add_scalar('Perf/Rounds_Per_Sec', rps, rnd)
                    # per-phase wall times, sampled on the snap round
                    # (train/gather/aggregate from run_round, plus eval)
                    for ph, secs in phases.items():
                        writer.add_scalar(f'Perf/{ph.capitalize()}_ms',
                                          secs * 1e3, rnd)
                print(f'| Val_Loss/Val_Acc: {val_loss:.3f} / {val_acc:.3f} |')
                print(f'| Val_Per_Class_Acc: {val_pc} ')
                print(f'| Poison Loss/Poison Acc: '
                      f'{poison_loss:.3f} / {poison_acc:.3f} |')
                if args.ckpt_dir:
                    os.makedirs(args.ckpt_dir, exist_ok=True)
                    save_checkpoint(
                        os.path.join(args.ckpt_dir, f'round_{rnd:06d}.pt'),
                        gm, rnd, args, cum_poison_acc_mean)

    if rank == 0:
        print('Training has finished!')
    history['final_params'] = gm.flat_params.detach().cpu()
    return history


def main(argv=None):
    args = args_parser(argv)
    try:
        run(args, progress=True)
    finally:
        pdist.teardown()


if __name__ == '__main__':
    main()
