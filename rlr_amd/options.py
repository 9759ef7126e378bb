"""CLI — reproduces the reference's full 21-flag surface
(reference: src/options.py:4-74) plus infrastructure flags that the
MI355X build adds (--seed, --ckpt_dir, --resume, --synthetic, --dtype,
--agents_per_stream, --no_tb).

Notes kept behavior-compatible with the reference:
  * default base_class is 5 (options.py:40 — the README's claim of 1 is
    wrong; code wins).
  * the driver forces server_lr = 1 unless aggr == 'sign'
    (reference federated.py:23) — done in finalize_args().
"""

import argparse


def args_parser(argv=None):
    parser = argparse.ArgumentParser(description="rlr_amd federated trainer")

    # ---- reference flag surface (options.py:7-71) ----
    parser.add_argument('--data', type=str, default='fmnist',
                        help="dataset to train on: fmnist | cifar10 | fedemnist")
    parser.add_argument('--num_agents', type=int, default=10,
                        help="number of agents K")
    parser.add_argument('--agent_frac', type=float, default=1,
                        help="fraction of agents sampled per round C")
    parser.add_argument('--num_corrupt', type=int, default=0,
                        help="number of corrupt agents")
    parser.add_argument('--rounds', type=int, default=200,
                        help="number of communication rounds R")
    parser.add_argument('--aggr', type=str, default='avg',
                        choices=['avg', 'comed', 'sign'],
                        help="aggregation rule")
    parser.add_argument('--local_ep', type=int, default=2,
                        help="number of local epochs E")
    parser.add_argument('--bs', type=int, default=256,
                        help="local batch size B")
    parser.add_argument('--client_lr', type=float, default=0.1,
                        help="client learning rate")
    parser.add_argument('--client_moment', type=float, default=0.9,
                        help="client momentum")
    parser.add_argument('--server_lr', type=float, default=1,
                        help="server learning rate (signSGD)")
    parser.add_argument('--base_class', type=int, default=5,
                        help="base class for the backdoor attack")
    parser.add_argument('--target_class', type=int, default=7,
                        help="target class for the backdoor attack")
    parser.add_argument('--poison_frac', type=float, default=0.0,
                        help="fraction of a corrupt agent's base-class data to trojan")
    parser.add_argument('--pattern_type', type=str, default='plus',
                        help="trojan pattern: plus | square | copyright | apple")
    parser.add_argument('--robustLR_threshold', type=int, default=0,
                        help="RLR vote threshold theta (0 disables the defense)")
    parser.add_argument('--clip', type=float, default=0,
                        help="L2 ball radius for the per-batch PGD projection (0 disables)")
    parser.add_argument('--noise', type=float, default=0,
                        help="gaussian noise std multiplier (std = noise*clip)")
    parser.add_argument('--top_frac', type=int, default=100,
                        help="number of top-Fisher coords for sign-agreement diagnostics")
    parser.add_argument('--snap', type=int, default=1,
                        help="evaluate every snap rounds")
    parser.add_argument('--device', type=str, default=None,
                        help="device; default cuda:LOCAL_RANK if available else cpu")
    parser.add_argument('--num_workers', type=int, default=0,
                        help="dataloader workers (kept for CLI parity; the GPU path "
                             "keeps datasets resident in HBM and does not use workers)")

    # ---- MI355X build additions ----
    parser.add_argument('--seed', type=int, default=42,
                        help="master seed; all RNG streams derive from it "
                             "world-size-invariantly")
    parser.add_argument('--ckpt_dir', type=str, default='',
                        help="directory for checkpoints (empty = no checkpointing)")
    parser.add_argument('--resume', type=str, default='',
                        help="path to a checkpoint to resume from")
    parser.add_argument('--synthetic', action='store_true', default=False,
                        help="use deterministic synthetic data shaped like the real "
                             "dataset (no download needed; bench default)")
    parser.add_argument('--dtype', type=str, default='fp32',
                        choices=['fp32', 'bf16'],
                        help="client compute dtype (updates stay fp64 as in the "
                             "reference, agent.py:63)")
    parser.add_argument('--agents_per_stream', type=int, default=0,
                        help="train up to this many agents concurrently on separate "
                             "HIP streams per rank (0 = auto)")
    parser.add_argument('--model', type=str, default=None,
                        choices=[None, 'resnet18'],
                        help="override the dataset->model mapping "
                             "(build extension: ResNet18 w/ BatchNorm)")
    parser.add_argument('--no_tb', action='store_true', default=False,
                        help="disable the TensorBoard writer")
    parser.add_argument('--no_hip_graphs', dest='hip_graphs',
                        action='store_false', default=True,
                        help="disable hipGraph capture of the training step "
                             "(debug; results are bitwise identical either "
                             "way)")
    parser.add_argument('--log_dir', type=str, default='logs',
                        help="TensorBoard log root")

    args = parser.parse_args(argv)
    return finalize_args(args)


def finalize_args(args):
    """Derived rules. server_lr forced to 1 unless aggr=='sign'
    (reference federated.py:23)."""
    args.server_lr = args.server_lr if args.aggr == 'sign' else 1.0
    if args.device is None:
        import torch
        if torch.cuda.is_available():
            import os
            lr = int(os.environ.get('LOCAL_RANK', 0))
            args.device = f'cuda:{lr}'
        else:
            args.device = 'cpu'
    return args


def default_args(**overrides):
    """Programmatic args (tests, bench): the parser's defaults + overrides."""
    args = args_parser([])
    for k, v in overrides.items():
        if not hasattr(args, k):
            raise AttributeError(f"unknown option {k}")
        setattr(args, k, v)
    return finalize_args(args)
