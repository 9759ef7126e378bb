"""TensorBoard + console logging (reference federated.py:27-31,
utils.py:287-303).  Scalar names and the run-name format are kept
reference-identical so existing dashboards read both; the build adds
Perf/* scalars (rounds/sec, phase timings)."""

from time import ctime


def run_name(args):
    """Reference federated.py:27-30 run-name string."""
    return (f"time:{ctime()}-clip_val:{args.clip}-noise_std:{args.noise}"
            f"-aggr:{args.aggr}-s_lr:{args.server_lr}-num_cor:{args.num_corrupt}"
            f"thrs_robustLR:{args.robustLR_threshold}"
            f"-num_corrupt:{args.num_corrupt}-pttrn:{args.pattern_type}")


def build_writer(args):
    if getattr(args, 'no_tb', False):
        return None
    try:
        from torch.utils.tensorboard import SummaryWriter
    except ImportError:
        return None
    return SummaryWriter(f"{args.log_dir}/{run_name(args)}")


def print_exp_details(args):
    """Reference utils.py:287-303."""
    print('======================================')
    print(f'    Dataset: {args.data}')
    print(f'    Global Rounds: {args.rounds}')
    print(f'    Aggregation Function: {args.aggr}')
    print(f'    Number of agents: {args.num_agents}')
    print(f'    Fraction of agents: {args.agent_frac}')
    print(f'    Batch size: {args.bs}')
    print(f'    Client_LR: {args.client_lr}')
    print(f'    Server_LR: {args.server_lr}')
    print(f'    Client_Momentum: {args.client_moment}')
    print(f'    RobustLR_threshold: {args.robustLR_threshold}')
    print(f'    Noise Ratio: {args.noise}')
    print(f'    Number of corrupt agents: {args.num_corrupt}')
    print(f'    Poison Frac: {args.poison_frac}')
    print(f'    Clip: {args.clip}')
    print('======================================')
