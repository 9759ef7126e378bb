"""CLI parity tests (reference src/options.py:1-74) and a parse guard for
every canned experiment in scripts/runner.sh."""

import re
from pathlib import Path

from rlr_amd.options import args_parser, default_args, finalize_args

REPO = Path(__file__).resolve().parents[1]

# the reference's 21 flags
REFERENCE_FLAGS = [
    'data', 'num_agents', 'agent_frac', 'num_corrupt', 'rounds', 'aggr',
    'local_ep', 'bs', 'client_lr', 'client_moment', 'server_lr',
    'base_class', 'target_class', 'poison_frac', 'pattern_type',
    'robustLR_threshold', 'clip', 'noise', 'top_frac', 'snap', 'device',
]


def test_all_reference_flags_exist():
    args = args_parser([])
    for name in REFERENCE_FLAGS:
        assert hasattr(args, name), f"missing reference flag --{name}"


def test_reference_defaults():
    """Defaults match reference options.py:7-66 (device aside, which the
    reference resolves at parse time the same way)."""
    a = args_parser([])
    assert a.data == 'fmnist' and a.num_agents == 10
    assert a.agent_frac == 1.0 and a.num_corrupt == 0
    assert a.rounds == 200 and a.aggr == 'avg'
    assert a.local_ep == 2 and a.bs == 256
    assert a.client_lr == 0.1 and a.client_moment == 0.9
    assert a.server_lr == 1.0
    assert a.base_class == 5 and a.target_class == 7
    assert a.poison_frac == 0.0 and a.pattern_type == 'plus'
    assert a.robustLR_threshold == 0 and a.clip == 0
    assert a.noise == 0 and a.top_frac == 100 and a.snap == 1


def test_server_lr_rule():
    """federated.py:23 — server_lr is forced to 1 unless aggr == 'sign'."""
    a = finalize_args(args_parser(['--server_lr', '100']))
    assert a.server_lr == 1
    a = finalize_args(args_parser(['--server_lr', '100', '--aggr', 'sign']))
    assert a.server_lr == 100
    assert default_args(server_lr=50).server_lr == 1
    assert default_args(server_lr=50, aggr='sign').server_lr == 50


def _runner_commands():
    """Extract each `$PY ... $SYN` invocation from scripts/runner.sh,
    following backslash continuations."""
    lines = (REPO / 'scripts' / 'runner.sh').read_text().splitlines()
    cmds, cur = [], None
    for ln in lines:
        s = ln.strip()
        if cur is not None:
            cur += ' ' + s.rstrip('\\').strip()
            if not s.endswith('\\'):
                cmds.append(cur)
                cur = None
        elif s.startswith('$PY '):
            body = s[4:].rstrip('\\').strip()
            if s.endswith('\\'):
                cur = body
            else:
                cmds.append(body)
    return [c.replace('$SYN', '--synthetic') for c in cmds]


def test_runner_experiments_parse():
    cmds = _runner_commands()
    assert len(cmds) == 9, cmds  # 3 datasets x {none, attack, attack+RLR}
    parsed = []
    for cmd in cmds:
        argv = [t for t in re.split(r'\s+', cmd) if t]
        a = finalize_args(args_parser(argv))
        parsed.append(a)
    # spot-check the reference's scales (runner.sh:12-38)
    fm, cf, fe = parsed[2], parsed[5], parsed[8]
    assert (fm.data, fm.num_agents, fm.robustLR_threshold) == ('fmnist', 10, 4)
    assert (cf.data, cf.num_agents, cf.num_corrupt,
            cf.robustLR_threshold) == ('cifar10', 40, 4, 8)
    assert (fe.data, fe.num_agents, fe.agent_frac, fe.num_corrupt,
            fe.local_ep, fe.bs) == ('fedemnist', 3383, 0.01, 338, 10, 64)
    assert all(a.synthetic for a in parsed)
