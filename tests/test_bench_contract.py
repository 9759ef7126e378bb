"""Driver-contract guard: `python bench.py` must emit ONE JSON line with
the agreed fields (the round driver parses this output verbatim).  Runs
the real benchmark on CPU at minimum size."""

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parents[1]


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, str(REPO / 'bench.py'), '--steps', '1',
         '--warmup', '0', '--agents_per_gpu', '1'],
        capture_output=True, text=True, timeout=600, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines()
                  if l.startswith('{') and l.endswith('}')]
    assert len(json_lines) == 1, out.stdout[-2000:]
    d = json.loads(json_lines[0])
    for field in ('metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
                  'ms_per_step', 'higher_is_better', 'scaling',
                  'vs_baseline', 'dtype', 'data', 'config'):
        assert field in d, field
    assert d['metric'] == 'fl_rounds_per_sec'
    assert d['n_gpus'] == 1 and d['steps'] == 1 and d['warmup'] == 0
    assert d['higher_is_better'] is True and d['scaling'] == 'weak'
    assert d['dtype'] == 'fp32' and d['data'] == 'synthetic'
    assert d['value'] > 0 and d['ms_per_step'] > 0
    cfg = d['config']
    assert cfg['model'] and 'global_batch' in cfg and 'parallelism' in cfg
