#!/usr/bin/env python3
"""Asserted defense-efficacy CI at full reference scale.

Runs the reference's FMNIST triple (runner.sh:12-18 semantics: 10 agents,
200 rounds, bs 256, 2 local epochs) on synthetic data through the HIP
path and ASSERTS tight bands from the round-1 measured curves
(BASELINE.md): backdoor accuracy saturates >=0.95 without a defense and
is crushed <=0.05 under RLR theta=4, while the two val accuracies stay
within 0.05 of each other.  Writes gpurun_out/defense_ci.json — copy into
profiles/ for the tracked record.
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))


def run_cfg(**over):
    from rlr_amd.federated import run
    from rlr_amd.options import default_args
    base = dict(data='fmnist', num_agents=10, rounds=200, snap=200,
                local_ep=2, bs=256, synthetic=True, no_tb=True,
                device='cuda:0')
    base.update(over)
    return run(default_args(**base))


def main():
    atk = run_cfg(num_corrupt=1, poison_frac=0.5)
    rlr = run_cfg(num_corrupt=1, poison_frac=0.5, robustLR_threshold=4)

    result = {
        'config': 'fmnist 10 agents, 200 rounds, bs 256, local_ep 2, '
                  'synthetic data (reference runner.sh:12-18 scale)',
        'attack_no_defense': {'val_acc': atk['val_acc'][-1],
                              'poison_acc': atk['poison_acc'][-1]},
        'attack_rlr_theta4': {'val_acc': rlr['val_acc'][-1],
                              'poison_acc': rlr['poison_acc'][-1]},
    }
    os.makedirs('gpurun_out', exist_ok=True)
    with open('gpurun_out/defense_ci.json', 'w') as f:
        json.dump(result, f, indent=2)
    print(json.dumps(result, indent=2))

    assert atk['poison_acc'][-1] >= 0.95, \
        f"attack without defense should saturate: {atk['poison_acc'][-1]}"
    assert rlr['poison_acc'][-1] <= 0.05, \
        f"RLR theta=4 should crush the backdoor: {rlr['poison_acc'][-1]}"
    assert abs(atk['val_acc'][-1] - rlr['val_acc'][-1]) <= 0.05, \
        (atk['val_acc'][-1], rlr['val_acc'][-1])
    print('DEFENSE CI PASS')


if __name__ == '__main__':
    main()
