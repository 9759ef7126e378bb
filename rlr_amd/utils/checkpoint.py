"""Checkpoint / resume — absent from the reference (nothing is ever saved,
federated.py:95), defined by this build:

one file per snap round containing the flat fp32 parameter vector (+ flat
float buffers for BN models), the round index, the full arg namespace, the
master seed, and cum_poison_acc_mean.  Every RNG stream is derived
statelessly from (seed, purpose, round/agent/...), so no RNG state needs
saving and a checkpoint written at any world size resumes at any other
world size bit-identically."""

import os

import torch

FORMAT_VERSION = 1


def save_checkpoint(path, global_model, rnd, args, cum_poison_acc_mean):
    state = {
        'version': FORMAT_VERSION,
        'round': rnd,
        'args': {k: v for k, v in vars(args).items() if k != 'device'},
        'seed': args.seed,
        'params': global_model.flat_params.detach().float().cpu(),
        'cum_poison_acc_mean': float(cum_poison_acc_mean),
    }
    if global_model.n_buffers:
        state['buffers'] = global_model.flat_buffers.detach().float().cpu()
    tmp = path + '.tmp'
    torch.save(state, tmp)
    os.replace(tmp, path)


def load_checkpoint(path, global_model=None, expect_args=None):
    """Load and (optionally) restore into `global_model`.  Refuses
    checkpoints whose model/dataset/param-count don't match the run being
    resumed — a silent mismatch would corrupt the weights at the copy."""
    state = torch.load(path, map_location='cpu', weights_only=False)
    assert state.get('version') == FORMAT_VERSION, \
        f"unknown checkpoint version {state.get('version')}"
    if expect_args is not None:
        saved = state.get('args', {})
        for key in ('data', 'model'):
            want = getattr(expect_args, key, None)
            got = saved.get(key)
            if got is not None and want is not None and got != want:
                raise ValueError(
                    f"checkpoint {path!r} was written for {key}={got!r}; "
                    f"this run has {key}={want!r} — refusing to resume")
    if global_model is not None:
        if state['params'].numel() != global_model.n_params:
            raise ValueError(
                f"checkpoint {path!r} holds {state['params'].numel()} params "
                f"but the model has {global_model.n_params} — wrong "
                f"model/dataset for this checkpoint")
        global_model.load_vector(
            state['params'].to(global_model.flat_params.device))
        if global_model.n_buffers and 'buffers' in state:
            global_model.flat_buffers.copy_(
                state['buffers'].to(global_model.flat_buffers.device))
    return state
