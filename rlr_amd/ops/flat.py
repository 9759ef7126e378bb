"""Flat-buffer vector ops for the training hot loop.

The reference packs/unpacks parameters into vectors on every batch
(parameters_to_vector / vector_to_parameters, agent.py:35,56,60,63) — K10 in
SURVEY.md §2b.  Here parameters, grads and momentum live permanently in flat
device buffers (module params are views), so "pack" is a no-op and the whole
post-backward sequence

    clip_grad_norm_(params, 10); optimizer.step();           (agent.py:50-51)
    if clip>0: PGD-project the cumulative update             (agent.py:54-60)

is three fused HIP kernels on the GPU path (one reduction + one update for
the SGD step, one reduction + one axpby for the projection).  CPU path is
the plain-torch equivalent (GPU-less CI)."""

import torch

from . import ext, force_eager


def _gpu(t):
    return t.is_cuda and not force_eager()


def clipped_sgd_step_(params: torch.Tensor, grads: torch.Tensor,
                      momentum: torch.Tensor, lr: float, mu: float,
                      max_norm: float):
    """torch.nn.utils.clip_grad_norm_(max_norm) + SGD(momentum) step, fused.
    clip_coef = max_norm / (||g||2 + 1e-6), applied only when < 1 (torch
    semantics); v = mu*v + g_clipped; p -= lr*v."""
    if _gpu(params):
        ext().clipped_sgd_step(params, grads, momentum, lr, mu, max_norm)
        return
    total_norm = torch.linalg.vector_norm(grads)
    clip_coef = max_norm / (total_norm + 1e-6)
    scale = torch.clamp(clip_coef, max=1.0)
    # v = mu*v + g*scale ; p -= lr*v
    momentum.mul_(mu).add_(grads * scale)
    params.add_(momentum, alpha=-lr)


def pgd_project_(params: torch.Tensor, theta0: torch.Tensor, clip: float):
    """Project (params - theta0) onto the L2 ball of radius `clip`
    (reference agent.py:54-60): update /= max(1, ||update||/clip)."""
    if _gpu(params):
        ext().pgd_project(params, theta0, clip)
        return
    update = params - theta0
    denom = torch.clamp(torch.linalg.vector_norm(update) / clip, min=1.0)
    params.copy_(theta0 + update / denom)


def delta64(params: torch.Tensor, theta0_64: torch.Tensor) -> torch.Tensor:
    """fp64 update vector double(p) - theta0 (reference agent.py:62-64)."""
    if _gpu(params):
        return ext().delta64(params, theta0_64)
    return params.double() - theta0_64
