"""Trojan injection (reference utils.py:160-284 semantics).

The pattern geometry is expressed as data — a (P,2) coordinate table plus a
write mode — shared by the CPU implementation (vectorized torch writes, no
per-pixel python loop) and the on-device HIP kernel (ops/csrc/poison.hip),
so poisoned batches never leave HBM on the GPU path (BASELINE.json).

Pattern coordinates reproduce the reference exactly:
  * cifar10 'plus' at (5,5), size 6: full plus for agent_idx==-1, or the
    4-way Distributed-Backdoor-Attack split by agent_idx % 4
    (utils.py:188-224); value 0 in all 3 channels.
  * fmnist 'square': rows/cols 21..25 = 255 (utils.py:227-230);
    'plus' at (5,5) size 5 = 255 (utils.py:244-253).
  * fedemnist: same shapes with value 0 / start 8 (utils.py:255-282).
  * 'copyright'/'apple': the reference loads PNG watermarks with OpenCV,
    inverts and resizes them, then adds (uint8, wrapping) / subtracts
    (normalized) them (utils.py:232-242,261-271).  This build generates the
    watermark masks procedurally (no binary assets are copied from the
    reference); the add-with-uint8-wraparound arithmetic is preserved.
"""

import math

import numpy as np
import torch

from ..utils.rng import np_rng


# ------------------------------------------------------------ watermark gen

def _watermark_mask(name: str) -> np.ndarray:
    """Procedural 28x28 uint8 watermark (already 'inverted': pattern bright
    on black, like the reference's bitwise_not'ed PNGs)."""
    yy, xx = np.mgrid[0:28, 0:28].astype(np.float64)
    cy, cx = 13.5, 13.5
    r = np.sqrt((yy - cy) ** 2 + (xx - cx) ** 2)
    m = np.zeros((28, 28), dtype=np.float64)
    if name == 'copyright':
        # ring + open 'C' arc
        m[(r > 9) & (r < 12)] = 255
        ang = np.arctan2(yy - cy, xx - cx)
        c_arc = (r > 4) & (r < 7) & (np.abs(ang) > 0.7)
        m[c_arc] = 255
    elif name == 'apple':
        # filled disc with a bite removed and a stem
        m[r < 10] = 255
        bite = np.sqrt((yy - 8) ** 2 + (xx - 22) ** 2) < 5
        m[bite] = 0
        m[3:8, 13:15] = 255
    else:
        raise ValueError(name)
    return m.astype(np.uint8)


# ------------------------------------------------------------- pattern spec

class PatternSpec:
    """mode: 'set'   -> raw[coords] = value (all channels for HWC)
             'add_wrap_u8' -> raw = uint8(raw + mask)   (wrapping, ref parity)
             'sub_float'   -> raw = raw - mask/255."""

    def __init__(self, mode, coords=None, value=None, mask=None):
        self.mode = mode
        self.coords = coords  # np.ndarray (P,2) int64 or None
        self.value = value
        self.mask = mask      # np.ndarray (28,28) uint8 or None


def _plus_coords(start, size, agent_idx, quarters):
    """Full plus or one DBA quarter; mirrors utils.py:188-224/244-253."""
    rows = []
    if not quarters or agent_idx == -1:
        rows += [(i, start) for i in range(start, start + size + 1)]
        mid = start + size // 2
        rows += [(mid, i) for i in range(start - size // 2, start + size // 2 + 1)]
    else:
        q = agent_idx % 4
        mid = start + size // 2
        if q == 0:
            rows += [(i, start) for i in range(start, start + size // 2 + 1)]
        elif q == 1:
            rows += [(i, start) for i in range(start + size // 2 + 1, start + size + 1)]
        elif q == 2:
            rows += [(mid, i) for i in range(start - size // 2, start + size // 4 + 1)]
        else:
            rows += [(mid, i) for i in range(start - size // 4 + 1, start + size // 2 + 1)]
    return np.array(rows, dtype=np.int64)


def pattern_spec(data: str, pattern_type: str, agent_idx: int = -1) -> PatternSpec:
    if data == 'cifar10':
        if pattern_type != 'plus':
            raise ValueError("cifar10 supports only the 'plus' (DBA) pattern "
                             "(reference utils.py:187)")
        # DBA split active for agent_idx >= 0 (reference utils.py:196)
        coords = _plus_coords(5, 6, agent_idx, quarters=(agent_idx != -1))
        return PatternSpec('set', coords=coords, value=0)
    if data == 'fmnist':
        if pattern_type == 'square':
            rs, cs = np.mgrid[21:26, 21:26]
            return PatternSpec('set',
                               coords=np.stack([rs.ravel(), cs.ravel()], 1),
                               value=255)
        if pattern_type == 'plus':
            # fmnist plus is NOT DBA-split; range(start, start+size) for the
            # vertical bar (utils.py:247-253 — note: size, not size+1)
            rows = [(i, 5) for i in range(5, 10)]
            rows += [(7, i) for i in range(3, 8)]
            return PatternSpec('set', coords=np.array(rows, dtype=np.int64),
                               value=255)
        if pattern_type in ('copyright', 'apple'):
            return PatternSpec('add_wrap_u8', mask=_watermark_mask(pattern_type))
        raise ValueError(pattern_type)
    if data == 'fedemnist':
        if pattern_type == 'square':
            rs, cs = np.mgrid[21:26, 21:26]
            return PatternSpec('set',
                               coords=np.stack([rs.ravel(), cs.ravel()], 1),
                               value=0.0)
        if pattern_type == 'plus':
            rows = [(i, 8) for i in range(8, 13)]
            rows += [(10, i) for i in range(6, 11)]
            return PatternSpec('set', coords=np.array(rows, dtype=np.int64),
                               value=0.0)
        if pattern_type in ('copyright', 'apple'):
            return PatternSpec('sub_float', mask=_watermark_mask(pattern_type))
        raise ValueError(pattern_type)
    raise ValueError(data)


# ---------------------------------------------------------------- appliers

def apply_pattern_(raw: torch.Tensor, spec: PatternSpec):
    """In-place pattern write on a batch of raw images.
    raw: (B,H,W) uint8 | (B,H,W,C) uint8 | (B,1,H,W)/(B,H,W) float."""
    if spec.mode == 'set':
        r = torch.as_tensor(spec.coords[:, 0], device=raw.device)
        c = torch.as_tensor(spec.coords[:, 1], device=raw.device)
        if raw.dim() == 4 and raw.shape[-1] == 3:          # (B,H,W,C)
            raw[:, r, c, :] = spec.value
        elif raw.dim() == 4:                               # (B,1,H,W)
            raw[:, :, r, c] = spec.value
        else:                                              # (B,H,W)
            raw[:, r, c] = spec.value
    elif spec.mode == 'add_wrap_u8':
        m = torch.as_tensor(spec.mask, device=raw.device)
        view = raw if raw.dim() == 3 else raw.squeeze(1)
        view += m                                          # uint8 wraps
    elif spec.mode == 'sub_float':
        m = torch.as_tensor(spec.mask, device=raw.device).float() / 255.0
        view = raw if raw.dim() == 3 else raw.squeeze(1)
        view -= m
    else:
        raise ValueError(spec.mode)


def apply_pattern_indexed_(storage: torch.Tensor, idx_t: torch.Tensor,
                           spec: PatternSpec):
    """Apply a pattern to storage[idx] in place.  On CUDA(HIP) tensors this
    is the on-device trojan kernel (ops/csrc/poison.hip) — raw data stays in
    HBM; the CPU path gathers/patches/scatters with torch ops."""
    if storage.is_cuda:
        from ..ops import ext
        if spec.mode == 'set':
            coords = torch.as_tensor(spec.coords, dtype=torch.int32,
                                     device=storage.device).contiguous()
            if storage.dtype == torch.uint8:
                ext().poison_set_u8(storage, idx_t, coords, int(spec.value))
            else:
                ext().poison_set_f32(storage, idx_t, coords,
                                     float(spec.value))
        elif spec.mode == 'add_wrap_u8':
            mask = torch.as_tensor(spec.mask, device=storage.device)
            ext().poison_addwrap_u8(storage, idx_t, mask)
        else:  # sub_float
            mask = torch.as_tensor(spec.mask, device=storage.device)
            ext().poison_subf(storage, idx_t, mask)
        return
    batch = storage[idx_t]
    apply_pattern_(batch, spec)
    storage[idx_t] = batch


def poison_dataset(dataset, args, data_idxs=None, poison_all=False,
                   agent_idx=-1):
    """Reference utils.py:160-178 semantics with a deterministic,
    world-size-invariant RNG ('poison' stream keyed by agent_idx).

    Writes the pattern into the dataset's RAW storage (uint8 `.data` for
    fmnist/cifar10, normalized `.inputs` for fedemnist) and relabels the
    poisoned samples to args.target_class."""
    all_idxs = (dataset.targets == args.base_class).nonzero().flatten().tolist()
    if data_idxs is not None:
        all_idxs = sorted(set(all_idxs).intersection(set(data_idxs)))

    poison_frac = 1.0 if poison_all else args.poison_frac
    n_poison = math.floor(poison_frac * len(all_idxs))
    rng = np_rng(args.seed, 'poison', agent_idx)
    poison_idxs = rng.permutation(np.array(all_idxs, dtype=np.int64))[:n_poison] \
        if all_idxs else np.array([], dtype=np.int64)

    if len(poison_idxs) == 0:
        return []

    spec = pattern_spec(args.data, args.pattern_type, agent_idx)
    storage = dataset.inputs if hasattr(dataset, 'inputs') else dataset.data
    idx_t = torch.as_tensor(poison_idxs, device=storage.device)
    apply_pattern_indexed_(storage, idx_t, spec)
    dataset.targets[idx_t.to(dataset.targets.device)] = args.target_class
    return poison_idxs.tolist()
