"""Op layer: hand-written HIP/CDNA4 kernels on the GPU path, plain PyTorch
only as the GPU-less CI fallback.

`ext()` loads the in-tree extension `rlr_amd/ops/_hip.so` (built by
`python -m rlr_amd.ops.build` / __graft_entry__.build()).  On a CUDA(HIP)
device the op layer REQUIRES the extension — a missing .so raises
immediately instead of silently falling back to eager (the round-end GPU
check records which native libraries were actually loaded)."""

import os

_EXT = None
_EXT_ERR = None


def ext():
    """The compiled HIP extension module; raises if unavailable."""
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            import torch  # noqa: F401  (loads libtorch first)
            from . import _hip  # type: ignore
            _EXT = _hip
        except ImportError as e:  # pragma: no cover
            _EXT_ERR = e
    if _EXT is None:
        raise RuntimeError(
            "rlr_amd HIP extension not built — run `python -m rlr_amd.ops.build` "
            f"(import error: {_EXT_ERR})")
    return _EXT


def have_ext() -> bool:
    try:
        ext()
        return True
    except RuntimeError:
        return False


def force_eager() -> bool:
    """Debug-only escape hatch; never set in production or benches."""
    return os.environ.get('RLR_AMD_FORCE_EAGER', '0') == '1'
