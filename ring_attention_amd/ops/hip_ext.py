"""Loader for the in-tree CDNA4 HIP extension (_ring_attn_hip.so).

The extension is built IN-TREE (setup.py build_ext --inplace or
__graft_entry__.build()) for gfx950 only; there is no JIT fallback.  On a GPU
box a missing extension is a hard error — the HIP kernels ARE the compute
path, never a silent eager fallback.
"""

from __future__ import annotations


import threading

_ext = None
_tried = False
_lock = threading.Lock()


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    with _lock:        # threaded first use (loopback tests): import once
        if _tried:
            return _ext
        try:
            import importlib
            _ext = importlib.import_module("ring_attention_amd._ring_attn_hip")
        except ImportError:
            _ext = None
        _tried = True
    return _ext


def available() -> bool:
    return _load() is not None


def require():
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "ring_attention_amd HIP extension (_ring_attn_hip) is not built. "
            "Build it in-tree with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950). The HIP kernels are the only GPU "
            "compute path — there is no eager fallback on GPU.")
    return ext


def decode_partial(q, k, v, sm_scale=-1.0):
    """Single-query decode partial via the HIP kernel: (out fp32, lse fp32)."""
    return require().decode_partial(q, k, v, sm_scale)
