"""Loader for the in-tree HIP extension (`relora_amd/ops/_relora_hip*.so`).

The extension is built IN-TREE (see `relora_amd/ops/build.py` /
`python setup.py build_ext --inplace`) so the .so ships with the repo snapshot
to GPU boxes. Policy: on a ROCm GPU the HIP kernels are
the compute path — if a tensor is on CUDA(=ROCm) and the extension is not
importable we raise, unless RELORA_AMD_ALLOW_FALLBACK=1 is set (debug only).
"""

import importlib
import os

_ext = None
_ext_err = None


def _try_load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return
    try:
        _ext = importlib.import_module("relora_amd.ops._relora_hip")
    except ImportError as e:
        _ext_err = e


def ext():
    """Return the loaded extension module or None."""
    _try_load()
    return _ext


def ext_or_raise():
    _try_load()
    if _ext is None:
        raise RuntimeError(
            "relora_amd HIP extension is not built but a tensor is on a ROCm GPU. "
            "Build it in-tree with `python -m relora_amd.ops.build` "
            "(or set RELORA_AMD_ALLOW_FALLBACK=1 to run the slow PyTorch fallback; "
            f"import error: {_ext_err})"
        )
    return _ext


def allow_fallback():
    return os.environ.get("RELORA_AMD_ALLOW_FALLBACK", "0") == "1"


_disabled_ops = None


def _op_disabled(op):
    """RELORA_AMD_DISABLE_OPS='attention,rmsnorm,...' routes the named ops
    through the torch composition on GPU — an explicit per-kernel A/B
    switch for measurement, never a silent fallback (the env must be set
    deliberately).  Parsed once per process."""
    global _disabled_ops
    if _disabled_ops is None:
        raw = os.environ.get("RELORA_AMD_DISABLE_OPS", "")
        _disabled_ops = {s.strip() for s in raw.split(",") if s.strip()}
    return op is not None and op in _disabled_ops


def use_hip(t, op=None):
    """True if op dispatch should take the HIP kernel path for tensor `t`.

    Raises when `t` is on GPU and the extension is missing (fail loudly —
    a silent eager fallback on the GPU box would invalidate benchmarks).
    `op` names the kernel for the RELORA_AMD_DISABLE_OPS A/B switch.
    """
    if not t.is_cuda:
        return False
    if _op_disabled(op):
        return False
    _try_load()
    if _ext is not None:
        return True
    if allow_fallback():
        return False
    ext_or_raise()
