"""openembedding_amd.ops — hot-path kernels.

Two backends:
  - torch: pure-torch fallback used on CPU (tests + oracle);
  - hip:   the in-tree CDNA4 extension (_embops.so, built from ops/csrc by
           setup.py / __graft_entry__.build for gfx950).

On a CUDA/ROCm device the HIP extension is REQUIRED: ops raise if it is not
importable, rather than silently running the slow fallback (so a GPU test
that passes is guaranteed to have run the native kernels).
Set OEAMD_ALLOW_TORCH_FALLBACK=1 to override for debugging only.
"""

from __future__ import annotations

import os

_ext = None
_ext_err: Exception | None = None


def _load_ext():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        import torch  # noqa: F401  (loads libc10/libtorch for the ext)
        from . import _embops  # built in-tree by setup.py build_ext --inplace
        _ext = _embops
    except Exception as e:  # noqa: BLE001
        _ext_err = e
    return _ext


def hip_available() -> bool:
    return _load_ext() is not None


def require_hip():
    ext = _load_ext()
    if ext is None:
        if os.environ.get("OEAMD_ALLOW_TORCH_FALLBACK") == "1":
            return None
        raise RuntimeError(
            "openembedding_amd HIP extension (_embops) is not built, but a "
            "ROCm device op was requested. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error was: {_ext_err!r}")
    return ext
