"""ops — compute kernels for the SAC hot path.

Two tiers:

- :mod:`.torch_ref` — pure-torch fp32 implementations (CPU path + oracle).
- ``_hip_ops`` — the in-tree HIP/CDNA4 extension (gfx950).  On a GPU box the
  native extension is REQUIRED: GPU execution with the extension missing
  raises rather than silently falling back to eager torch.

Use :func:`native` to get the extension module, :func:`has_native` to probe.
"""

from __future__ import annotations

import importlib
import os

_native = None
_native_err: Exception | None = None


def _try_load():
    global _native, _native_err
    if _native is not None or _native_err is not None:
        return
    try:
        import torch  # noqa: F401 — extension links against torch's libs
        _native = importlib.import_module("distributed_sac_amd.ops._hip_ops")
    except Exception as e:  # pragma: no cover - depends on build state
        _native_err = e


def has_native() -> bool:
    _try_load()
    return _native is not None


def native():
    """Return the HIP extension module, raising loudly if unavailable."""
    _try_load()
    if _native is None:
        raise RuntimeError(
            "distributed_sac_amd HIP extension (_hip_ops) is not built/loadable. "
            "Build it in-tree with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Underlying error: {_native_err!r}")
    return _native


USE_NATIVE_ENV = "DSAC_DISABLE_NATIVE"


def native_enabled() -> bool:
    """Native kernels are mandatory on GPU unless explicitly disabled for
    debugging via DSAC_DISABLE_NATIVE=1."""
    return os.environ.get(USE_NATIVE_ENV, "0") != "1"
