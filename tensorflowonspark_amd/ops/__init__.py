"""HIP/CDNA4 op extension loader.

The extension (``tfosr_hip_ops``) is built **in-tree** by ``__graft_entry__.build()``
(or ``python setup.py build_ext --inplace``) for gfx950 only — no CUDA paths, no
hipify output. On a GPU box, ops on CUDA tensors *require* the extension: a
missing .so raises instead of silently falling back to eager PyTorch (so a green
GPU test always means the HIP kernels ran). CPU tensors use plain PyTorch
reference implementations — those are also the numerics references the GPU tests
compare against.
"""

import glob
import logging
import os

import torch

logger = logging.getLogger(__name__)

#: must match TFOSR_API_VERSION in csrc/bindings.cpp
API_VERSION = 4

_ext = None
_ext_checked = False


def _find_ext_path():
    here = os.path.dirname(os.path.abspath(__file__))
    for pat in ("tfosr_hip_ops*.so",):
        hits = glob.glob(os.path.join(here, pat))
        if hits:
            return hits[0]
    return None


def get_ext(required=False):
    """Return the loaded HIP extension module, or None.

    required=True (the CUDA-tensor path) raises if the extension is absent —
    GPU execution must never silently fall back to eager PyTorch.
    """
    global _ext, _ext_checked
    if not _ext_checked:
        _ext_checked = True
        path = _find_ext_path()
        if path is not None:
            try:
                import importlib.util
                spec = importlib.util.spec_from_file_location("tfosr_hip_ops", path)
                mod = importlib.util.module_from_spec(spec)
                spec.loader.exec_module(mod)
                got = mod.api_version() if hasattr(mod, "api_version") else 0
                if got != API_VERSION:
                    raise RuntimeError(
                        "tfosr_hip_ops.so is stale (api {} != {}): rebuild "
                        "with `python __graft_entry__.py`".format(
                            got, API_VERSION))
                _ext = mod
                logger.info("loaded HIP ops extension: %s", path)
            except Exception as e:
                logger.error("failed to load HIP ops extension %s: %s", path, e)
                _ext = None
    if required and _ext is None:
        if os.environ.get("TFOS_ALLOW_EAGER_FALLBACK"):
            return None
        raise RuntimeError(
            "tfosr_hip_ops extension not built but a CUDA tensor hit a fused op. "
            "Run __graft_entry__.build() (hipcc --offload-arch=gfx950) first, or "
            "set TFOS_ALLOW_EAGER_FALLBACK=1 to debug with eager PyTorch.")
    return _ext


def hip_available(x=None):
    """True when the fast path applies: CUDA tensor + extension present."""
    if x is not None and not x.is_cuda:
        return False
    if not torch.cuda.is_available():
        return False
    return get_ext(required=True) is not None
