"""Python surface of the hand-written gfx950 HIP ops.

The extension `_autodist_hip` is built IN-TREE (ops/build.py; .so lands next
to this file) so it travels to GPU boxes with the repo snapshot. On a GPU
box the HIP path is MANDATORY: if CUDA devices are visible and the extension
is missing, importing ops raises — a silent eager fallback would invalidate
every benchmark (see repo rules).

On CPU-only machines (CI) the torch fallbacks are used silently; every
kernel has a numerics test comparing HIP vs the plain fp32 torch reference
(tests/test_gpu_kernels.py, @pytest.mark.gpu).
"""
import os

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        import importlib.util
        here = os.path.dirname(os.path.abspath(__file__))
        sos = [f for f in os.listdir(here)
               if f.startswith("_autodist_hip") and f.endswith(".so")]
        if sos:
            spec = importlib.util.spec_from_file_location(
                "_autodist_hip", os.path.join(here, sos[0]))
            mod = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(mod)
            _ext = mod
    except Exception as exc:  # noqa: BLE001
        _ext = None
        if torch.cuda.is_available():
            raise RuntimeError(
                f"GPU visible but _autodist_hip extension failed to load: {exc}. "
                "Run `python -m autodist_amd.ops.build` first.") from exc
    if _ext is None and torch.cuda.is_available():
        raise RuntimeError(
            "GPU visible but the _autodist_hip extension is not built. "
            "Run `python -m autodist_amd.ops.build` (refusing to fall back "
            "to eager on a GPU box).")
    return _ext


def has_gpu_ops() -> bool:
    if not torch.cuda.is_available():
        return False
    return _load() is not None


def ext():
    e = _load()
    if e is None:
        raise RuntimeError("_autodist_hip extension not available")
    return e


# -- kernel wrappers (HIP on GPU, torch fallback on CPU) --------------------

def segment_coalesce(indices: torch.Tensor, values: torch.Tensor):
    """Sort-by-index + segment-sum of duplicate rows (sparse accumulator)."""
    if indices.is_cuda and has_gpu_ops():
        return ext().segment_coalesce(indices, values)
    uniq, inv = torch.unique(indices, sorted=True, return_inverse=True)
    out = torch.zeros((uniq.shape[0],) + tuple(values.shape[1:]),
                      dtype=values.dtype, device=values.device)
    out.index_add_(0, inv, values)
    return uniq, out
