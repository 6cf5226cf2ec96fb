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

_FUSED_CLASSES = ("SGD", "Adam", "AdamW", "Adagrad", "RMSprop")


def has_fused(cls_name: str) -> bool:
    return cls_name in _FUSED_CLASSES


def fused_apply(cls_name: str, param: torch.Tensor, grad: torch.Tensor,
                state: dict, hyper: dict) -> bool:
    """One-kernel optimizer update over flat bucket buffers. Returns False
    when this optimizer class has no fused kernel (caller falls back to the
    torch implementation)."""
    if not (param.is_cuda and has_gpu_ops()):
        return False
    e = ext()
    if cls_name == "SGD":
        momentum = float(hyper.get("momentum", 0.0))
        buf = None
        first = False
        if momentum != 0.0:
            if "momentum_buffer" not in state:
                state["momentum_buffer"] = torch.empty_like(param)
                first = True
            buf = state["momentum_buffer"]
        e.fused_sgd(param, grad, buf, float(hyper["lr"]), momentum,
                    float(hyper.get("dampening", 0.0)),
                    float(hyper.get("weight_decay", 0.0)),
                    bool(hyper.get("nesterov", False)), first,
                    bool(hyper.get("maximize", False)))
        return True
    if cls_name in ("Adam", "AdamW"):
        state["step"] += 1
        step = float(state["step"])
        beta1, beta2 = hyper.get("betas", (0.9, 0.999))
        bc1 = 1.0 - beta1 ** step
        sqrt_bc2 = (1.0 - beta2 ** step) ** 0.5
        wd_default = 0.01 if cls_name == "AdamW" else 0.0
        e.fused_adam(param, grad, state["exp_avg"], state["exp_avg_sq"],
                     float(hyper["lr"]), float(beta1), float(beta2),
                     float(hyper.get("eps", 1e-8)),
                     float(hyper.get("weight_decay", wd_default)),
                     cls_name == "AdamW", bc1, sqrt_bc2,
                     bool(hyper.get("maximize", False)))
        return True
    if cls_name == "Adagrad":
        state["step"] += 1
        clr = float(hyper["lr"]) / (
            1.0 + (float(state["step"]) - 1.0)
            * float(hyper.get("lr_decay", 0.0)))
        e.fused_adagrad(param, grad, state["sum"], clr,
                        float(hyper.get("eps", 1e-10)),
                        float(hyper.get("weight_decay", 0.0)),
                        bool(hyper.get("maximize", False)))
        return True
    if cls_name == "RMSprop":
        e.fused_rmsprop(param, grad, state["square_avg"],
                        state.get("grad_avg"), state.get("momentum_buffer"),
                        float(hyper["lr"]), float(hyper.get("alpha", 0.99)),
                        float(hyper.get("eps", 1e-8)),
                        float(hyper.get("weight_decay", 0.0)),
                        float(hyper.get("momentum", 0.0)),
                        bool(hyper.get("maximize", False)))
        return True
    return False


def scale_cast_bf16(inp: torch.Tensor, out: torch.Tensor, scale: float):
    if inp.is_cuda and has_gpu_ops():
        ext().scale_cast_bf16(inp, out, float(scale))
    else:
        out.copy_(inp if scale == 1.0 else inp * scale)


def cast_back_f32(wire: torch.Tensor, out: torch.Tensor):
    if wire.is_cuda and has_gpu_ops():
        ext().cast_back_f32(wire, out)
    else:
        out.copy_(wire)


def ef_compress(flat: torch.Tensor, err: torch.Tensor, wire: torch.Tensor,
                scale: float):
    """flat += err; wire = bf16(flat*scale); err = flat - fp32(wire)/scale."""
    if flat.is_cuda and has_gpu_ops():
        ext().ef_compress(flat, err, wire, float(scale))
    else:
        flat.add_(err)
        torch.mul(flat, scale, out=err)       # reuse err as tmp
        wire.copy_(err)
        err.copy_(flat).sub_(wire.to(torch.float32) / scale)


def segment_coalesce(indices: torch.Tensor, values: torch.Tensor):
    """Sort-by-index + segment-sum of duplicate rows (sparse accumulator)."""
    if indices.is_cuda and has_gpu_ops():
        return ext().segment_coalesce(indices, values)
    uniq, inv = torch.unique(indices, sorted=True, return_inverse=True)
    out = torch.zeros((uniq.shape[0],) + tuple(values.shape[1:]),
                      dtype=values.dtype, device=values.device)
    out.index_add_(0, inv, values)
    return uniq, out
