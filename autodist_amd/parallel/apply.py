"""Optimizer update application — the engine's own `ResourceApply*` layer.

Reference context: TF executes per-variable `ResourceApply*` ops recognized in
autodist/kernel/common/op_info.py:24-68 and re-created by the partitioner
(partitioner.py:570-573). The MI355X engine applies updates ITSELF (it must:
PS owners update only their shard, with shard-local optimizer state), with
semantics bit-matching torch.optim so single-GPU training and checkpoints are
interchangeable.

Dense updates run through torch._foreach_* multi-tensor ops (one horizontally
fused launch group per bucket); on gfx950 the hand-written HIP multi-tensor
kernels in ops/csrc/multi_tensor.hip replace them (one kernel launch per
bucket per step, LDS-free pure-bandwidth kernels tuned for HBM3E).

Supported exactly: SGD (momentum/nesterov/dampening/weight_decay/maximize),
Adam, AdamW, Adagrad, RMSprop. Sparse rows: sparse_apply_* variants implement
the `SparseApply*` table (op_info.py:73-117).
"""
import math
from typing import Dict, List

import torch

_SUPPORTED = ("SGD", "Adam", "AdamW", "Adagrad", "RMSprop", "Adamax",
              "NAdam", "RAdam", "Adadelta", "ASGD", "Rprop")


def is_supported(cls_name: str) -> bool:
    return cls_name in _SUPPORTED


def make_state(cls_name: str, param: torch.Tensor, hyper: dict) -> Dict[str, torch.Tensor]:
    """Initialize optimizer state for one param/shard, matching torch.optim
    lazy-init semantics."""
    if cls_name == "SGD":
        return {}  # momentum buffer lazily = first grad (torch semantics)
    if cls_name in ("Adam", "AdamW"):
        return {"step": torch.zeros((), dtype=torch.float32),
                "exp_avg": torch.zeros_like(param),
                "exp_avg_sq": torch.zeros_like(param)}
    if cls_name == "Adagrad":
        init = float(hyper.get("initial_accumulator_value", 0.0))
        return {"step": torch.zeros((), dtype=torch.float32),
                "sum": torch.full_like(param, init)}
    if cls_name == "RMSprop":
        s = {"square_avg": torch.zeros_like(param)}
        if hyper.get("momentum", 0) > 0:
            s["momentum_buffer"] = torch.zeros_like(param)
        if hyper.get("centered", False):
            s["grad_avg"] = torch.zeros_like(param)
        return s
    if cls_name == "Adamax":
        return {"step": torch.zeros((), dtype=torch.float32),
                "exp_avg": torch.zeros_like(param),
                "exp_inf": torch.zeros_like(param)}
    if cls_name == "NAdam":
        return {"step": torch.zeros((), dtype=torch.float32),
                "mu_product": torch.ones((), dtype=torch.float32),
                "exp_avg": torch.zeros_like(param),
                "exp_avg_sq": torch.zeros_like(param)}
    if cls_name == "RAdam":
        return {"step": torch.zeros((), dtype=torch.float32),
                "exp_avg": torch.zeros_like(param),
                "exp_avg_sq": torch.zeros_like(param)}
    if cls_name == "Adadelta":
        return {"step": torch.zeros((), dtype=torch.float32),
                "square_avg": torch.zeros_like(param),
                "acc_delta": torch.zeros_like(param)}
    if cls_name == "ASGD":
        return {"step": torch.zeros((), dtype=torch.float32),
                "eta": torch.tensor(float(hyper.get("lr", 1e-2))),
                "mu": torch.ones(()),
                "ax": param.detach().clone().to(torch.float32)}
    if cls_name == "Rprop":
        return {"step": torch.zeros((), dtype=torch.float32),
                "prev": torch.zeros_like(param),
                "step_size": torch.full_like(
                    param, float(hyper.get("lr", 1e-2)))}
    raise NotImplementedError(f"optimizer {cls_name} not supported for engine apply")


def _maybe_weight_decay(grads, params, wd):
    if wd != 0:
        grads = torch._foreach_add(grads, params, alpha=wd)
    return grads


def apply_sgd(params: List[torch.Tensor], grads: List[torch.Tensor],
              states: List[dict], hyper: dict):
    lr = hyper["lr"]
    momentum = hyper.get("momentum", 0.0)
    dampening = hyper.get("dampening", 0.0)
    nesterov = hyper.get("nesterov", False)
    wd = hyper.get("weight_decay", 0.0)
    maximize = hyper.get("maximize", False)
    if maximize:
        grads = torch._foreach_neg(grads)
    grads = _maybe_weight_decay(grads, params, wd)
    if momentum != 0:
        bufs = []
        fresh = []
        for g, st in zip(grads, states):
            if "momentum_buffer" not in st:
                st["momentum_buffer"] = torch.clone(g).detach()
                fresh.append(True)
            else:
                fresh.append(False)
            bufs.append(st["momentum_buffer"])
        stale = [i for i, f in enumerate(fresh) if not f]
        if stale:
            torch._foreach_mul_([bufs[i] for i in stale], momentum)
            torch._foreach_add_([bufs[i] for i in stale],
                                [grads[i] for i in stale], alpha=1 - dampening)
        if nesterov:
            grads = torch._foreach_add(grads, bufs, alpha=momentum)
        else:
            grads = bufs
    torch._foreach_add_(params, grads, alpha=-lr)


def _adam_impl(params, grads, states, hyper, decoupled_wd: bool):
    lr = hyper["lr"]
    beta1, beta2 = hyper.get("betas", (0.9, 0.999))
    eps = hyper.get("eps", 1e-8)
    wd = hyper.get("weight_decay", 0.01 if decoupled_wd else 0.0)
    amsgrad = hyper.get("amsgrad", False)
    maximize = hyper.get("maximize", False)
    if maximize:
        grads = torch._foreach_neg(grads)
    if decoupled_wd:
        if wd != 0:
            torch._foreach_mul_(params, 1 - lr * wd)
    else:
        grads = _maybe_weight_decay(grads, params, wd)
    exp_avgs = [st["exp_avg"] for st in states]
    exp_avg_sqs = [st["exp_avg_sq"] for st in states]
    for st in states:
        st["step"] += 1
    steps = [float(st["step"]) for st in states]
    torch._foreach_lerp_(exp_avgs, grads, 1 - beta1)
    torch._foreach_mul_(exp_avg_sqs, beta2)
    torch._foreach_addcmul_(exp_avg_sqs, grads, grads, value=1 - beta2)
    if amsgrad:
        for st, v in zip(states, exp_avg_sqs):
            if "max_exp_avg_sq" not in st:
                st["max_exp_avg_sq"] = torch.zeros_like(v)
            torch.maximum(st["max_exp_avg_sq"], v, out=st["max_exp_avg_sq"])
        exp_avg_sqs = [st["max_exp_avg_sq"] for st in states]
    # per-tensor bias correction (steps may differ across shards)
    for p, m, v, step in zip(params, exp_avgs, exp_avg_sqs, steps):
        bc1 = 1 - beta1 ** step
        bc2 = 1 - beta2 ** step
        denom = (v.sqrt() / math.sqrt(bc2)).add_(eps)
        p.addcdiv_(m, denom, value=-lr / bc1)


def apply_adam(params, grads, states, hyper):
    _adam_impl(params, grads, states, hyper, decoupled_wd=False)


def apply_adamw(params, grads, states, hyper):
    _adam_impl(params, grads, states, hyper, decoupled_wd=True)


def apply_adagrad(params, grads, states, hyper):
    lr = hyper["lr"]
    lr_decay = hyper.get("lr_decay", 0.0)
    eps = hyper.get("eps", 1e-10)
    wd = hyper.get("weight_decay", 0.0)
    grads = _maybe_weight_decay(grads, params, wd)
    for st in states:
        st["step"] += 1
    sums = [st["sum"] for st in states]
    torch._foreach_addcmul_(sums, grads, grads, value=1.0)
    for p, g, s, st in zip(params, grads, sums, states):
        clr = lr / (1 + (float(st["step"]) - 1) * lr_decay)
        p.addcdiv_(g, s.sqrt().add_(eps), value=-clr)


def apply_rmsprop(params, grads, states, hyper):
    lr = hyper["lr"]
    alpha = hyper.get("alpha", 0.99)
    eps = hyper.get("eps", 1e-8)
    wd = hyper.get("weight_decay", 0.0)
    momentum = hyper.get("momentum", 0.0)
    centered = hyper.get("centered", False)
    grads = _maybe_weight_decay(grads, params, wd)
    sq = [st["square_avg"] for st in states]
    torch._foreach_mul_(sq, alpha)
    torch._foreach_addcmul_(sq, grads, grads, value=1 - alpha)
    if centered:
        gavg = [st["grad_avg"] for st in states]
        torch._foreach_lerp_(gavg, grads, 1 - alpha)
        avg = [s.addcmul(ga, ga, value=-1).sqrt_().add_(eps)
               for s, ga in zip(sq, gavg)]
    else:
        avg = [s.sqrt().add_(eps) for s in sq]
    if momentum > 0:
        bufs = [st["momentum_buffer"] for st in states]
        torch._foreach_mul_(bufs, momentum)
        for b, g, a in zip(bufs, grads, avg):
            b.addcdiv_(g, a)
        torch._foreach_add_(params, bufs, alpha=-lr)
    else:
        for p, g, a in zip(params, grads, avg):
            p.addcdiv_(g, a, value=-lr)


def apply_adamax(params, grads, states, hyper):
    lr = hyper["lr"]
    beta1, beta2 = hyper.get("betas", (0.9, 0.999))
    eps = hyper.get("eps", 1e-8)
    wd = hyper.get("weight_decay", 0.0)
    grads = _maybe_weight_decay(grads, params, wd)
    for p, g, st in zip(params, grads, states):
        st["step"] += 1
        step = float(st["step"])
        st["exp_avg"].lerp_(g, 1 - beta1)
        torch.maximum(st["exp_inf"].mul_(beta2), g.abs().add_(eps),
                      out=st["exp_inf"])
        clr = lr / (1 - beta1 ** step)
        p.addcdiv_(st["exp_avg"], st["exp_inf"], value=-clr)


def apply_nadam(params, grads, states, hyper):
    lr = hyper["lr"]
    beta1, beta2 = hyper.get("betas", (0.9, 0.999))
    eps = hyper.get("eps", 1e-8)
    wd = hyper.get("weight_decay", 0.0)
    psi = hyper.get("momentum_decay", 4e-3)
    decoupled = hyper.get("decoupled_weight_decay", False)
    if wd != 0 and decoupled:
        torch._foreach_mul_(params, 1 - lr * wd)
    elif wd != 0:
        grads = torch._foreach_add(grads, params, alpha=wd)
    for p, g, st in zip(params, grads, states):
        st["step"] += 1
        step = float(st["step"])
        bc2 = 1 - beta2 ** step
        mu = beta1 * (1.0 - 0.5 * 0.96 ** (step * psi))
        mu_next = beta1 * (1.0 - 0.5 * 0.96 ** ((step + 1) * psi))
        st["mu_product"] *= mu
        mu_product = float(st["mu_product"])
        st["exp_avg"].lerp_(g, 1 - beta1)
        st["exp_avg_sq"].mul_(beta2).addcmul_(g, g, value=1 - beta2)
        denom = st["exp_avg_sq"].div(bc2).sqrt_().add_(eps)
        p.addcdiv_(g, denom, value=-lr * (1 - mu) / (1 - mu_product))
        p.addcdiv_(st["exp_avg"], denom,
                   value=-lr * mu_next / (1 - mu_product * mu_next))


def apply_radam(params, grads, states, hyper):
    lr = hyper["lr"]
    beta1, beta2 = hyper.get("betas", (0.9, 0.999))
    eps = hyper.get("eps", 1e-8)
    wd = hyper.get("weight_decay", 0.0)
    decoupled = hyper.get("decoupled_weight_decay", False)
    if wd != 0 and decoupled:
        torch._foreach_mul_(params, 1 - lr * wd)
    elif wd != 0:
        grads = torch._foreach_add(grads, params, alpha=wd)
    rho_inf = 2.0 / (1 - beta2) - 1
    for p, g, st in zip(params, grads, states):
        st["step"] += 1
        step = float(st["step"])
        bc1 = 1 - beta1 ** step
        bc2 = 1 - beta2 ** step
        st["exp_avg"].lerp_(g, 1 - beta1)
        st["exp_avg_sq"].mul_(beta2).addcmul_(g, g, value=1 - beta2)
        m_hat = st["exp_avg"] / bc1
        rho_t = rho_inf - 2 * step * (beta2 ** step) / bc2
        if rho_t > 5.0:
            rect = math.sqrt((rho_t - 4) * (rho_t - 2) * rho_inf
                             / ((rho_inf - 4) * (rho_inf - 2) * rho_t))
            adaptive_lr = math.sqrt(bc2) / st["exp_avg_sq"].sqrt().add_(eps)
            p.add_(m_hat * lr * adaptive_lr * rect, alpha=-1.0)
        else:
            p.add_(m_hat * lr, alpha=-1.0)


def apply_adadelta(params, grads, states, hyper):
    lr = hyper["lr"]
    rho = hyper.get("rho", 0.9)
    eps = hyper.get("eps", 1e-6)
    wd = hyper.get("weight_decay", 0.0)
    grads = _maybe_weight_decay(grads, params, wd)
    for p, g, st in zip(params, grads, states):
        st["step"] += 1
        st["square_avg"].mul_(rho).addcmul_(g, g, value=1 - rho)
        std = st["square_avg"].add(eps).sqrt_()
        delta = st["acc_delta"].add(eps).sqrt_().div_(std).mul_(g)
        st["acc_delta"].mul_(rho).addcmul_(delta, delta, value=1 - rho)
        p.add_(delta, alpha=-lr)


def apply_asgd(params, grads, states, hyper):
    lr = hyper["lr"]
    lambd = hyper.get("lambd", 1e-4)
    alpha = hyper.get("alpha", 0.75)
    t0 = hyper.get("t0", 1e6)
    wd = hyper.get("weight_decay", 0.0)
    grads = _maybe_weight_decay(grads, params, wd)
    for p, g, st in zip(params, grads, states):
        st["step"] += 1
        step = float(st["step"])
        eta = float(st["eta"])
        mu = float(st["mu"])
        p.mul_(1 - lambd * eta)
        p.add_(g, alpha=-eta)
        if mu != 1:
            st["ax"].add_(p.sub(st["ax"]).mul_(mu))
        else:
            st["ax"].copy_(p)
        st["eta"] = torch.tensor(lr / (1 + lambd * lr * step) ** alpha)
        st["mu"] = torch.tensor(1.0 / max(1.0, step - t0))


def apply_rprop(params, grads, states, hyper):
    etaminus, etaplus = hyper.get("etas", (0.5, 1.2))
    step_min, step_max = hyper.get("step_sizes", (1e-6, 50.0))
    for p, g, st in zip(params, grads, states):
        st["step"] += 1
        sign = g.mul(st["prev"]).sign()
        mult = torch.where(sign > 0, etaplus,
                           torch.where(sign < 0, etaminus, 1.0))
        st["step_size"].mul_(mult).clamp_(step_min, step_max)
        g = g.clone()
        g[sign.eq(-1)] = 0
        p.addcmul_(g.sign(), st["step_size"], value=-1)
        st["prev"].copy_(g)


_APPLY = {"SGD": apply_sgd, "Adam": apply_adam, "AdamW": apply_adamw,
          "Adagrad": apply_adagrad, "RMSprop": apply_rmsprop,
          "Adamax": apply_adamax, "NAdam": apply_nadam, "RAdam": apply_radam,
          "Adadelta": apply_adadelta, "ASGD": apply_asgd,
          "Rprop": apply_rprop}


def apply_dense(cls_name: str, params, grads, states, hyper):
    """Apply one optimizer update to a list of params/shards. On GPU,
    tensors route through the hand-written fused HIP kernels (one launch
    per tensor instead of 3-5 _foreach passes); non-contiguous/odd-dtype
    shards fall back to the torch multi-tensor path."""
    if not params:
        return
    if params[0].is_cuda:
        from autodist_amd.ops import api as ops_api
        if ops_api.has_gpu_ops() and ops_api.has_fused(cls_name):
            rest = ([], [], [])
            for pp, gg, st in zip(params, grads, states):
                if (pp.is_contiguous() and gg.is_contiguous()
                        and pp.dtype == torch.float32
                        and gg.dtype == torch.float32
                        and ops_api.fused_apply(cls_name, pp, gg, st,
                                                hyper)):
                    continue
                rest[0].append(pp)
                rest[1].append(gg)
                rest[2].append(st)
            if rest[0]:
                _APPLY[cls_name](rest[0], rest[1], rest[2], hyper)
            return
    _APPLY[cls_name](params, grads, states, hyper)


def apply_flat(cls_name: str, param: torch.Tensor, grad: torch.Tensor,
               state: dict, hyper: dict):
    """Bucket-level update over flat buffers: ONE hand-written gfx950 HIP
    kernel launch on GPU (ops/csrc/multi_tensor.hip), torch fallback on CPU."""
    from autodist_amd.ops import api as ops_api
    if param.is_cuda and ops_api.has_gpu_ops():
        if ops_api.fused_apply(cls_name, param, grad, state, hyper):
            return
    apply_dense(cls_name, [param], [grad], [state], hyper)


# -- sparse (row-wise) applies: the SparseApply* table ----------------------

def apply_sparse_rows(cls_name: str, param: torch.Tensor, rows: torch.Tensor,
                      row_grads: torch.Tensor, state: dict, hyper: dict):
    """Row-sparse update: only `rows` of param are touched (embedding grads).
    Matches torch.optim sparse semantics (SGD/Adagrad support sparse grads;
    sparse Adam follows torch.optim.SparseAdam)."""
    if cls_name == "SGD":
        # torch SGD rejects sparse grads with momentum/weight_decay; match it
        if hyper.get("momentum", 0) or hyper.get("weight_decay", 0):
            raise ValueError(
                "sparse SGD does not support momentum/weight_decay "
                "(torch.optim.SGD semantics)")
        param.index_add_(0, rows, row_grads, alpha=-hyper["lr"])
        return
    if cls_name == "Adagrad":
        state["step"] += 1
        lr = hyper["lr"] / (1 + (float(state["step"]) - 1) * hyper.get("lr_decay", 0.0))
        s_rows = state["sum"].index_select(0, rows)
        s_rows.addcmul_(row_grads, row_grads, value=1.0)
        state["sum"].index_copy_(0, rows, s_rows)
        upd = row_grads / s_rows.sqrt().add_(hyper.get("eps", 1e-10))
        param.index_add_(0, rows, upd, alpha=-lr)
        return
    if cls_name in ("Adam", "AdamW", "SparseAdam"):
        # SparseAdam semantics: moments updated only on touched rows,
        # bias correction by a global step count.
        state["step"] += 1
        step = float(state["step"])
        beta1, beta2 = hyper.get("betas", (0.9, 0.999))
        eps = hyper.get("eps", 1e-8)
        lr = hyper["lr"]
        m_rows = state["exp_avg"].index_select(0, rows)
        v_rows = state["exp_avg_sq"].index_select(0, rows)
        m_rows.mul_(beta1).add_(row_grads, alpha=1 - beta1)
        v_rows.mul_(beta2).addcmul_(row_grads, row_grads, value=1 - beta2)
        state["exp_avg"].index_copy_(0, rows, m_rows)
        state["exp_avg_sq"].index_copy_(0, rows, v_rows)
        bc1 = 1 - beta1 ** step
        bc2 = 1 - beta2 ** step
        upd = m_rows / bc1 / (v_rows.sqrt() / math.sqrt(bc2)).add_(eps)
        param.index_add_(0, rows, upd, alpha=-lr)
        return
    raise NotImplementedError(f"sparse apply for {cls_name} not supported")
