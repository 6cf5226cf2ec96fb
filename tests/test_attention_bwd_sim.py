"""CPU simulation of the attention-backward kernel's EXACT tile algorithm
(ops/csrc/attention_bwd.hip) vs torch autograd.

Validates the math the draft kernels implement — per-row online m/l stats,
delta = rowsum(dO*O), per-32-key-tile dS with the scale folded in, bf16
quantization at the LDS staging points, and the K2 mirror accumulation — so
the only risk left for the GPU validation (round 2) is the mechanical HIP
lane mapping, which reuses GPU-verified forward patterns."""
import math

import torch


def _sim_attention_bwd(q, k, v, do, scale):
    """Mirror of attn_bwd_q_kernel + attn_bwd_kv_kernel tile semantics
    (fp32 stats; dS/P staged through bf16 like the LDS round trip)."""
    S = q.shape[0]
    qf, kf, vf, dof = q.float(), k.float(), v.float(), do.float()
    s_full = (qf @ kf.T) * scale
    # pass A: online m/l over 32-key tiles == exact rowwise logsumexp parts
    m = torch.full((S,), -1e30)
    l = torch.zeros(S)
    for kt in range(0, S, 32):
        t = s_full[:, kt:kt + 32]
        m_new = torch.maximum(m, t.max(-1).values)
        l = l * torch.exp(m - m_new) + torch.exp(t - m_new[:, None]).sum(-1)
        m = m_new
    # forward output (for delta) with bf16 P staging like the fwd kernel
    o = torch.zeros(S, q.shape[1])
    for kt in range(0, S, 32):
        p = (torch.exp(s_full[:, kt:kt + 32] - m[:, None]) / l[:, None])
        p = p.to(torch.bfloat16).float()
        o = o + p * l[:, None] / l[:, None] @ vf[kt:kt + 32]  # == p @ v
    delta = (dof * o).sum(-1)
    # K1 pass B: dQ
    dq = torch.zeros_like(qf)
    for kt in range(0, S, 32):
        p = torch.exp(s_full[:, kt:kt + 32] - m[:, None]) / l[:, None]
        dp = dof @ vf[kt:kt + 32].T
        ds = (p * (dp - delta[:, None]) * scale).to(torch.bfloat16).float()
        dq = dq + ds @ kf[kt:kt + 32]
    # K2: dK, dV per key tile, mirrored stats per q column
    dk = torch.zeros_like(kf)
    dv = torch.zeros_like(vf)
    for k0 in range(0, S, 16):
        for qt in range(0, S, 32):
            sp = (kf[k0:k0 + 16] @ qf[qt:qt + 32].T) * scale  # [key][q]
            m_q = m[qt:qt + 32]
            l_q = l[qt:qt + 32]
            d_q = delta[qt:qt + 32]
            pp = torch.exp(sp - m_q[None, :]) / l_q[None, :]
            dpp = vf[k0:k0 + 16] @ dof[qt:qt + 32].T
            dsp = (pp * (dpp - d_q[None, :]) * scale)
            pp_q = pp.to(torch.bfloat16).float()
            dsp_q = dsp.to(torch.bfloat16).float()
            dv[k0:k0 + 16] += pp_q @ dof[qt:qt + 32]
            dk[k0:k0 + 16] += dsp_q @ qf[qt:qt + 32]
    return dq, dk, dv, o


def test_bwd_algorithm_matches_autograd():
    torch.manual_seed(0)
    S, D = 96, 64
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(S, D, dtype=torch.bfloat16)
    k = torch.randn(S, D, dtype=torch.bfloat16)
    v = torch.randn(S, D, dtype=torch.bfloat16)
    do = torch.randn(S, D, dtype=torch.bfloat16)
    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    o_ref = torch.softmax((qf @ kf.T) * scale, -1) @ vf
    o_ref.backward(do.float())
    dq, dk, dv, o = _sim_attention_bwd(q, k, v, do, scale)
    assert (o - o_ref.detach()).abs().max() < 3e-2
    assert (dq - qf.grad).abs().max() < 6e-2, (dq - qf.grad).abs().max()
    assert (dk - kf.grad).abs().max() < 6e-2, (dk - kf.grad).abs().max()
    assert (dv - vf.grad).abs().max() < 6e-2, (dv - vf.grad).abs().max()


def test_bwd_algorithm_with_outlier_keys():
    """Spiked keys force large m jumps across tiles (the stats path)."""
    torch.manual_seed(1)
    S, D = 64, 64
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(S, D, dtype=torch.bfloat16)
    k = torch.randn(S, D, dtype=torch.bfloat16)
    k[55] = (q[3].float() * 4).to(torch.bfloat16)
    v = torch.randn(S, D, dtype=torch.bfloat16)
    do = torch.randn(S, D, dtype=torch.bfloat16)
    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    (torch.softmax((qf @ kf.T) * scale, -1) @ vf).backward(do.float())
    dq, dk, dv, _ = _sim_attention_bwd(q, k, v, do, scale)
    assert (dq - qf.grad).abs().max() < 8e-2
    assert (dk - kf.grad).abs().max() < 8e-2
    assert (dv - vf.grad).abs().max() < 8e-2
