"""Fused bf16 LayerNorm(+residual) kernels vs fp32 torch reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture
def ext():
    from autodist_amd.ops import api
    if not api.has_gpu_ops():
        pytest.skip("no GPU ops")
    return api.ext()


@pytest.mark.parametrize("N,H", [(64, 768), (37, 1024), (128, 3072),
                                 (16, 100)])
def test_ln_fwd_bwd_vs_torch(ext, N, H):
    torch.manual_seed(0)
    x = torch.randn(N, H, device="cuda", dtype=torch.bfloat16)
    res = torch.randn_like(x)
    gamma = torch.randn(H, device="cuda") * 0.5 + 1.0
    beta = torch.randn(H, device="cuda") * 0.1
    dy = torch.randn_like(x)
    y, u, mean, rstd = ext.ln_fwd(x, res, gamma, beta, 1e-12)
    # fp32 reference on the same bf16-quantized inputs
    xf = (x.float() + res.float()).requires_grad_(True)
    gf = gamma.clone().requires_grad_(True)
    bf = beta.clone().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(xf, (H,), gf, bf, 1e-12)
    yr.backward(dy.float())
    assert (y.float() - yr.detach()).abs().max().item() < 3e-2
    # u is stored bf16: allow bf16 quantization of the fp32 sum
    assert torch.allclose(u.float(), x.float() + res.float(), atol=1e-2,
                          rtol=1e-2)
    dx, dgamma, dbeta = ext.ln_bwd(dy, u, gamma, mean, rstd)
    assert (dx.float() - xf.grad).abs().max().item() < 3e-2, \
        (dx.float() - xf.grad).abs().max()
    # column sums over bf16 inputs: looser tol, relative to magnitude
    assert (dgamma - gf.grad).abs().max().item() < \
        0.02 * gf.grad.abs().max().item() + 0.05
    assert (dbeta - bf.grad).abs().max().item() < \
        0.02 * bf.grad.abs().max().item() + 0.05


def test_fused_ln_module_autograd():
    from autodist_amd.ops.fused_ln import FusedLayerNorm
    torch.manual_seed(1)
    H = 768
    m = FusedLayerNorm(H).cuda()
    x = torch.randn(8, 16, H, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    r = torch.randn_like(x, requires_grad=True)
    y = m(x, residual=r)
    assert y.dtype == torch.bfloat16
    y.float().pow(2).mean().backward()
    assert x.grad is not None and r.grad is not None
    assert torch.equal(x.grad, r.grad)  # identity residual gradient
    assert m.weight.grad is not None and m.weight.grad.dtype == torch.float32
    # reference
    xf = (x.detach().float() + r.detach().float()).requires_grad_(True)
    yr = torch.nn.functional.layer_norm(xf, (H,), m.weight.detach(),
                                        m.bias.detach(), m.eps)
    yr.pow(2).mean().backward()
    assert (x.grad.float() - xf.grad).abs().max().item() < 2e-2


def test_bert_layer_fused_ln_trains():
    """BERT layer with fused LN under autocast: bf16 stream, finite grads,
    close to the fp32 reference layer."""
    from autodist_amd.models.bert import BertConfig, BertLayer
    torch.manual_seed(2)
    cfg = BertConfig(hidden=256, heads=4, intermediate=512, dropout=0.0)
    layer = BertLayer(cfg).cuda()
    x = torch.randn(4, 64, 256, device="cuda")
    with torch.autocast("cuda", torch.bfloat16):
        out = layer(x)
    assert out.dtype == torch.bfloat16
    out.float().pow(2).mean().backward()
    for p in layer.parameters():
        assert p.grad is None or torch.isfinite(p.grad.float()).all()


def test_col_sum_and_fused_linear():
    from autodist_amd.ops import api
    torch.manual_seed(5)
    dy = torch.randn(4096, 768, device="cuda", dtype=torch.bfloat16)
    ref = dy.float().sum(0)
    got = api.ext().col_sum(dy)
    assert (got - ref).abs().max().item() < 0.05 * ref.abs().max().item() + 0.1
    # FusedLinear fwd+bwd vs nn.Linear under autocast
    from autodist_amd.ops.fused_linear import FusedLinear
    lin_ref = torch.nn.Linear(768, 512).cuda()
    lin = FusedLinear(768, 512).cuda()
    with torch.no_grad():
        lin.weight.copy_(lin_ref.weight)
        lin.bias.copy_(lin_ref.bias)
    x = torch.randn(8, 32, 768, device="cuda")
    xr = x.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    with torch.autocast("cuda", torch.bfloat16):
        yr = lin_ref(xr)
        yf = lin(xf)
    assert torch.allclose(yr, yf, atol=2e-2, rtol=1e-2)
    g = torch.randn_like(yr)
    yr.backward(g)
    yf.backward(g)
    assert lin_ref.weight.grad.dtype == lin.weight.grad.dtype
    assert (lin.weight.grad - lin_ref.weight.grad).abs().max().item() < 0.2
    assert (lin.bias.grad - lin_ref.bias.grad).abs().max().item() < \
        0.05 * lin_ref.bias.grad.abs().max().item() + 0.2
    assert (xf.grad - xr.grad).abs().max().item() < 0.1


def test_fused_ce_vs_torch():
    """Online-softmax CE kernels vs F.cross_entropy (loss + input grads)."""
    from autodist_amd.parallel.vocab_parallel import _FusedCERows
    torch.manual_seed(6)
    N, V = 512, 4000
    logits = torch.randn(N, V, device="cuda", dtype=torch.bfloat16) * 3
    targets = torch.randint(0, V, (N,), device="cuda")
    lg = logits.clone().requires_grad_(True)
    loss = _FusedCERows.apply(lg, targets).mean()
    loss.backward()
    lr = logits.float().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lr, targets)
    ref.backward()
    assert abs(loss.item() - ref.item()) < 2e-3 * abs(ref.item()) + 1e-3
    err = (lg.grad.float() - lr.grad).abs().max().item()
    assert err < 5e-5, f"dlogits err {err}"  # grads are O(1/N)
