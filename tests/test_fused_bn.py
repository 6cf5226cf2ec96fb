"""Fused BN numerics (CPU fallback path) vs torch reference; the GPU HIP
kernels are checked against the same references in test_gpu_kernels.py."""
import copy

import torch

from autodist_amd.ops.fused_bn import FusedBatchNorm2d, fused_bn_train


def test_fused_bn_forward_backward_matches_torch():
    torch.manual_seed(0)
    N, C, H, W = 4, 16, 5, 5
    x = torch.randn(N, C, H, W, requires_grad=True)
    res = torch.randn(N, C, H, W, requires_grad=True)
    w = torch.nn.Parameter(torch.rand(C) + 0.5)
    b = torch.nn.Parameter(torch.randn(C))
    rm, rv = torch.zeros(C), torch.ones(C)

    x2 = x.detach().clone().requires_grad_(True)
    res2 = res.detach().clone().requires_grad_(True)
    w2 = torch.nn.Parameter(w.detach().clone())
    b2 = torch.nn.Parameter(b.detach().clone())
    rm2, rv2 = torch.zeros(C), torch.ones(C)

    y = fused_bn_train(x, w, b, rm, rv, momentum=0.1, eps=1e-5, relu=True,
                       residual=res)
    y_ref = torch.relu(torch.nn.functional.batch_norm(
        x2, rm2, rv2, w2, b2, training=True, momentum=0.1, eps=1e-5) + res2)
    assert torch.allclose(y, y_ref, atol=1e-5)
    assert torch.allclose(rm, rm2, atol=1e-6)
    assert torch.allclose(rv, rv2, atol=1e-5)

    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(res.grad, res2.grad, atol=1e-6)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
    assert torch.allclose(b.grad, b2.grad, atol=1e-4)


def test_fused_module_state_dict_compatible():
    m = FusedBatchNorm2d(32, relu=True)
    ref = torch.nn.BatchNorm2d(32)
    assert set(m.state_dict().keys()) == set(ref.state_dict().keys())
    # load torch BN weights into fused module
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.normal_()
    m.load_state_dict(ref.state_dict())
    assert torch.equal(m.weight, ref.weight)


def test_resnet18_fused_matches_unfused_cpu():
    torch.manual_seed(0)
    from autodist_amd.models.resnet import resnet18
    m_ref = resnet18(num_classes=10, fused=False)
    m_fused = resnet18(num_classes=10, fused=True)
    m_fused.load_state_dict(m_ref.state_dict())
    x = torch.randn(2, 3, 64, 64)
    y_ref = m_ref(x)
    y_fused = m_fused(x)
    assert torch.allclose(y_ref, y_fused, atol=1e-4), \
        (y_ref - y_fused).abs().max()
    loss_r = y_ref.square().mean()
    loss_f = y_fused.square().mean()
    loss_r.backward()
    loss_f.backward()
    for (n1, p1), (n2, p2) in zip(m_ref.named_parameters(),
                                  m_fused.named_parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-3), \
            f"{n1}: {(p1.grad - p2.grad).abs().max()}"


def test_fused_bn_eval_mode():
    torch.manual_seed(1)
    m = FusedBatchNorm2d(8, relu=False)
    ref = torch.nn.BatchNorm2d(8)
    ref.load_state_dict(m.state_dict())
    x = torch.randn(3, 8, 4, 4)
    m.train()
    ref.train()
    m(x)
    ref(x)
    m.eval()
    ref.eval()
    x2 = torch.randn(3, 8, 4, 4)
    assert torch.allclose(m(x2), ref(x2), atol=1e-5)
