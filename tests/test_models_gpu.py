"""GPU training-step checks for the CNN benchmark families with the fused
gfx950 BN kernels composed in (bf16 autocast, channels_last)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("maker,size", [("densenet121", 64),
                                        ("vgg16_bn", 64),
                                        ("inception_v3", 299)])
def test_cnn_family_fused_step(maker, size):
    from autodist_amd.models.densenet import densenet121
    from autodist_amd.models.inception import inception_v3
    from autodist_amd.models.vgg import vgg16
    makers = {
        "densenet121": lambda: densenet121(num_classes=16, fused=True),
        "vgg16_bn": lambda: vgg16(num_classes=16, batch_norm=True,
                                  fused=True),
        "inception_v3": lambda: inception_v3(num_classes=16, fused=True),
    }
    device = torch.device("cuda", 0)
    torch.manual_seed(0)
    model = makers[maker]().to(device).to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    x = torch.randn(4, 3, size, size, device=device).contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 16, (4,), device=device)
    with torch.autocast("cuda", torch.bfloat16):
        loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    for p in model.parameters():
        assert torch.isfinite(p).all()
        assert p.grad is None or torch.isfinite(p.grad).all()


def test_fused_vs_unfused_densenet_forward():
    """Fused-BN DenseNet forward matches the unfused fp32 forward."""
    from autodist_amd.models.densenet import densenet121
    torch.manual_seed(0)
    m_ref = densenet121(num_classes=8, fused=False)
    m_fused = densenet121(num_classes=8, fused=True)
    m_fused.load_state_dict(m_ref.state_dict())
    device = torch.device("cuda", 0)
    m_ref = m_ref.to(device)
    m_fused = m_fused.to(device).to(memory_format=torch.channels_last)
    x = torch.randn(2, 3, 64, 64, device=device)
    y_ref = m_ref(x)
    y_fused = m_fused(x.contiguous(memory_format=torch.channels_last))
    assert torch.allclose(y_ref, y_fused, atol=5e-2), \
        (y_ref - y_fused).abs().max()
