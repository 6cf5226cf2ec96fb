"""End-to-end GPU tests: engine on a real MI355X (world=1; the driver's
8-GPU scale run covers multi-rank RCCL)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_resnet50_train_step_bf16():
    from autodist_amd.graph_item import GraphItem
    from autodist_amd.models.resnet import resnet50
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.strategy import AllReduce

    device = torch.device("cuda", 0)
    torch.manual_seed(0)
    model = resnet50(num_classes=64).to(device).to(
        memory_format=torch.channels_last)
    g = GraphItem()
    g.extend_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.005, momentum=0.9)
    g.extend_optimizer_info(opt)
    engine = DistributedEngine(g, AllReduce().build(g, ResourceSpec()),
                               rank=0, world_size=1, device=device).setup()
    x = torch.randn(4, 3, 224, 224, device=device).contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 64, (4,), device=device)
    losses = []
    for _ in range(4):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    torch.cuda.synchronize()
    assert losses[-1] < losses[0]
    assert all(v == v for v in losses)


def test_gpu_engine_matches_cpu_reference():
    """Same seed/data: GPU engine (HIP fused apply) must track a CPU torch
    training loop within fp32 tolerance."""
    import copy
    device = torch.device("cuda", 0)
    torch.manual_seed(3)
    model_cpu = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                                    torch.nn.Linear(64, 8))
    model_gpu = copy.deepcopy(model_cpu).to(device)
    opt_cpu = torch.optim.AdamW(model_cpu.parameters(), lr=1e-2,
                                weight_decay=0.01)

    from autodist_amd.graph_item import GraphItem
    from autodist_amd.parallel.engine import DistributedEngine
    from autodist_amd.resource_spec import ResourceSpec
    from autodist_amd.strategy import AllReduce
    g = GraphItem()
    g.extend_model(model_gpu)
    opt_gpu = torch.optim.AdamW(model_gpu.parameters(), lr=1e-2,
                                weight_decay=0.01)
    g.extend_optimizer_info(opt_gpu)
    engine = DistributedEngine(g, AllReduce().build(g, ResourceSpec()),
                               rank=0, world_size=1, device=device).setup()
    for step in range(5):
        torch.manual_seed(100 + step)
        x, y = torch.randn(16, 32), torch.randn(16, 8)
        opt_cpu.zero_grad()
        torch.nn.functional.mse_loss(model_cpu(x), y).backward()
        opt_cpu.step()
        opt_gpu.zero_grad()
        torch.nn.functional.mse_loss(
            model_gpu(x.to(device)), y.to(device)).backward()
        opt_gpu.step()
    torch.cuda.synchronize()
    for pc, pg in zip(model_cpu.parameters(), model_gpu.parameters()):
        assert torch.allclose(pc, pg.detach().cpu(), atol=1e-4), \
            (pc - pg.detach().cpu()).abs().max()


def test_native_extension_loaded():
    """Guard against silent eager fallback on GPU boxes."""
    from autodist_amd.ops import api
    assert api.has_gpu_ops()
    import os
    here = os.path.dirname(os.path.abspath(api.__file__))
    assert any(f.startswith("_autodist_hip") and f.endswith(".so")
               for f in os.listdir(here))
