"""Fused BN(+ReLU)(+residual) numerics — CPU fallback vs torch composition,
and the full GPU-kernel comparison (gpu-marked)."""
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from dear_pytorch_amd.ops.fused_bn import FusedBNAct2d


def _ref(x, bn, relu, residual=None):
    y = bn(x)
    if residual is not None:
        y = y + residual
    return F.relu(y) if relu else y


@pytest.mark.parametrize("relu,res", [(False, False), (True, False),
                                      (True, True)])
def test_cpu_fallback_matches_torch(relu, res):
    torch.manual_seed(0)
    C = 8
    fused = FusedBNAct2d(C, relu=relu)
    ref_bn = nn.BatchNorm2d(C)
    ref_bn.load_state_dict(fused.state_dict())
    x = torch.randn(4, C, 6, 6, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    r = torch.randn(4, C, 6, 6) if res else None
    y1 = fused(x, residual=r)
    y2 = _ref(x2, ref_bn, relu, r)
    assert torch.allclose(y1, y2, atol=1e-6)
    y1.sum().backward(); y2.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(fused.running_mean, ref_bn.running_mean, atol=1e-6)


def test_state_dict_compatible_with_batchnorm():
    fused = FusedBNAct2d(16, relu=True)
    plain = nn.BatchNorm2d(16)
    fused.load_state_dict(plain.state_dict())
    plain.load_state_dict(fused.state_dict())


def test_fused_resnet_cpu_matches_plain():
    from dear_pytorch_amd import models
    torch.manual_seed(0)
    a = models.get_cnn("resnet18", num_classes=10, fused_bn=False)
    torch.manual_seed(0)
    b = models.get_cnn("resnet18", num_classes=10, fused_bn=True)
    b.load_state_dict(a.state_dict())
    x = torch.randn(2, 3, 64, 64)
    ya, yb = a(x), b(x)
    assert torch.allclose(ya, yb, atol=1e-5)
    ya.sum().backward(); yb.sum().backward()
    for (n, pa), (_, pb) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(pa.grad, pb.grad, atol=1e-4), n


@pytest.mark.gpu
@pytest.mark.parametrize("relu,res,C,hw", [
    (False, False, 64, 14), (True, False, 64, 14), (True, True, 256, 14),
    (True, False, 3, 9),  # C < wavefront
    (True, True, 130, 7),  # C not multiple of block
])
def test_gpu_kernel_matches_torch_fp32(relu, res, C, hw):
    torch.manual_seed(1)
    dev = torch.device("cuda:0")
    fused = FusedBNAct2d(C, relu=relu).to(dev)
    ref_bn = nn.BatchNorm2d(C).to(dev)
    ref_bn.load_state_dict(fused.state_dict())
    N = 8
    x = torch.randn(N, C, hw, hw, device=dev) \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    r = None
    r2 = None
    if res:
        r = torch.randn(N, C, hw, hw, device=dev) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        r2 = r.detach().clone().requires_grad_(True)
    y1 = fused(x, residual=r)
    y2 = _ref(x2, ref_bn, relu, r2)
    assert torch.allclose(y1, y2, atol=1e-4), \
        (y1 - y2).abs().max().item()
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-3), \
        (x.grad - x2.grad).abs().max().item()
    assert torch.allclose(fused.weight.grad, ref_bn.weight.grad, atol=1e-2)
    assert torch.allclose(fused.bias.grad, ref_bn.bias.grad, atol=1e-2)
    if res:
        assert torch.allclose(r.grad, r2.grad, atol=1e-4)
    assert torch.allclose(fused.running_mean, ref_bn.running_mean, atol=1e-4)
    assert torch.allclose(fused.running_var, ref_bn.running_var, atol=1e-4)


@pytest.mark.gpu
def test_gpu_eval_mode_matches():
    torch.manual_seed(2)
    dev = torch.device("cuda:0")
    C = 64
    fused = FusedBNAct2d(C, relu=True).to(dev)
    ref_bn = nn.BatchNorm2d(C).to(dev)
    # populate running stats
    x = torch.randn(8, C, 14, 14, device=dev) \
        .to(memory_format=torch.channels_last)
    fused(x)
    ref_bn.load_state_dict(fused.state_dict())
    fused.eval(); ref_bn.eval()
    with torch.no_grad():
        y1 = fused(x)
        y2 = F.relu(ref_bn(x))
    assert torch.allclose(y1, y2, atol=1e-5)


@pytest.mark.gpu
def test_gpu_fused_resnet50_e2e_matches_plain():
    """Full-model check: fused-BN ResNet-50 tracks the plain model's loss
    trajectory and parameters over 3 DeAR training steps (fp32, NHWC)."""
    import dear_pytorch_amd as dear
    from dear_pytorch_amd import models
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    a = models.get_cnn("resnet50", num_classes=100, fused_bn=False).to(dev)
    b = models.get_cnn("resnet50", num_classes=100, fused_bn=True).to(dev)
    b.load_state_dict(a.state_dict())
    a = a.to(memory_format=torch.channels_last)
    b = b.to(memory_format=torch.channels_last)
    x = torch.randn(8, 3, 224, 224, device=dev) \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 100, (8,), device=dev)
    lossf = nn.CrossEntropyLoss()
    # single fwd+bwd: losses and gradients must agree tightly
    la = lossf(a(x), y)
    la.backward()
    lb = lossf(b(x), y)
    lb.backward()
    assert abs(la.item() - lb.item()) < 5e-3 * max(1.0, la.item()), \
        (la.item(), lb.item())
    # relative grad agreement (Frobenius): stable against deep-net
    # reduction-order amplification, unlike max-abs
    for (n, pa), (_, pb) in zip(a.named_parameters(), b.named_parameters()):
        if pa.grad is None or pb.grad is None:
            continue
        ref = pa.grad.norm().item()
        diff = (pa.grad - pb.grad).norm().item()
        assert diff <= 0.1 * ref + 1e-4, (n, diff, ref)
    for m in (a, b):
        m.zero_grad(set_to_none=True)
    # 3 training steps: trajectories stay close (loose — 53 BN layers
    # amplify reduction-order differences chaotically)
    oa = dear.DistributedOptimizer(
        torch.optim.SGD(a.parameters(), lr=0.01, momentum=0.9), model=a)
    ob = dear.DistributedOptimizer(
        torch.optim.SGD(b.parameters(), lr=0.01, momentum=0.9), model=b)
    for i in range(3):
        la = lossf(a(x), y)
        la.backward()
        oa.step()
        lb = lossf(b(x), y)
        lb.backward()
        ob.step()
        assert abs(la.item() - lb.item()) < 0.15 * max(1.0, la.item()), \
            (i, la.item(), lb.item())
    oa.synchronize(); ob.synchronize()


def test_fused_densenet_cpu_matches_plain():
    from dear_pytorch_amd import models
    torch.manual_seed(0)
    a = models.get_cnn("densenet121", num_classes=10, fused_bn=False)
    torch.manual_seed(0)
    b = models.get_cnn("densenet121", num_classes=10, fused_bn=True)
    b.load_state_dict(a.state_dict())
    x = torch.randn(2, 3, 64, 64)
    ya, yb = a(x), b(x)
    assert torch.allclose(ya, yb, atol=1e-5)
    ya.sum().backward(); yb.sum().backward()
    for (n, pa), (_, pb) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(pa.grad, pb.grad, atol=1e-4), n


def test_fused_inceptionv4_cpu_matches_plain():
    from dear_pytorch_amd import models
    torch.manual_seed(0)
    a = models.get_cnn("inceptionv4", num_classes=10, fused_bn=False)
    torch.manual_seed(0)
    b = models.get_cnn("inceptionv4", num_classes=10, fused_bn=True)
    b.load_state_dict(a.state_dict())
    a.eval(); b.eval()  # cheap single pass, eval avoids dropout
    x = torch.randn(1, 3, 299, 299)
    with torch.no_grad():
        ya, yb = a(x), b(x)
    assert torch.allclose(ya, yb, atol=1e-5)
