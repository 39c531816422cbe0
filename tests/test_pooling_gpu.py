"""MgxMaxPool2d numerics vs nn.MaxPool2d (fp32 reference)."""
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu


def _mk(C, N=4, H=30, W=30, dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(N, C, H, W, generator=g).to('cuda', dtype)
    return x.to(memory_format=torch.channels_last)


# (k, s, p, H, W): resnet stem (3,2,1), vgg (2,2,0), odd sizes
CONFIGS = [
    (3, 2, 1, 112, 112),
    (2, 2, 0, 112, 112),
    (2, 2, 0, 56, 56),
    (3, 2, 1, 31, 33),
    (2, 2, 0, 30, 30),
    (3, 1, 1, 14, 14),
]


@pytest.mark.parametrize('k,s,p,H,W', CONFIGS)
@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_forward_backward_matches_torch(k, s, p, H, W, dtype):
    from mgwfbp_amd.kernels.pooling import MgxMaxPool2d
    C = 64
    ref = nn.MaxPool2d(k, stride=s, padding=p)
    ours = MgxMaxPool2d(k, stride=s, padding=p)
    x1 = _mk(C, H=H, W=W, dtype=dtype).requires_grad_(True)
    x2 = x1.detach().clone().requires_grad_(True)
    y = ours(x1)
    y_ref = ref(x2)
    assert y.shape == y_ref.shape
    assert y.is_contiguous(memory_format=torch.channels_last)
    assert torch.equal(y, y_ref), (y - y_ref).abs().max().item()
    dy = torch.randn_like(y_ref)
    y.backward(dy)
    y_ref.backward(dy)
    if k <= s:
        # disjoint windows: exactly one contribution per input — bitwise
        assert torch.equal(x1.grad, x2.grad), \
            (x1.grad - x2.grad).abs().max().item()
    else:
        # overlapping windows: torch's atomic scatter adds in arbitrary
        # order vs our fixed gather order — last-ulp differences allowed
        tol = 1e-6 if dtype == torch.float32 else 1e-2
        assert torch.allclose(x1.grad.float(), x2.grad.float(),
                              atol=tol, rtol=tol), \
            (x1.grad - x2.grad).abs().max().item()


def test_resnet50_stem_pool_converted_and_trains():
    from mgwfbp_amd import models
    from mgwfbp_amd.kernels.pooling import convert_maxpool, MgxMaxPool2d
    net = models.resnet50(num_classes=10).cuda().to(
        memory_format=torch.channels_last)
    convert_maxpool(net)
    assert any(isinstance(m, MgxMaxPool2d) for m in net.modules())
    x = torch.randn(2, 3, 64, 64, device='cuda').to(
        memory_format=torch.channels_last)
    with torch.autocast('cuda', torch.bfloat16):
        loss = net(x).sum()
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


def test_vgg_pools_converted(capsys):
    from mgwfbp_amd import models
    from mgwfbp_amd.kernels.pooling import convert_maxpool, MgxMaxPool2d
    net = models.vgg16i(num_classes=10).cuda().to(
        memory_format=torch.channels_last)
    convert_maxpool(net)
    n = sum(1 for m in net.modules() if isinstance(m, MgxMaxPool2d))
    assert n == 5, n


def test_cpu_fallback_passthrough():
    """On CPU (or odd configs) the module must behave exactly like
    nn.MaxPool2d."""
    from mgwfbp_amd.kernels.pooling import MgxMaxPool2d
    x = torch.randn(2, 8, 10, 10)
    ours = MgxMaxPool2d(2, stride=2)
    ref = nn.MaxPool2d(2, stride=2)
    assert torch.equal(ours(x), ref(x))
