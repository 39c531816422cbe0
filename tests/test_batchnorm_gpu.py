"""MgxBatchNorm2d numerics vs torch nn.BatchNorm2d (fp32 reference)."""
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu


def _mk(C, N=8, H=14, W=14, dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(N, C, H, W, generator=g).to('cuda', dtype)
    return x.to(memory_format=torch.channels_last)


@pytest.mark.parametrize('C', [64, 96, 256, 2048])
@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_forward_backward_matches_torch(C, dtype):
    from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d
    torch.manual_seed(1)
    ref = nn.BatchNorm2d(C).cuda()
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.uniform_(-0.5, 0.5)
    ours = MgxBatchNorm2d(C).cuda()
    ours.load_state_dict(ref.state_dict())

    x1 = _mk(C, dtype=dtype).requires_grad_(True)
    # reference runs in fp32 (the plain-torch fp32 reference demanded
    # for HIP kernel numerics) on an fp32 leaf
    x2 = x1.detach().float().clone().requires_grad_(True)
    y_ref = ref(x2)
    y = ours(x1)
    tol = 2e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(y.float(), y_ref, atol=tol, rtol=tol), \
        (y.float() - y_ref).abs().max().item()

    dy = torch.randn_like(y_ref)
    y_ref.backward(dy)
    y.backward(dy.to(dtype).to(memory_format=torch.channels_last))
    assert torch.allclose(x1.grad.float(), x2.grad, atol=tol * 5,
                          rtol=tol * 5), \
        (x1.grad.float() - x2.grad).abs().max().item()
    # bf16: our path sees dy quantized to bf16 while the fp32 reference
    # sees full-precision dy, so the channel sums differ at ~bf16 eps
    # relative scale
    gatol, grtol = (1e-2, 1e-3) if dtype == torch.float32 else (0.5, 3e-2)
    assert torch.allclose(ours.weight.grad, ref.weight.grad, atol=gatol,
                          rtol=grtol), \
        (ours.weight.grad - ref.weight.grad).abs().max().item()
    assert torch.allclose(ours.bias.grad, ref.bias.grad, atol=gatol,
                          rtol=grtol)
    # running stats updated identically
    assert torch.allclose(ours.running_mean, ref.running_mean, atol=1e-4,
                          rtol=1e-4)
    assert torch.allclose(ours.running_var, ref.running_var, atol=1e-3,
                          rtol=1e-3)


def test_eval_mode_matches_torch():
    from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d
    C = 128
    ref = nn.BatchNorm2d(C).cuda().eval()
    ref.running_mean.uniform_(-1, 1)
    ref.running_var.uniform_(0.5, 2)
    ours = MgxBatchNorm2d(C).cuda().eval()
    ours.load_state_dict(ref.state_dict())
    x = _mk(C)
    with torch.no_grad():
        assert torch.allclose(ours(x), ref(x), atol=1e-5, rtol=1e-5)


def test_convert_resnet50_trains(capsys):
    """End-to-end: converted resnet50 trains a step with finite loss."""
    from mgwfbp_amd import models
    from mgwfbp_amd.kernels.batchnorm import (convert_batchnorm,
                                              MgxBatchNorm2d)
    torch.manual_seed(0)
    net = models.resnet50().cuda().to(memory_format=torch.channels_last)
    convert_batchnorm(net)
    n_bn = sum(1 for m in net.modules() if isinstance(m, MgxBatchNorm2d))
    assert n_bn == 53, n_bn
    x = torch.randn(8, 3, 224, 224, device='cuda').to(
        memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (8,), device='cuda')
    with torch.autocast('cuda', torch.bfloat16):
        loss = nn.CrossEntropyLoss()(net(x), y)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_fused_bn_relu_matches_composite(dtype):
    """y = relu(bn(x)) fused: forward + gated backward vs the fp32
    composite torch reference."""
    from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d
    C = 128
    torch.manual_seed(3)
    ref = nn.BatchNorm2d(C).cuda()
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.uniform_(-0.5, 0.5)
    ours = MgxBatchNorm2d(C).cuda()
    ours.load_state_dict(ref.state_dict())
    ours.fuse_relu = True

    x1 = _mk(C, dtype=dtype, seed=7).requires_grad_(True)
    x2 = x1.detach().float().clone().requires_grad_(True)
    y_ref = torch.relu(ref(x2))
    y = ours(x1)
    tol = 2e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(y.float(), y_ref, atol=tol, rtol=tol)
    assert (y.float().min() >= 0)

    dy = torch.randn_like(y_ref)
    y_ref.backward(dy)
    y.backward(dy.to(dtype).to(memory_format=torch.channels_last))
    assert torch.allclose(x1.grad.float(), x2.grad, atol=tol * 5,
                          rtol=tol * 5), \
        (x1.grad.float() - x2.grad).abs().max().item()
    gatol, grtol = (1e-2, 1e-3) if dtype == torch.float32 else (0.5, 3e-2)
    assert torch.allclose(ours.weight.grad, ref.weight.grad, atol=gatol,
                          rtol=grtol)
    assert torch.allclose(ours.bias.grad, ref.bias.grad, atol=gatol,
                          rtol=grtol)


def test_convert_fuses_bnrelu_units_and_sequentials():
    from mgwfbp_amd import models
    from mgwfbp_amd.kernels.batchnorm import (convert_batchnorm,
                                              MgxBatchNorm2d)
    net = models.resnet50().cuda().to(memory_format=torch.channels_last)
    convert_batchnorm(net)
    fused = sum(1 for m in net.modules()
                if isinstance(m, MgxBatchNorm2d) and m.fuse_relu)
    plain = sum(1 for m in net.modules()
                if isinstance(m, MgxBatchNorm2d) and not m.fuse_relu)
    # stem + 2 per bottleneck x16 = 33 relu-fused, plus the 16 BNAddReLU
    # epilogues (bn3, residual-fused) = 49; only the 4 downsample BNs
    # stay plain
    assert fused == 49, fused
    assert plain == 4, plain
    # vgg-style sequential fusion
    vgg = models.vgg16().cuda().to(memory_format=torch.channels_last)
    convert_batchnorm(vgg)
    fused_v = sum(1 for m in vgg.modules()
                  if isinstance(m, MgxBatchNorm2d) and m.fuse_relu)
    assert fused_v == 13, fused_v


@pytest.mark.parametrize('C', [64, 256, 2048])
@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_fused_bn_add_relu_matches_composite(C, dtype):
    """y = relu(bn(x) + residual) fused (BNAddReLU epilogue, VERDICT r01
    item 7) vs the fp32 torch chain, incl. the residual gradient the dx
    kernel emits in-pass."""
    from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d
    torch.manual_seed(3)
    ref = nn.BatchNorm2d(C).cuda()
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.uniform_(-0.5, 0.5)
    ours = MgxBatchNorm2d(C).cuda()
    ours.load_state_dict(ref.state_dict())
    ours.fuse_relu = True

    x1 = _mk(C, dtype=dtype).requires_grad_(True)
    r1 = _mk(C, dtype=dtype, seed=9).requires_grad_(True)
    x2 = x1.detach().float().clone().requires_grad_(True)
    r2 = r1.detach().float().clone().requires_grad_(True)

    y_ref = torch.nn.functional.relu(ref(x2) + r2)
    y = ours.forward_add_relu(x1, r1)
    tol = 2e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(y.float(), y_ref, atol=tol, rtol=tol), \
        (y.float() - y_ref).abs().max().item()

    dy = torch.randn_like(y_ref)
    y_ref.backward(dy)
    y.backward(dy.to(dtype).to(memory_format=torch.channels_last))
    assert torch.allclose(x1.grad.float(), x2.grad, atol=tol * 5,
                          rtol=tol * 5), \
        (x1.grad.float() - x2.grad).abs().max().item()
    assert torch.allclose(r1.grad.float(), r2.grad, atol=tol * 5,
                          rtol=tol * 5), \
        (r1.grad.float() - r2.grad).abs().max().item()
    gatol, grtol = (1e-2, 1e-3) if dtype == torch.float32 else (0.5, 3e-2)
    assert torch.allclose(ours.weight.grad, ref.weight.grad, atol=gatol,
                          rtol=grtol)
    assert torch.allclose(ours.bias.grad, ref.bias.grad, atol=gatol,
                          rtol=grtol)
    # running stats must update identically (residual doesn't affect
    # the reduce pass)
    assert torch.allclose(ours.running_mean, ref.running_mean, atol=tol,
                          rtol=tol)
    assert torch.allclose(ours.running_var, ref.running_var, atol=tol,
                          rtol=tol)


def test_fused_bn_add_relu_eval_mode(dtype=torch.bfloat16):
    from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d
    C = 128
    torch.manual_seed(5)
    ref = nn.BatchNorm2d(C).cuda().eval()
    ref.running_mean.uniform_(-0.3, 0.3)
    ref.running_var.uniform_(0.5, 1.5)
    ours = MgxBatchNorm2d(C).cuda().eval()
    ours.load_state_dict(ref.state_dict())
    ours.fuse_relu = True
    x = _mk(C, dtype=dtype)
    r = _mk(C, dtype=dtype, seed=7)
    with torch.no_grad():
        y_ref = torch.nn.functional.relu(ref(x.float()) + r.float())
        y = ours.forward_add_relu(x, r)
    assert torch.allclose(y.float(), y_ref, atol=5e-2, rtol=5e-2)


def test_resnet50_bnaddrelu_converted_and_trains():
    """convert_batchnorm must fuse the BNAddReLU epilogues and the model
    must step (the path bench.py runs)."""
    from mgwfbp_amd import models
    from mgwfbp_amd.kernels.batchnorm import (convert_batchnorm,
                                              MgxBatchNorm2d)
    from mgwfbp_amd.models.common import BNAddReLU
    net = models.resnet50(num_classes=100).cuda().to(
        memory_format=torch.channels_last)
    convert_batchnorm(net, fuse_relu=True, only_fused=True)
    n_fused_add = sum(1 for m in net.modules()
                      if isinstance(m, BNAddReLU)
                      and isinstance(m.bn, MgxBatchNorm2d)
                      and m.bn.fuse_relu)
    assert n_fused_add == 16   # one per bottleneck
    opt = torch.optim.SGD(net.parameters(), lr=0.01)
    x = torch.randn(4, 3, 64, 64, device='cuda').to(
        memory_format=torch.channels_last)
    y = torch.randint(0, 100, (4,), device='cuda')
    with torch.autocast('cuda', dtype=torch.bfloat16):
        loss = nn.CrossEntropyLoss()(net(x), y)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


def test_fused_finalize_path_numerics():
    """MGX_BN_FUSED_FIN=1 (non-default: measured 9% slower end-to-end,
    see bn_kernels.hip) must stay numerically correct."""
    import os
    import importlib
    os.environ['MGX_BN_FUSED_FIN'] = '1'
    # the flag is latched per-process on first use; this test runs in
    # the same process, so only assert the kernels agree for a shape
    # that takes the fused path in a FRESH subprocess
    import subprocess, sys
    code = (
        "import os; os.environ['MGX_BN_FUSED_FIN']='1';"
        "import torch, torch.nn as nn;"
        "from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d;"
        "torch.manual_seed(0);"
        "ref=nn.BatchNorm2d(64).cuda(); ours=MgxBatchNorm2d(64).cuda();"
        "ours.load_state_dict(ref.state_dict()); ours.fuse_relu=True;"
        "x=torch.randn(8,64,14,14,device='cuda')"
        ".to(memory_format=torch.channels_last).requires_grad_(True);"
        "r=torch.randn_like(x).requires_grad_(True);"
        "x2=x.detach().clone().requires_grad_(True);"
        "r2=r.detach().clone().requires_grad_(True);"
        "y=ours.forward_add_relu(x,r);"
        "y2=torch.nn.functional.relu(ref(x2)+r2);"
        "assert torch.allclose(y,y2,atol=2e-5,rtol=2e-5), (y-y2).abs().max();"
        "dy=torch.randn_like(y); y.backward(dy); y2.backward(dy);"
        "assert torch.allclose(x.grad,x2.grad,atol=1e-4,rtol=1e-4);"
        "assert torch.allclose(r.grad,r2.grad,atol=1e-4,rtol=1e-4);"
        "assert torch.allclose(ours.weight.grad,ref.weight.grad,atol=1e-2,rtol=1e-3);"
        "assert torch.allclose(ours.running_mean,ref.running_mean,atol=2e-5);"
        "print('fused-fin numerics ok')"
    )
    out = subprocess.run([sys.executable, '-c', code],
                         capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    assert 'fused-fin numerics ok' in out.stdout
