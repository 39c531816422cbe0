"""Micro-benchmark: MgxBatchNorm2d vs nn.BatchNorm2d per ResNet-50 BN
shape (fwd+bwd, bf16 channels_last, cuda-event timed)."""
import sys

import torch
import torch.nn as nn

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mgwfbp_amd.kernels.batchnorm import MgxBatchNorm2d  # noqa: E402

# (C, H, W) with bs 128 — every distinct BN shape in resnet50
SHAPES = [(64, 112, 112), (64, 56, 56), (256, 56, 56), (128, 56, 56),
          (128, 28, 28), (512, 28, 28), (256, 28, 28), (256, 14, 14),
          (1024, 14, 14), (512, 14, 14), (512, 7, 7), (2048, 7, 7)]
N = 128


def time_one(mod, x, iters=20):
    dy = torch.randn_like(x)
    for _ in range(3):
        y = mod(x)
        y.backward(dy)
        x.grad = None
        mod.weight.grad = None
        mod.bias.grad = None
    torch.cuda.synchronize()
    sf = torch.cuda.Event(True)
    mf = torch.cuda.Event(True)
    eb = torch.cuda.Event(True)
    tf = tb = 0.0
    for _ in range(iters):
        sf.record()
        y = mod(x)
        mf.record()
        y.backward(dy)
        eb.record()
        torch.cuda.synchronize()
        tf += sf.elapsed_time(mf)
        tb += mf.elapsed_time(eb)
        x.grad = None
        mod.weight.grad = None
        mod.bias.grad = None
    return tf / iters, tb / iters


def time_raw(x, iters=20):
    """Raw extension calls without autograd, to isolate host overhead."""
    from mgwfbp_amd.kernels.batchnorm import _load
    ext = _load()
    C = x.size(1)
    w = torch.ones(C, device='cuda')
    b = torch.zeros(C, device='cuda')
    rm = torch.zeros(C, device='cuda')
    rv = torch.ones(C, device='cuda')
    dy = torch.randn_like(x)
    for _ in range(3):
        y, mean, invstd = ext.bn_fwd_train(x, w, b, rm, rv, 0.1, 1e-5,
                                           False)
        ext.bn_bwd(dy, x, mean, invstd, w, b, False)
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    m = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    tf = tb = 0.0
    for _ in range(iters):
        s.record()
        y, mean, invstd = ext.bn_fwd_train(x, w, b, rm, rv, 0.1, 1e-5,
                                           False)
        m.record()
        ext.bn_bwd(dy, x, mean, invstd, w, b, False)
        e.record()
        torch.cuda.synchronize()
        tf += s.elapsed_time(m)
        tb += m.elapsed_time(e)
    return tf / iters, tb / iters


def main():
    dtype = torch.bfloat16 if len(sys.argv) < 2 else \
        {'fp32': torch.float32, 'bf16': torch.bfloat16}[sys.argv[1]]
    tot_ref = tot_ours = 0.0
    print('%22s %10s %10s %10s %10s' % ('shape', 'ref fwd', 'ref bwd',
                                        'mgx fwd', 'mgx bwd'))
    for C, H, W in SHAPES:
        x = torch.randn(N, C, H, W, device='cuda', dtype=dtype) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        ref = nn.BatchNorm2d(C).cuda()
        ours = MgxBatchNorm2d(C).cuda()
        ours.load_state_dict(ref.state_dict())
        with torch.autocast('cuda', torch.bfloat16,
                            enabled=dtype == torch.bfloat16):
            rf, rb = time_one(ref, x)
            of, ob = time_one(ours, x)
        xf, xb = time_raw(x.detach())
        print('%22s %9.3f %9.3f %9.3f %9.3f | raw %6.3f %6.3f ms'
              % (str((C, H, W)), rf, rb, of, ob, xf, xb))
        tot_ref += rf + rb
        tot_ours += of + ob
    print('TOTAL per-pass: ref %.3f ms  mgx %.3f ms' % (tot_ref, tot_ours))


if __name__ == '__main__':
    main()
