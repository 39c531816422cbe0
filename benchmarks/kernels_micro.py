"""Achieved-bandwidth micro-benchmark for the gfx950 hot-path kernels.

fused SGD / pack / unpack / l2norm over a ResNet-50-sized parameter set
(161 tensors, 25.56M elements). Every kernel is HBM-bound; the score is
achieved GB/s vs the ~6300 GB/s practical ceiling
(MI355X_MICROARCH.md).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from mgwfbp_amd import models  # noqa: E402
from mgwfbp_amd import kernels as K  # noqa: E402


def timed(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t) / iters


def main():
    net = models.resnet50().cuda()
    params = [p.detach().clone() for p in net.parameters()]
    grads = [torch.randn_like(p) for p in params]
    moms = [torch.zeros_like(p) for p in params]
    wds = [1e-4] * len(params)
    n = sum(p.numel() for p in params)
    print('%d tensors, %.2fM elements' % (len(params), n / 1e6))

    fused = K.FusedSGD(params, grads, moms, wds, momentum=0.9)
    t = timed(lambda: fused.step(0.1))
    # bytes: read p,g,m + write p,m = 5 passes fp32
    print('fused_sgd:   %7.1f us  %7.0f GB/s' % (t * 1e6,
                                                 n * 4 * 5 / t / 1e9))

    offsets = []
    off = 0
    for p in params:
        offsets.append(off)
        off += (p.numel() + 63) // 64 * 64
    flat32 = torch.zeros(off, dtype=torch.float32, device='cuda')
    flat16 = torch.zeros(off, dtype=torch.bfloat16, device='cuda')
    table = K.PackTable(grads, offsets)
    t = timed(lambda: table.pack(flat32))
    print('pack fp32:   %7.1f us  %7.0f GB/s' % (t * 1e6,
                                                 n * 4 * 2 / t / 1e9))
    t = timed(lambda: table.pack(flat16))
    print('pack bf16:   %7.1f us  %7.0f GB/s' % (t * 1e6,
                                                 n * 6 / t / 1e9))
    t = timed(lambda: table.unpack(flat16))
    print('unpack bf16: %7.1f us  %7.0f GB/s' % (t * 1e6,
                                                 n * 6 / t / 1e9))
    t = timed(lambda: table.l2norm_sq())
    print('l2norm_sq:   %7.1f us  %7.0f GB/s' % (t * 1e6,
                                                 n * 4 / t / 1e9))
    t = timed(lambda: K.scale_inplace(flat32, 0.5))
    print('scale:       %7.1f us  %7.0f GB/s' % (t * 1e6,
                                                 off * 4 * 2 / t / 1e9))


if __name__ == '__main__':
    main()
