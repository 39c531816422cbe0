"""MgxMaxPool2d — NHWC max pooling on the gfx950 HIP kernels.

Drop-in replacement for nn.MaxPool2d (square kernel/stride/padding, no
dilation/ceil_mode). torch's NHWC maxpool backward scatters through
atomics (`max_pool_backward_nhwc`: 440 us/call on VGG-16's pools — 11%
of that model's step; 311 us on resnet50's stem pool); ours saves a
window-local uint8 argmax in forward and gathers in backward — no
atomics, no zero-fill pass, deterministic (bn_kernels.hip).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .batchnorm import _load


class _MaxPoolFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, s, p):
        ext = _load()
        y, idx = ext.maxpool_fwd(x, k, s, p)
        ctx.k, ctx.s, ctx.p = k, s, p
        ctx.in_shape = x.shape
        ctx.save_for_backward(idx)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load()
        (idx,) = ctx.saved_tensors
        n, c, h, w = ctx.in_shape
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = ext.maxpool_bwd(dy, idx, n, c, h, w, ctx.k, ctx.s, ctx.p)
        return dx, None, None, None


class MgxMaxPool2d(nn.MaxPool2d):
    def _use_hip_path(self, x):
        k, s, p = self.kernel_size, self.stride, self.padding
        return (x.is_cuda and x.dim() == 4 and x.size(1) % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)
                and isinstance(k, int) and isinstance(s, int)
                and isinstance(p, int) and self.dilation == 1
                and not self.ceil_mode and not self.return_indices
                and k * k <= 255 and _load() is not False)

    def forward(self, x):
        if not self._use_hip_path(x):
            return super().forward(x)
        return _MaxPoolFunc.apply(x, self.kernel_size, self.stride,
                                  self.padding)


def convert_maxpool(module):
    """Recursively swap nn.MaxPool2d -> MgxMaxPool2d (in place).
    Pool layers have no parameters, so this is config-only."""
    for name, child in module.named_children():
        if type(child) is nn.MaxPool2d:
            k = child.kernel_size
            s = child.stride if child.stride is not None else k
            setattr(module, name,
                    MgxMaxPool2d(k, stride=s, padding=child.padding,
                                 dilation=child.dilation,
                                 ceil_mode=child.ceil_mode))
        else:
            convert_maxpool(child)
    return module
