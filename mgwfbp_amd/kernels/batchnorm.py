"""MgxBatchNorm2d — NHWC BatchNorm on the gfx950 HIP kernels.

Drop-in replacement for nn.BatchNorm2d (same parameters/buffers, so
checkpoints interchange). On GPU with channels_last inputs it runs the
bf16-native kernels in bn_kernels.hip (one reduce + one finalize + one
normalize forward; one reduce + one elementwise backward) instead of
MIOpen's six fp32 kernels plus autocast cast pairs — the top hot spot in
the ResNet-50 profile (profiles/resnet50_n1_bf16_kernel_stats.md).
"""
from __future__ import annotations

import importlib

import torch
import torch.nn as nn

_ext = None


def _load():
    global _ext
    if _ext is None:
        try:
            _ext = importlib.import_module(
                'mgwfbp_amd.kernels.mgx_bn_ext')
        except ImportError:
            _ext = False
    return _ext


class _BatchNormFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum,
                eps):
        ext = _load()
        y, mean, invstd = ext.bn_fwd_train(
            x, weight, bias, running_mean, running_var, momentum, eps,
            False)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load()
        x, weight, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx, dgamma, dbeta = ext.bn_bwd(dy, x, mean, invstd, weight)
        return dx, dgamma, dbeta, None, None, None, None


class MgxBatchNorm2d(nn.BatchNorm2d):
    def _use_hip_path(self, x):
        return (x.is_cuda and x.dim() == 4 and x.size(1) % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)
                and self.affine and self.track_running_stats
                and _load() is not False)

    def forward(self, x):
        if not self._use_hip_path(x):
            return super().forward(x)
        if self.training:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            momentum = self.momentum
            if momentum is None:
                momentum = 1.0 / float(self.num_batches_tracked)
            return _BatchNormFunc.apply(x, self.weight, self.bias,
                                        self.running_mean,
                                        self.running_var, momentum,
                                        self.eps)
        ext = _load()
        return ext.bn_fwd_eval(x, self.weight, self.bias,
                               self.running_mean, self.running_var,
                               self.eps, False)


def convert_batchnorm(module):
    """Recursively swap nn.BatchNorm2d -> MgxBatchNorm2d (in place),
    keeping parameters, buffers and config."""
    for name, child in module.named_children():
        if type(child) is nn.BatchNorm2d:
            bn = MgxBatchNorm2d(child.num_features, eps=child.eps,
                                momentum=child.momentum,
                                affine=child.affine,
                                track_running_stats=
                                child.track_running_stats)
            bn = bn.to(device=child.weight.device
                       if child.affine else 'cpu',
                       dtype=child.weight.dtype if child.affine
                       else torch.float32)
            bn.load_state_dict(child.state_dict())
            setattr(module, name, bn)
        else:
            convert_batchnorm(child)
    return module
