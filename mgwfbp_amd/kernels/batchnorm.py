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
                eps, relu):
        ext = _load()
        y, mean, invstd = ext.bn_fwd_train(
            x, weight, bias, running_mean, running_var, momentum, eps,
            relu)
        ctx.relu = relu
        ctx.save_for_backward(x, weight, bias, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load()
        x, weight, bias, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx, dgamma, dbeta = ext.bn_bwd(dy, x, mean, invstd, weight, bias,
                                       ctx.relu)
        return dx, dgamma, dbeta, None, None, None, None, None


class _BatchNormAddReLUFunc(torch.autograd.Function):
    """y = relu(bn(x) + residual) in one normalize pass; backward gates
    dy by the recomputed post-add sign and emits the residual gradient
    in the same dx kernel (bn_kernels.hip ADD path)."""

    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                momentum, eps):
        ext = _load()
        y, mean, invstd = ext.bn_fwd_train(
            x, weight, bias, running_mean, running_var, momentum, eps,
            True, residual)
        ctx.save_for_backward(x, residual, weight, bias, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load()
        x, residual, weight, bias, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx, dgamma, dbeta, dres = ext.bn_bwd(dy, x, mean, invstd, weight,
                                             bias, True, residual)
        return (dx, dres, dgamma, dbeta, None, None, None, None)


class MgxBatchNorm2d(nn.BatchNorm2d):
    fuse_relu = False

    def _use_hip_path(self, x):
        return (x.is_cuda and x.dim() == 4 and x.size(1) % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)
                and self.affine and self.track_running_stats
                and _load() is not False)

    def forward(self, x):
        if not self._use_hip_path(x):
            y = super().forward(x)
            if self.fuse_relu:
                y = torch.nn.functional.relu(y, inplace=True)
            return y
        if self.training:
            momentum = self.momentum
            if momentum is None:
                # cumulative-average mode is the only consumer of the
                # counter; with a fixed momentum the per-layer
                # .add_(1) is 53 dead launches per resnet50 step
                self.num_batches_tracked.add_(1)
                momentum = 1.0 / float(self.num_batches_tracked)
            return _BatchNormFunc.apply(x, self.weight, self.bias,
                                        self.running_mean,
                                        self.running_var, momentum,
                                        self.eps, self.fuse_relu)
        ext = _load()
        return ext.bn_fwd_eval(x, self.weight, self.bias,
                               self.running_mean, self.running_var,
                               self.eps, self.fuse_relu)

    def forward_add_relu(self, x, residual):
        """Fused y = relu(bn(x) + residual) (the ResNet post-add
        activation). Falls back to the unfused chain when the HIP path
        can't run (CPU, odd layout)."""
        if not (self._use_hip_path(x) and residual.is_cuda
                and residual.shape == x.shape):
            y = nn.BatchNorm2d.forward(self, x)
            return torch.nn.functional.relu(y + residual, inplace=True)
        if residual.dtype != x.dtype:
            # autocast leaves the downsample BN's output in fp32 while
            # the conv output is bf16: one cast beats losing the fusion
            residual = residual.to(x.dtype)
        residual = residual.contiguous(
            memory_format=torch.channels_last)
        if self.training:
            momentum = self.momentum
            if momentum is None:
                self.num_batches_tracked.add_(1)
                momentum = 1.0 / float(self.num_batches_tracked)
            return _BatchNormAddReLUFunc.apply(
                x, residual, self.weight, self.bias, self.running_mean,
                self.running_var, momentum, self.eps)
        ext = _load()
        return ext.bn_fwd_eval(x, self.weight, self.bias,
                               self.running_mean, self.running_var,
                               self.eps, True, residual)


def _mgx_from(child, fuse_relu=False):
    bn = MgxBatchNorm2d(child.num_features, eps=child.eps,
                        momentum=child.momentum, affine=child.affine,
                        track_running_stats=child.track_running_stats)
    bn = bn.to(device=child.weight.device if child.affine else 'cpu',
               dtype=child.weight.dtype if child.affine
               else torch.float32)
    bn.load_state_dict(child.state_dict())
    bn.fuse_relu = fuse_relu
    return bn


def convert_batchnorm(module, fuse_relu=True, only_fused=False):
    """Recursively swap nn.BatchNorm2d -> MgxBatchNorm2d (in place),
    keeping parameters, buffers and config.

    With ``fuse_relu``, BN+ReLU pairs become ONE fused op:
    - a ``models.common.BNReLU`` unit gets a relu-fused MgxBatchNorm2d;
    - inside an nn.Sequential, a (BatchNorm2d, ReLU) pair becomes
      (fused MgxBatchNorm2d, Identity) — covers VGG-style stacks.

    ``only_fused`` leaves plain (no-relu) BatchNorm2d on the MIOpen path
    and converts only the fusable pairs — MIOpen's NHWC BN is slightly
    faster per-kernel on large layers, so converting everything costs
    more than the relu fusion saves on big models (measured A/B,
    profiles/README.md).
    """
    from ..models.common import BNReLU, BNAddReLU
    children = list(module.named_children())
    names = [n for n, _ in children]
    for idx, (name, child) in enumerate(children):
        if isinstance(child, (BNReLU, BNAddReLU)):
            if type(child.bn) is nn.BatchNorm2d and fuse_relu:
                child.bn = _mgx_from(child.bn, fuse_relu=True)
        elif type(child) is nn.BatchNorm2d:
            relu_next = (fuse_relu and isinstance(module, nn.Sequential)
                         and idx + 1 < len(children)
                         and isinstance(children[idx + 1][1], nn.ReLU))
            if relu_next:
                setattr(module, name, _mgx_from(child, fuse_relu=True))
                setattr(module, names[idx + 1], nn.Identity())
            elif not only_fused:
                setattr(module, name, _mgx_from(child, fuse_relu=False))
        else:
            convert_batchnorm(child, fuse_relu=fuse_relu,
                              only_fused=only_fused)
    return module
