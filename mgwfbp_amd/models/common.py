"""Shared model building blocks."""
import torch.nn as nn
import torch.nn.functional as F


class BNReLU(nn.Module):
    """BatchNorm2d + ReLU as ONE unit.

    Keeping the pair in a single module lets the MI355X fused BN+ReLU
    kernel (kernels/batchnorm.py convert_batchnorm) replace both ops
    with one normalize pass forward and a recomputed-sign gated
    backward — eliminating the separate relu fwd/bwd activation passes
    (~2 ms/step on ResNet-50 bs128).
    """

    def __init__(self, num_features):
        super().__init__()
        self.bn = nn.BatchNorm2d(num_features)

    def forward(self, x):
        y = self.bn(x)
        if getattr(self.bn, 'fuse_relu', False):
            return y          # relu applied inside the fused kernel
        return F.relu(y, inplace=True)


class BNAddReLU(nn.Module):
    """BatchNorm2d + residual add + ReLU as ONE unit.

    The ResNet block epilogue ``relu(bn(conv(x)) + identity)``: keeping
    the triple in one module lets the MI355X fused kernel replace the
    whole chain with one normalize pass forward (y = relu(bn(x)+res),
    residual loaded alongside x — saving the bn-out store, the add's
    read/read/write and the relu's read/write: 4 full-tensor HBM passes)
    and a recomputed-sign gated backward that emits the residual
    gradient in the same dx pass (kernels/bn_kernels.hip, VERDICT r01
    item 7).
    """

    def __init__(self, num_features):
        super().__init__()
        self.bn = nn.BatchNorm2d(num_features)

    def forward(self, x, residual):
        bn = self.bn
        fused = getattr(bn, 'forward_add_relu', None)
        if fused is not None and getattr(bn, 'fuse_relu', False):
            return fused(x, residual)
        return F.relu(bn(x) + residual, inplace=True)
