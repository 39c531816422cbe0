"""BN+Add+ReLU fused kernels on one bottleneck-epilogue shape, for PMC
counter capture (FETCH_SIZE/WRITE_SIZE -> achieved HBM bytes vs the
analytic minimum; LDS conflicts should be zero — these kernels do not
stage through LDS on the streaming path)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from mgwfbp_amd.kernels.batchnorm import _load

ext = _load()
# layer2 bn3 epilogue: C=512, 28x28, bs128 (51.4M elems)
C, H, W, N = 512, 28, 28, 128
x = torch.randn(N, C, H, W, device='cuda', dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
res = torch.randn_like(x)
w = torch.ones(C, device='cuda')
b = torch.zeros(C, device='cuda')
rm = torch.zeros(C, device='cuda')
rv = torch.ones(C, device='cuda')
dy = torch.randn_like(x)
for _ in range(10):
    y, mean, invstd = ext.bn_fwd_train(x, w, b, rm, rv, 0.1, 1e-5, True,
                                       res)
    ext.bn_bwd(dy, x, mean, invstd, w, b, True, res)
torch.cuda.synchronize()
elems = N * C * H * W
print('done; elems=%d  fwd_norm min bytes=%d  bwd_dx min bytes=%d'
      % (elems, elems * 2 * 3, elems * 2 * 5))
