"""Single-shape BN kernel run for PMC counter capture."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from mgwfbp_amd.kernels.batchnorm import _load

ext = _load()
C, H, W, N = 64, 112, 112, 128
x = torch.randn(N, C, H, W, device='cuda', dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
w = torch.ones(C, device='cuda')
b = torch.zeros(C, device='cuda')
rm = torch.zeros(C, device='cuda')
rv = torch.ones(C, device='cuda')
dy = torch.randn_like(x)
for _ in range(10):
    y, mean, invstd = ext.bn_fwd_train(x, w, b, rm, rv, 0.1, 1e-5, False)
    ext.bn_bwd(dy, x, mean, invstd, w, b, False)
torch.cuda.synchronize()
print('done')
