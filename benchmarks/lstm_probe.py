"""Is the fused MIOpen RNN path live? Times nn.LSTM configurations."""
import time

import torch

print('cudnn(miopen) available:', torch.backends.cudnn.is_available(),
      'version:', torch.backends.cudnn.version(), flush=True)

T, N, H = 200, 4, 800
for kwargs, tag in [
        (dict(bidirectional=True), 'bi'),
        (dict(bidirectional=False), 'uni'),
]:
    for flat in (False, True):
        lstm = torch.nn.LSTM(H, H, **kwargs).cuda()
        if flat:
            lstm.flatten_parameters()
        x = torch.randn(T, N, H, device='cuda', requires_grad=True)
        for _ in range(3):
            y, _ = lstm(x)
            y.sum().backward()
            x.grad = None
        torch.cuda.synchronize()
        t = time.time()
        iters = 5
        for _ in range(iters):
            y, _ = lstm(x)
            y.sum().backward()
            x.grad = None
        torch.cuda.synchronize()
        print('%s flat=%s: %.1f ms/iter' % (tag, flat,
                                            (time.time() - t) / iters * 1e3),
              flush=True)

# bigger batch to see batch sensitivity
for n in (4, 32, 128):
    lstm = torch.nn.LSTM(H, H, bidirectional=True).cuda()
    lstm.flatten_parameters()
    x = torch.randn(T, n, H, device='cuda')
    for _ in range(2):
        y, _ = lstm(x)
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(5):
        y, _ = lstm(x)
    torch.cuda.synchronize()
    print('fwd-only bs%d: %.1f ms' % (n, (time.time() - t) / 5 * 1e3),
          flush=True)
