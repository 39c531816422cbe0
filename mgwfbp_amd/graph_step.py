"""hipGraph-captured training step.

Captures the ENTIRE training step — arena zero_grad, autocast forward,
backward (whose post-accumulate hooks enqueue the merged-group RCCL
all-reduces on the comm stream), device-side synchronize waits, and the
fused multi-tensor SGD — into one hipGraph, then replays it per
iteration. Replay eliminates every per-step host launch (the resnet20
CIFAR config is launch-bound: hundreds of small kernels per 4.5 ms
step) while performing identical work; fresh batch data is copied into
the static input buffers before each replay.

This is the MI355X answer to a tracing compiler: HIP streams provide
the comm/compute overlap, hipGraphs remove the launch overhead
(BASELINE.json north star).

The fused SGD reads its learning rate from a 1-element device buffer,
so the LR schedule keeps working across replays (``set_lr`` before each
``step``) — no re-capture needed. Loss values are read from the static
loss tensor after replay when needed.
"""
from __future__ import annotations

import torch

from .settings import logger


class GraphedTrainStep:
    def __init__(self, trainer, optimizer, warmup=3):
        if not torch.cuda.is_available():
            raise RuntimeError('hipGraph capture needs a GPU')
        if trainer.dnn in ('lstm', 'lstman4'):
            raise RuntimeError('graph capture not supported for '
                               'variable-length RNN workloads')
        self.trainer = trainer
        self.optimizer = optimizer
        x, y = trainer.fetch_data()
        self.static_x = x.clone()
        self.static_y = y.clone()
        self.static_loss = None
        # LR through a device buffer: replays follow the LR schedule
        # without re-capture (set_lr fills it before replay). Without
        # the fused SGD there is no device LR buffer — a replayed graph
        # would bake in the capture-time LR (mid warmup-ramp) and
        # silently train at a frozen tiny LR, so refuse to capture and
        # let the caller fall back to eager.
        self._fused = getattr(optimizer, '_fused_sgd', None)
        if self._fused is None:
            raise RuntimeError(
                'hipGraph capture needs the fused SGD (device LR '
                'buffer); the LR schedule would freeze at capture time '
                'otherwise. Enable MGX_USE_FUSED_SGD or run eager.')
        self._fused.enable_lr_buffer()
        self._fused.set_lr(optimizer.param_groups[0]['lr'])

        # side-stream warmup, then capture (torch.cuda.graph idiom)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                self._full_step()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._full_step()
        logger.info('training step captured as hipGraph '
                    '(%s, batch %d)', trainer.dnn, self.static_x.size(0))

    def _full_step(self):
        self.optimizer.zero_grad()
        with self.trainer.autocast():
            out = self.trainer.net(self.static_x)
            loss = self.trainer.criterion(out, self.static_y)
        loss.backward()
        self.trainer.update_model()
        self.static_loss = loss.detach()

    def set_lr(self, lr):
        if self._fused is not None:
            self._fused.set_lr(lr)

    def step(self):
        x, y = self.trainer.fetch_data()
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.graph.replay()

    def loss(self):
        return float(self.static_loss.item())
