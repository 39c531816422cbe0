"""Profiling: layerwise backward timing + all-reduce alpha/beta sweep.

Reference equivalents (/root/reference/profiling.py):
- ``Profiling`` (:13-92): per-parameter backward hooks that
  ``torch.cuda.synchronize()`` inside EVERY hook — serializing backward
  while profiling. Here hipEvents (torch.cuda.Event on ROCm) are recorded
  per hook and read once per pass, so the profiled backward keeps its
  natural overlap (SURVEY.md §5.1 MI355X note).
- ``benchmark(trainer)`` (:95-147): 5 warmup + 50 timed fwd+bwd passes;
  returns (seq_layernames, layerwise_times, sizes) in FORWARD order for
  the solver.
- ``CommunicationProfiler`` (:150-183): all-reduce latency sweep for the
  alpha/beta fit. The reference swept 32KB-2MB (10GbE scale); xGMI needs
  the sweep extended to 256MB to pin beta at ~6.6e-12 s/B scale.
"""
from __future__ import annotations

import time

import torch

from .settings import logger


class Profiling:
    """Layerwise backward profiler via post-accumulate-grad hooks."""

    def __init__(self, model):
        if isinstance(model, torch.nn.Module) is False:
            raise ValueError('expected a torch.nn.Module')
        self._model = model
        self._use_events = torch.cuda.is_available()
        self._hook_handles = []
        self._named = [(k, v) for k, v in model.named_parameters()
                       if v.requires_grad]
        self._backward_seq_keys = []       # actual backward firing order
        self._backward_seq_sizes = []
        self._times_per_key = {k: [] for k, _ in self._named}
        self._recording = False
        self._pass_records = []            # (name, event-or-walltime)
        self._start_event = None
        self._start_time = None
        self._seq_locked = False
        for name, p in self._named:
            self._hook_handles.append(
                p.register_post_accumulate_grad_hook(self._make_hook(name, p)))

    def _make_hook(self, name, p):
        def hook(_):
            if not self._recording:
                if not self._seq_locked:
                    # first (warmup) pass establishes the backward order
                    if name not in self._backward_seq_keys:
                        self._backward_seq_keys.append(name)
                        self._backward_seq_sizes.append(p.numel())
                    if len(self._backward_seq_keys) == len(self._named):
                        self._seq_locked = True
                return
            if self._use_events:
                ev = torch.cuda.Event(enable_timing=True)
                ev.record()
                self._pass_records.append((name, ev))
            else:
                self._pass_records.append((name, time.time()))
        return hook

    def start(self):
        self._recording = True
        self._pass_records = []
        if self._use_events:
            self._start_event = torch.cuda.Event(enable_timing=True)
            self._start_event.record()
        else:
            self._start_time = time.time()

    def finish_pass(self):
        """After loss.backward(): convert this pass's stamps to per-layer
        deltas (cumulative differences, reference profiling.py:70-89)."""
        self._recording = False
        if not self._pass_records:
            return
        if self._use_events:
            torch.cuda.synchronize()
            stamps = [(n, self._start_event.elapsed_time(ev) / 1e3)
                      for n, ev in self._pass_records]
        else:
            stamps = [(n, t - self._start_time)
                      for n, t in self._pass_records]
        prev = 0.0
        for name, t in stamps:
            self._times_per_key[name].append(t - prev)
            prev = t
        self._pass_records = []

    def get_layerwise_times(self):
        """Mean per-layer backward time, in backward order."""
        keys = self._backward_seq_keys
        times = []
        for k in keys:
            samples = self._times_per_key[k]
            times.append(sum(samples) / len(samples) if samples else 0.0)
        return keys, times, list(self._backward_seq_sizes)

    def stop(self):
        for h in self._hook_handles:
            h.remove()
        self._hook_handles = []


def benchmark(trainer, num_warmup=5, num_iters=50):
    """Profile per-layer backward times of trainer's model.

    Returns (seq_layernames, layerwise_times, sizes) in FORWARD order
    (reversed backward order), the contract the solver and
    DistributedOptimizer expect (reference profiling.py:147,
    dist_trainer.py:45).
    """
    p = Profiling(trainer.net)
    for i in range(num_warmup + num_iters):
        inputs, labels = trainer.fetch_data()
        hidden = None
        if trainer.dnn in ('lstm',):
            hidden = trainer.net.init_hidden()
        if i >= num_warmup:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
        trainer.net.zero_grad(set_to_none=False)
        # profile under the SAME autocast the training loop uses so the
        # solver's layer times (and MIOpen's tuned configs) match the
        # dtype that actually runs
        with trainer.autocast():
            if trainer.dnn == 'lstm':
                outputs, hidden = trainer.net(inputs, hidden)
                loss = trainer.criterion(
                    outputs.contiguous().view(-1, trainer.net.ntokens),
                    labels.contiguous().view(-1))
            elif trainer.dnn == 'lstman4':
                outputs, output_sizes = trainer.net(inputs[0], inputs[1])
                loss = trainer.ctc_loss(outputs, labels[0], output_sizes,
                                        labels[1])
            else:
                outputs = trainer.net(inputs)
                loss = trainer.criterion(outputs, labels)
        if i >= num_warmup:
            p.start()
        loss.backward()
        if i >= num_warmup:
            p.finish_pass()
    seq_keys, times, sizes = p.get_layerwise_times()
    p.stop()
    trainer.net.zero_grad(set_to_none=False)
    # return in forward order
    return seq_keys[::-1], times[::-1], sizes[::-1]


class CommunicationProfiler:
    """All-reduce latency sweep feeding the alpha/beta fit."""

    def __init__(self, allreduce_fn, synchronize_fn, sizes=None):
        self.allreduce_fn = allreduce_fn
        self.synchronize_fn = synchronize_fn
        self.sizes = sizes

    def benchmark(self, num_iters=10, dtype=torch.float32):
        if self.sizes is None:
            small = [1024 * i for i in (1, 2, 4, 8, 16, 32, 64, 128)]
            large = [2 ** k for k in range(18, 27)]   # 256K .. 64M elems
            self.sizes = small + large
        device = ('cuda:%d' % torch.cuda.current_device()
                  if torch.cuda.is_available() else 'cpu')
        times = []
        for n in self.sizes:
            data = torch.randn(n, dtype=dtype, device=device)
            # warmup
            for _ in range(3):
                self.synchronize_fn(self.allreduce_fn(data))
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            stime = time.time()
            for _ in range(num_iters):
                self.synchronize_fn(self.allreduce_fn(data))
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            times.append((time.time() - stime) / num_iters)
        logger.debug('comm sweep: %s', list(zip(self.sizes, times)))
        return self.sizes, times

    def benchmark_host_overhead(self, num_calls=100, numel=262144,
                                dtype=torch.float32):
        """Mean HOST time to enqueue one async all-reduce.

        Measured launch-to-launch WITHOUT waiting: this is the cost each
        collective puts on the host thread (python wrapper + RCCL
        enqueue) even when the device side overlaps perfectly — the
        per-call constant the MG-WFBP solver amortizes by merging
        (the serialized sweep above hides it inside
        one launch+completion round-trip).
        """
        device = ('cuda:%d' % torch.cuda.current_device()
                  if torch.cuda.is_available() else 'cpu')
        data = torch.randn(numel, dtype=dtype, device=device)
        # warmup, fully drained
        for _ in range(5):
            self.synchronize_fn(self.allreduce_fn(data))
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        handles = []
        stime = time.time()
        for _ in range(num_calls):
            handles.append(self.allreduce_fn(data))
        host_per_call = (time.time() - stime) / num_calls
        for h in handles:
            self.synchronize_fn(h)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        logger.debug('host enqueue overhead: %.2f us/call',
                     host_per_call * 1e6)
        return host_per_call
