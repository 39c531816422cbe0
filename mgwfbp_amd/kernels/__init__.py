"""Loaders and Python wrappers for the in-tree HIP extensions.

Two extensions, built IN-TREE by ``setup.py build_ext --inplace`` (or
``__graft_entry__.build()``) so the .so files travel with the repo
snapshot to GPU boxes:

- ``mgx_kernels_ext``  (mgwfbp_amd/kernels/mgx_kernels.hip): fused SGD,
  pack/unpack(+cast/scale), L2 norm — the gfx950 hot-path kernels.
- ``mgx_comm_ext``     (mgwfbp_amd/comm/comm_core.hip): RCCL comm core.

Policy: on a GPU box the HIP path MUST run — ops raise loudly if the
extension is missing while CUDA is available (no silent eager fallback).
On CPU (the test tier) pure-torch reference implementations are used; the
same reference implementations back the kernel numerics tests.
"""
from __future__ import annotations

import importlib

import torch

_kernels_ext = None
_comm_ext = None


class ExtensionMissing(RuntimeError):
    pass


def _try_import(modname):
    try:
        return importlib.import_module(modname)
    except ImportError as e:
        return e


def load_kernels(required=None):
    """Import the mgx_kernels extension.

    required=None: required iff a GPU is visible (fail-loud policy).
    """
    global _kernels_ext
    if _kernels_ext is not None:
        return _kernels_ext
    mod = _try_import('mgwfbp_amd.kernels.mgx_kernels_ext')
    if isinstance(mod, ImportError):
        req = torch.cuda.is_available() if required is None else required
        if req:
            raise ExtensionMissing(
                'mgx_kernels_ext (gfx950 HIP kernels) is not built but a GPU '
                'is present. Build it in-tree: python setup.py build_ext '
                '--inplace  (original error: %s)' % mod)
        return None
    _kernels_ext = mod
    return _kernels_ext


def load_comm_core():
    """Import the RCCL comm-core extension (GPU only, always required)."""
    global _comm_ext
    if _comm_ext is not None:
        return _comm_ext
    mod = _try_import('mgwfbp_amd.comm.mgx_comm_ext')
    if isinstance(mod, ImportError):
        raise ExtensionMissing(
            'mgx_comm_ext (RCCL comm core) is not built. Build it in-tree: '
            'python setup.py build_ext --inplace (original error: %s)' % mod)
    _comm_ext = mod
    return _comm_ext


def hip_available():
    """True when the HIP kernels can run here (GPU + built extension)."""
    return torch.cuda.is_available() and load_kernels(required=True) is not None


# --------------------------------------------------------------------------
# Fused multi-tensor ops with cached device-side chunk tables.
# --------------------------------------------------------------------------

class FusedSGD:
    """One-launch SGD+momentum+weight-decay over a fixed tensor list.

    Replaces the reference's per-tensor torch.optim.SGD inner loop
    (reference dl_trainer.py:244-248 / distributed_optimizer.py:431) with a
    single multi-tensor HIP kernel launch; chunk tables are built once
    (tensor addresses are stable) and cached on device.
    """

    def __init__(self, params, grads, momenta, weight_decays,
                 momentum=0.0, dampening=0.0, nesterov=False):
        self.ext = load_kernels(required=True)
        assert self.ext is not None
        self.momentum = momentum
        self.dampening = dampening
        self.nesterov = nesterov
        t_tensors, t_chunks, t_n = self.ext.build_sgd_table(
            list(params), list(grads),
            list(momenta) if momentum != 0.0 else [],
            [float(w) for w in weight_decays])
        self._t_tensors = t_tensors
        self._t_chunks = t_chunks
        self._nchunks = int(t_n.item())
        self._lr_buf = torch.empty(0)   # disabled by default

    def enable_lr_buffer(self):
        """Switch the kernel to read LR from a device buffer — required
        under hipGraph capture so the LR schedule survives replay (the
        host fills the buffer before each replay, outside the graph)."""
        if self._lr_buf.numel() == 0:
            dev = self._t_tensors.device
            self._lr_buf = torch.zeros(1, dtype=torch.float32, device=dev)
        return self._lr_buf

    def set_lr(self, lr):
        self._lr_buf.fill_(float(lr))

    def step(self, lr, grad_scale=1.0):
        if self._lr_buf.numel() and \
                not torch.cuda.is_current_stream_capturing():
            # buffer mode outside capture: keep the device LR current
            self._lr_buf.fill_(float(lr))
        self.ext.multi_tensor_sgd(self._t_tensors, self._t_chunks,
                                  self._nchunks, float(lr), self._lr_buf,
                                  self.momentum, self.dampening,
                                  self.nesterov, float(grad_scale))


class PackTable:
    """Cached descriptor/chunk table for pack/unpack/norm over one group."""

    def __init__(self, tensors, offsets):
        self.ext = load_kernels(required=True)
        assert self.ext is not None
        t_descs, t_chunks, t_n = self.ext.build_pack_table(
            list(tensors), [int(o) for o in offsets])
        self._t_descs = t_descs
        self._t_chunks = t_chunks
        self._nchunks = int(t_n.item())

    def pack(self, flat, scale=1.0):
        self.ext.multi_tensor_pack(self._t_descs, self._t_chunks,
                                   self._nchunks, flat, float(scale))

    def unpack(self, flat, scale=1.0):
        self.ext.multi_tensor_unpack(self._t_descs, self._t_chunks,
                                     self._nchunks, flat, float(scale))

    def l2norm_sq(self):
        return self.ext.l2norm_sq(self._t_descs, self._t_chunks,
                                  self._nchunks)


def scale_inplace(buf, scale):
    ext = load_kernels()
    if ext is not None and buf.is_cuda:
        ext.scale_inplace(buf, float(scale))
    else:
        buf.mul_(scale)


_clip_scratch = {}   # device -> 1-elem fp32 norm^2 scratch (stable addr
                     # so the clip is hipGraph-capturable)


def l2norm_clip_(buf, max_norm, eps=1e-6):
    """Clip ``buf`` to L2 norm <= max_norm entirely on device — no
    ``.item()`` host round-trip (the reference's clip at
    distributed_optimizer.py:380-389 host-syncs per merged tensor)."""
    ext = load_kernels()
    if ext is not None and buf.is_cuda:
        scratch = _clip_scratch.get(buf.device)
        if scratch is None:
            scratch = torch.zeros(1, dtype=torch.float32,
                                  device=buf.device)
            _clip_scratch[buf.device] = scratch
        ext.l2norm_clip_(buf, scratch, float(max_norm), float(eps))
    else:
        coef = torch.clamp(max_norm / (buf.norm(2) + eps), max=1.0)
        buf.mul_(coef)


# --------------------------------------------------------------------------
# Pure-torch reference implementations (CPU test tier + numerics checks).
# --------------------------------------------------------------------------

def sgd_reference(params, grads, momenta, weight_decays, lr, momentum=0.0,
                  dampening=0.0, nesterov=False, grad_scale=1.0):
    """fp32 reference of the fused SGD kernel (same math, per tensor)."""
    with torch.no_grad():
        for i, (p, g) in enumerate(zip(params, grads)):
            d_p = g * grad_scale + weight_decays[i] * p
            if momentum != 0.0:
                buf = momenta[i]
                buf.mul_(momentum).add_(d_p, alpha=1.0 - dampening)
                d_p = d_p.add(buf, alpha=momentum) if nesterov else buf
            p.add_(d_p, alpha=-lr)


def pack_reference(tensors, offsets, flat, scale=1.0):
    with torch.no_grad():
        for t, off in zip(tensors, offsets):
            n = t.numel()
            flat[off:off + n].copy_(t.reshape(-1).to(flat.dtype))
            if scale != 1.0:
                flat[off:off + n].mul_(scale)


def unpack_reference(tensors, offsets, flat, scale=1.0):
    with torch.no_grad():
        for t, off in zip(tensors, offsets):
            n = t.numel()
            src = flat[off:off + n].to(t.dtype)
            if scale != 1.0:
                src = src * scale
            t.reshape(-1).copy_(src)
