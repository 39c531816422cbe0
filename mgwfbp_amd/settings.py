"""Global settings for the MG-WFBP MI355X framework.

Same role as the reference's ``settings.py`` (flags + shared logger +
experiment PREFIX naming; /root/reference/settings.py:7-53), but every flag
is overridable from the environment (``MGX_<NAME>``) instead of
edit-the-source.

MI355X-specific defaults:
- ``CONNECTION`` defaults to ``'xgmi'`` — intra-node xGMI point-to-point
  links (7 x ~153 GB/s per GPU), not 10GbE/56GbIB.
- ``ADAPTIVE_ABC``: measure alpha/beta online over RCCL at startup (the
  reference's dead ``_benchmark_communication`` path, promoted to default
  on GPU: reference distributed_optimizer.py:105-127 was never called).
- ``COMM_DTYPE``: dtype gradients are all-reduced in ('fp32' or 'bf16');
  replaces the reference's apex-fp16 flag (settings.py:25).
"""
import logging
import os
import socket


def _env_bool(name, default):
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() in ('1', 'true', 'yes', 'on')


def _env_str(name, default):
    return os.environ.get(name, default)


def _env_int(name, default):
    v = os.environ.get(name)
    return default if v is None else int(v)


def _env_float(name, default):
    v = os.environ.get(name)
    return default if v is None else float(v)


DEBUG = _env_bool('MGX_DEBUG', False)
WARMUP = _env_bool('MGX_WARMUP', True)          # LR warmup in first epochs
DELAY_COMM = _env_bool('MGX_DELAY_COMM', False)
CONNECTION = _env_str('MGX_CONNECTION', 'xgmi')  # 'xgmi' | '10GbE' | '56GbIB'
FP16 = _env_bool('MGX_FP16', False)              # legacy flag: comm in half precision
COMM_DTYPE = _env_str('MGX_COMM_DTYPE', 'fp16' if FP16 else 'fp32')
ADAPTIVE_MERGE = _env_bool('MGX_ADAPTIVE_MERGE', True)   # MG-WFBP solver vs threshold grouping
ADAPTIVE_ABC = _env_bool('MGX_ADAPTIVE_ABC', True)       # measure alpha/beta online on GPU
# Per-collective HOST cost (async-enqueue + hook bookkeeping, seconds)
# added to the solver's per-call constant. On xGMI the device alpha is
# O(10us) so the host launch path dominates what merging can save; the
# online sweep measures it (CommunicationProfiler.benchmark_host_overhead)
# and overrides this default. 0 disables.
ALPHA_HOST = _env_float('MGX_ALPHA_HOST', 0.0)
TENSORBOARD = _env_bool('MGX_TENSORBOARD', False)
MAX_EPOCHS = _env_int('MGX_MAX_EPOCHS', 200)
USE_HIP_KERNELS = _env_bool('MGX_USE_HIP_KERNELS', True)  # hand-written gfx950 kernels on GPU
USE_FUSED_SGD = _env_bool('MGX_USE_FUSED_SGD', True)
EXCHANGE_MODE = _env_str('MGX_EXCHANGE_MODE', 'MODEL_MG')
UPDATE_ITER = _env_int('MGX_UPDATE_ITER', 1)
# Deterministic single-stream fallback (disables comm-stream overlap) for
# debugging stream races.
DETERMINISTIC = _env_bool('MGX_DETERMINISTIC', False)

# Experiment prefix string, mirroring the reference's log-dir naming
# (reference settings.py:13-36): encodes the flags that affect a run.
PREFIX = EXCHANGE_MODE
if WARMUP:
    PREFIX += '-gwarmup'
if DELAY_COMM:
    PREFIX += '-dc'
PREFIX += '-' + CONNECTION
if COMM_DTYPE != 'fp32':
    PREFIX += '-' + COMM_DTYPE
if ADAPTIVE_MERGE:
    PREFIX += '-ada'


hostname = socket.gethostname()
logger = logging.getLogger(hostname)
if not logger.handlers:
    logger.setLevel(logging.DEBUG if DEBUG else logging.INFO)
    _formatter = logging.Formatter(
        '%(asctime)s [%(filename)s:%(lineno)d] %(levelname)s %(message)s')
    _handler = logging.StreamHandler()
    _handler.setFormatter(_formatter)
    logger.addHandler(_handler)


def add_file_handler(logfile):
    """Attach a per-run file handler (reference dist_trainer.py:138-141)."""
    os.makedirs(os.path.dirname(logfile), exist_ok=True)
    hdl = logging.FileHandler(logfile)
    hdl.setFormatter(_formatter)
    logger.addHandler(hdl)
    return hdl
