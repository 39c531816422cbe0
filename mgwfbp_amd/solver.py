"""MG-WFBP merge-group solver — pure functions, no torch/hardware deps.

Implements the merged-gradient WFBP grouping of Shi et al. (INFOCOM'19 /
TPDS): given per-layer backward times and an all-reduce cost model
``t(s) = alpha + beta * s_bytes``, decide which consecutive (in backward
order) layer gradients to merge into one flat all-reduce buffer so that
communication maximally overlaps the remaining backward computation.

Behavioral parity with the reference ``_generate_groups_mgwfbp`` /
``_generate_groups_with_threshold`` (reference
distributed_optimizer.py:140-261), re-expressed as pure, unit-testable
functions (the reference buries them in the optimizer class).

Conventions (same as the reference):
- ``seq_layernames``, ``layerwise_times`` (tb) and ``sizes`` are in
  FORWARD order; index L-1 is the model's last layer = the FIRST gradient
  ready in backward.
- Returned ``groups`` is a list of lists of layer names in backward
  completion order; each group's all-reduce fires when its LAST listed
  member's gradient arrives.
- ``key_groupidx_maps`` maps layer name -> group index.

MI355X note: over intra-node xGMI, alpha is O(10 us) (RCCL launch + hop
latency) — two orders below the reference's 10GbE table (reference
distributed_optimizer.py:172-177) — so the ``t_wait < alpha`` merge
condition fires far less and the win shifts to amortizing per-call launch
overhead; alpha/beta are therefore MEASURED at startup (see
profiling.CommunicationProfiler) rather than hardcoded.
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

# Fitted alpha/beta fallback tables keyed by world size, used only when no
# measured values are available (e.g. CPU/gloo tests). The 'xgmi' entries
# are MI355X-plausible priors (RCCL launch ~20us; ring over 7x153GB/s
# links: beta ~= 2(P-1)/P / 153e9); they are replaced by the online fit on
# real hardware. The ethernet/IB tables mirror the reference's measured
# clusters (reference distributed_optimizer.py:166-177) for A/B parity runs.
ALPHA_BETA_TABLES = {
    'xgmi': {
        2: (2.0e-05, 6.6e-12),
        4: (2.0e-05, 9.9e-12),
        8: (2.0e-05, 1.15e-11),
        16: (2.5e-05, 1.25e-11),
    },
    '56GbIB': {
        2: (2.554691138304671e-06, 9.837548167872609e-11),
        4: (4.204298980348825e-05, 2.0589360830118177e-10),
        8: (9.75367204301171e-05, 3.0568230536676206e-10),
        16: (0.00023583677659915685, 4.0594787739537565e-10),
    },
    '10GbE': {
        2: (2.554691138304671e-06, 9.837548167872609e-11),
        4: (4.204298980348825e-05, 2.0589360830118177e-10),
        8: (0.0005230272768511732, 8.570746975492128e-10),
        16: (0.0009080981007148093, 7.395651186836712e-10),
    },
}


def lookup_alpha_beta(connection: str, nworkers: int) -> Tuple[float, float]:
    """Fallback alpha/beta for a connection type and world size."""
    table = ALPHA_BETA_TABLES.get(connection) or ALPHA_BETA_TABLES['xgmi']
    if nworkers in table:
        return table[nworkers]
    # nearest known world size (keeps solver usable at odd P)
    key = min(table.keys(), key=lambda k: abs(k - nworkers))
    return table[key]


def predict_allreduce_time(alpha: float, beta: float, size_bytes: float) -> float:
    """Cost model t = alpha + beta * bytes (reference utils.py:151-154)."""
    if size_bytes == 0:
        return 0.0
    return alpha + beta * size_bytes


def predict_from_table(table_sizes, table_times, size_bytes):
    """Piecewise-linear interpolation over a MEASURED (bytes -> seconds)
    all-reduce sweep — the reference's ``size_commtime_dict`` hook
    (reference distributed_optimizer.py:197-199, :210-212), which its
    entry points never populate; here the online sweep can feed it so
    non-linear RCCL protocol switches (LL/LL128/simple) are captured
    instead of forced through one alpha+beta line.

    table_sizes must be ascending. Extrapolates linearly from the last
    segment above the table, and scales proportionally below it.
    """
    if size_bytes <= 0:
        return 0.0
    n = len(table_sizes)
    if n == 0:
        raise ValueError('empty comm table')
    if n == 1 or size_bytes <= table_sizes[0]:
        return table_times[0] * size_bytes / table_sizes[0]             if size_bytes < table_sizes[0] else table_times[0]
    import bisect
    i = bisect.bisect_left(table_sizes, size_bytes)
    if i >= n:
        i = n - 1
    lo, hi = i - 1, i
    if table_sizes[hi] == table_sizes[lo]:
        return table_times[hi]
    f = (size_bytes - table_sizes[lo]) / (table_sizes[hi] - table_sizes[lo])
    return table_times[lo] + f * (table_times[hi] - table_times[lo])


def predict_sparse_allgather_time(alpha: float, beta: float, numel: int,
                                  density: float, nworkers: int,
                                  nbytes: int = 4,
                                  index_bytes: int = 8) -> float:
    """Cost of one merge group's top-k sparse exchange.

    The sparse path (distributed_optimizer._sparse_allgather_async)
    replaces the dense all-reduce with TWO all-gathers — values
    (``nbytes``/elem) and indices (int64, ``index_bytes``/elem) — each
    rank contributing k = ceil(numel * density) elements. A ring
    all-gather receives (P-1)*k elements per rank, so:

        t = 2*alpha + beta * (nbytes + index_bytes) * k * (P-1)

    This is the reference's sparse cost-model hook made real
    (reference utils.py:104-149 ``predict_density_time`` effectively
    returns a constant 0.001 — SURVEY §2.1); the top-k compute on the
    GPU is not modeled (it overlaps backward like the collectives do).
    """
    if numel == 0:
        return 0.0
    k = max(1, int(numel * density))
    payload = (nbytes + index_bytes) * k * max(nworkers - 1, 1)
    return 2.0 * alpha + beta * payload


def _comm_start_times(tc: List[float], tb: Sequence[float],
                      taob: Sequence[float], L: int) -> List[float]:
    """Earliest all-reduce start per layer given channel serialization.

    taoc[l] = max(taoc[l+1] + tc[l+1], taob[l] + tb[l]) walking l from L-1
    down (reference __calculate_comm_start, distributed_optimizer.py:187-192):
    layer l's comm starts when its gradient is ready AND the previously
    fired comm (layer l+1, earlier in backward) has drained.
    """
    taoc = [0.0] * L
    taoc[L - 1] = taob[L - 1] + tb[L - 1]
    for l in range(L - 2, -1, -1):
        taoc[l] = max(taoc[l + 1] + tc[l + 1], taob[l] + tb[l])
    return taoc


def generate_groups_mgwfbp(
    seq_layernames: Sequence[str],
    layerwise_times: Sequence[float],
    sizes: Sequence[int],
    alpha: float,
    beta: float,
    nbytes: int = 4,
    size_commtime: 'Tuple[Sequence[float], Sequence[float]]' = None,
    alpha_host: float = 0.0,
    density: float = 1.0,
    nworkers: int = 2,
) -> Tuple[List[List[str]], Dict[str, int], Dict[str, float]]:
    """Solve the optimal merged-gradient grouping.

    Args (forward order): layer names, per-layer backward times (s),
    per-layer element counts; cost-model alpha (s), beta (s/B); bytes per
    element of the comm dtype.

    ``alpha_host`` is the per-collective HOST cost (async-enqueue launch
    path + hook bookkeeping) that is paid once per group regardless of
    device-side overlap. On xGMI the device alpha is O(10us) — two
    orders below the reference's 10GbE table — so amortizing this host
    launch cost is what merging actually buys; it is added to every
    collective's cost AND to the startup saving in the merge condition
    (the reference's ``t_wait < alpha`` test,
    reference distributed_optimizer.py:239-241).

    Returns (groups, key_groupidx_maps, stats). stats carries the solver's
    predicted timeline (reference logs these at
    distributed_optimizer.py:256-259).
    """
    L = len(seq_layernames)
    if L == 0:
        return [], {}, {}
    if len(set(seq_layernames)) != L:
        raise ValueError('duplicate layer names passed to solver')
    if not (len(layerwise_times) == len(sizes) == L):
        raise ValueError('seq_layernames/layerwise_times/sizes length mismatch')

    tb = list(layerwise_times)
    p = [int(s) for s in sizes]          # merged element counts (mutated)

    if density < 1.0:
        # sparse top-k exchange: cost follows the all-gather payload,
        # not the dense all-reduce (and each group pays TWO collective
        # launches — values + indices)
        def comm_cost(size_bytes):
            if size_bytes == 0:
                return 0.0
            return 2.0 * alpha_host + predict_sparse_allgather_time(
                alpha, beta, size_bytes // max(nbytes, 1), density,
                nworkers, nbytes)
    elif size_commtime is not None:
        t_sizes, t_times = size_commtime

        def comm_cost(size_bytes):
            if size_bytes == 0:
                return 0.0
            return alpha_host + predict_from_table(t_sizes, t_times,
                                                   size_bytes)
    else:
        def comm_cost(size_bytes):
            if size_bytes == 0:
                return 0.0
            return alpha_host + predict_allreduce_time(alpha, beta,
                                                       size_bytes)

    tc = [comm_cost(s * nbytes) for s in p]
    # Gradient-ready offsets: taob[L-1] = 0 (backward starts at the last
    # layer); taob[l] = taob[l+1] + tb[l+1].
    taob = [0.0] * L
    for l in range(L - 2, -1, -1):
        taob[l] = taob[l + 1] + tb[l + 1]
    taoc = _comm_start_times(tc, tb, taob, L)
    tc_sum_before = sum(tc)

    def merge_into_prev(l: int) -> None:
        # Defer layer l's payload to ride with layer l-1 (the next gradient
        # in backward order).
        p[l - 1] += p[l]
        p[l] = 0
        tc[l] = 0.0
        tc[l - 1] = comm_cost(p[l - 1] * nbytes)

    groups: List[List[str]] = []
    group: List[str] = []
    key_groupidx_maps: Dict[str, int] = {}
    idx = 0
    key_groupidx_maps[seq_layernames[L - 1]] = idx
    for l in range(L - 1, 0, -1):
        key = seq_layernames[l]
        group.append(key)
        key_groupidx_maps[key] = idx
        ready_next = taob[l - 1] + tb[l - 1]   # when the next gradient lands
        merged = False
        if ready_next < taoc[l] + tc[l]:       # comm of l still busy then
            if taoc[l] > ready_next:
                # comm hasn't even started: free to merge, saves an alpha
                merge_into_prev(l)
                taoc = _comm_start_times(tc, tb, taob, L)
                merged = True
            else:
                t_wait = ready_next - taoc[l]
                # waiting < startup saved (device latency + host launch;
                # the sparse path saves TWO launches per merged-away
                # group: values + indices all-gathers)
                saved = ((alpha + alpha_host) * 2.0 if density < 1.0
                         else alpha + alpha_host)
                if t_wait < saved:
                    merge_into_prev(l)
                    taoc = _comm_start_times(tc, tb, taob, L)
                    merged = True
        if not merged:
            idx += 1
            groups.append(group)
            group = []
    key = seq_layernames[0]
    key_groupidx_maps[key] = idx
    group.append(key)
    groups.append(group)

    stats = {
        'predicted_nonoverlapped_time': taoc[0] + tc[0] - (taob[0] + tb[0]),
        'predicted_total_time': taoc[0] + tc[0],
        'tc_sum_before_merge': tc_sum_before,
        'tc_sum_after_merge': sum(tc),
        'num_groups': len(groups),
    }
    return groups, key_groupidx_maps, stats


def generate_groups_with_threshold(
    seq_layernames: Sequence[str],
    sizes: Sequence[int],
    threshold: int,
) -> Tuple[List[List[str]], Dict[str, int]]:
    """Greedy grouping by cumulative element count (reference
    distributed_optimizer.py:140-162).

    Walks layers in backward order accumulating numel; a group closes as
    soon as its cumulative count reaches ``threshold``. threshold=0 gives
    per-layer groups (pure WFBP); a huge threshold gives one group
    (single-shot all-reduce) — the reference A/B endpoints
    (batch_dist_mpi.sh:2).
    """
    L = len(seq_layernames)
    groups: List[List[str]] = []
    group: List[str] = []
    key_groupidx_maps: Dict[str, int] = {}
    idx = 0
    sub_size = 0
    for l in range(L - 1, -1, -1):          # backward order
        key = seq_layernames[l]
        numel = int(sizes[l])
        sub_size += numel
        key_groupidx_maps[key] = idx
        group.append(key)
        if sub_size >= threshold:
            idx += 1
            groups.append(group)
            group = []
            sub_size = 0
    if group:
        groups.append(group)
    return groups, key_groupidx_maps


def fit_alpha_beta(sizes_bytes: Sequence[float],
                   times_s: Sequence[float]) -> Tuple[float, float]:
    """Least-squares fit of t = alpha + beta*size from a comm sweep.

    Replaces the reference's sklearn LinearRegression
    (distributed_optimizer.py:112-116) with a closed-form fit.
    Clamps alpha to >= 0 (a negative intercept from noise would make the
    solver merge everything).
    """
    n = len(sizes_bytes)
    if n == 0:
        raise ValueError('empty sweep')
    if n == 1:
        return 0.0, times_s[0] / max(sizes_bytes[0], 1.0)
    sx = sum(sizes_bytes)
    sy = sum(times_s)
    sxx = sum(x * x for x in sizes_bytes)
    sxy = sum(x * y for x, y in zip(sizes_bytes, times_s))
    denom = n * sxx - sx * sx
    if denom == 0:
        return 0.0, sy / sx if sx else 0.0
    beta = (n * sxy - sx * sy) / denom
    alpha = (sy - beta * sx) / n
    if alpha < 0:
        # refit beta through origin-ish: keep slope, floor intercept
        alpha = 0.0
    if beta < 0:
        beta = 0.0
    return alpha, beta
