"""Solver analysis: predicted comm/compute timelines for MG-WFBP vs
WFBP vs single-group on measured MI355X layer profiles.

Pure CPU — uses the GPU-profiled per-layer backward times
(profiles/profile_<model>.json, produced by benchmarks/dump_profile.py)
and the xGMI alpha/beta priors (replaced by the online fit in real
multi-GPU runs). Reproduces the reference's internal A/B methodology
(reference batch_dist_mpi.sh:2) analytically.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from mgwfbp_amd import solver  # noqa: E402


def timeline(tb, tc_sizes, alpha, beta, groups, key_pos, nbytes=4):
    """Simulated non-overlapped comm time for a given grouping."""
    # groups listed in backward order; group i's comm starts when its
    # last member's gradient is ready and the channel is free
    ready = 0.0
    chan_free = 0.0
    finish = 0.0
    # per-layer ready times in backward order
    L = len(tb)
    bwd_ready = []
    acc = 0.0
    for l in range(L - 1, -1, -1):
        acc += tb[l]
        bwd_ready.append((l, acc))
    ready_map = dict(bwd_ready)
    total_bwd = acc
    for g in groups:
        gsize = sum(tc_sizes[key_pos[k]] for k in g)
        g_ready = max(ready_map[key_pos[k]] for k in g)
        start = max(g_ready, chan_free)
        chan_free = start + solver.predict_allreduce_time(
            alpha, beta, gsize * nbytes)
        finish = chan_free
    return finish - total_bwd, finish, total_bwd


def analyze(profile_path, nworkers=8):
    with open(profile_path) as f:
        prof = json.load(f)
    seq = prof['seq_layernames']
    tb = prof['layerwise_times']
    sizes = prof['sizes']
    key_pos = {k: i for i, k in enumerate(seq)}
    alpha, beta = solver.lookup_alpha_beta('xgmi', nworkers)
    rows = []
    mg_groups, _, stats = solver.generate_groups_mgwfbp(
        seq, tb, sizes, alpha, beta, 4)
    for name, groups in [
            ('mgwfbp', mg_groups),
            ('wfbp', solver.generate_groups_with_threshold(
                seq, sizes, 0)[0]),
            ('single', solver.generate_groups_with_threshold(
                seq, sizes, 1 << 40)[0])]:
        nono, fin, tbwd = timeline(tb, sizes, alpha, beta, groups,
                                   key_pos)
        rows.append((name, len(groups), nono, fin, tbwd))
    return prof, alpha, beta, rows


def main():
    out = ['# Predicted comm/compute overlap on xGMI (P=8 prior '
           'alpha/beta)',
           '',
           'Per-layer backward times measured on 1x MI355X '
           '(benchmarks/dump_profile.py); all-reduce model t = alpha + '
           'beta*bytes with xgmi priors (the live path fits alpha/beta '
           'online at startup on multi-GPU runs).',
           '']
    for model in ('resnet50', 'vgg16i', 'resnet20'):
        path = os.path.join('profiles', 'profile_%s.json' % model)
        if not os.path.exists(path):
            continue
        prof, alpha, beta, rows = analyze(path)
        out.append('## %s (bs %d, %d layers, backward %.2f ms)'
                   % (model, prof['batch_size'],
                      len(prof['seq_layernames']),
                      sum(prof['layerwise_times']) * 1e3))
        out.append('alpha=%.2e s, beta=%.2e s/B' % (alpha, beta))
        out.append('')
        out.append('| arm | groups | non-overlapped comm | iter time '
                   '(bwd+exposed comm) |')
        out.append('|---|---|---|---|')
        for name, ng, nono, fin, tbwd in rows:
            out.append('| %s | %d | %.3f ms | %.3f ms |'
                       % (name, ng, nono * 1e3, fin * 1e3))
        out.append('')
    report = '\n'.join(out)
    print(report)
    with open('profiles/solver_analysis.md', 'w') as f:
        f.write(report + '\n')


if __name__ == '__main__':
    main()
