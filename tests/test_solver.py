"""Unit tests for the MG-WFBP merge-group solver (pure math, no GPU).

Mirrors the behavior of the reference solver
(/root/reference/distributed_optimizer.py:140-261) on synthetic inputs.
"""
import math

import pytest

from mgwfbp_amd import solver


def _names(n):
    return ['layer%03d' % i for i in range(n)]


class TestThresholdGrouping:
    def test_threshold_zero_is_wfbp(self):
        names = _names(5)
        sizes = [10, 20, 30, 40, 50]
        groups, gmap = solver.generate_groups_with_threshold(names, sizes, 0)
        # every layer its own group, backward order
        assert groups == [[n] for n in reversed(names)]
        assert gmap == {n: i for i, n in enumerate(reversed(names))}

    def test_huge_threshold_single_group(self):
        names = _names(6)
        sizes = [100] * 6
        groups, gmap = solver.generate_groups_with_threshold(names, sizes, 10**9)
        assert len(groups) == 1
        assert groups[0] == list(reversed(names))
        assert all(gmap[n] == 0 for n in names)

    def test_threshold_partitions_cover_all(self):
        names = _names(7)
        sizes = [3, 5, 7, 11, 13, 17, 19]
        groups, gmap = solver.generate_groups_with_threshold(names, sizes, 20)
        flat = [k for g in groups for k in g]
        assert sorted(flat) == sorted(names)
        assert flat == list(reversed(names))  # backward order preserved
        # groups close as soon as cumulative numel >= threshold
        for g in groups[:-1]:
            tot = sum(sizes[names.index(k)] for k in g)
            assert tot >= 20
        # map consistent with groups
        for gi, g in enumerate(groups):
            for k in g:
                assert gmap[k] == gi


class TestMgwfbpSolver:
    def test_large_alpha_merges_everything(self):
        # alpha >> everything: comm never keeps up, always merges
        names = _names(10)
        tb = [1e-4] * 10
        sizes = [1000] * 10
        groups, gmap, stats = solver.generate_groups_mgwfbp(
            names, tb, sizes, alpha=1.0, beta=1e-12)
        assert len(groups) == 1
        assert groups[0] == list(reversed(names))
        assert stats['num_groups'] == 1

    def test_zero_comm_cost_keeps_wfbp(self):
        # alpha=0, beta tiny: comm always finishes before the next grad;
        # no merge condition fires -> per-layer groups (WFBP).
        names = _names(8)
        tb = [1e-3] * 8
        sizes = [10] * 8
        groups, _, stats = solver.generate_groups_mgwfbp(
            names, tb, sizes, alpha=0.0, beta=1e-15)
        assert len(groups) == 8
        assert all(len(g) == 1 for g in groups)
        # fully overlapped: non-overlapped time ~ cost of final layer's comm
        assert stats['predicted_nonoverlapped_time'] <= 1e-9

    def test_groups_cover_all_layers_exactly_once(self):
        names = _names(20)
        tb = [(i % 5 + 1) * 1e-5 for i in range(20)]
        sizes = [(i * 37) % 9000 + 100 for i in range(20)]
        groups, gmap, _ = solver.generate_groups_mgwfbp(
            names, tb, sizes, alpha=5e-5, beta=3e-10)
        flat = [k for g in groups for k in g]
        assert sorted(flat) == sorted(names)
        assert flat == list(reversed(names))
        for gi, g in enumerate(groups):
            for k in g:
                assert gmap[k] == gi

    def test_duplicate_names_rejected(self):
        with pytest.raises(ValueError):
            solver.generate_groups_mgwfbp(['a', 'a'], [1e-4, 1e-4], [1, 1],
                                          1e-5, 1e-10)

    def test_small_alpha_merges_less_than_big_alpha(self):
        # MI355X regime: tiny alpha (xGMI) should produce >= as many groups
        # as an ethernet-sized alpha on the same model.
        names = _names(30)
        tb = [2e-5] * 30
        sizes = [250000] * 30  # 1MB fp32 each
        g_small, _, _ = solver.generate_groups_mgwfbp(
            names, tb, sizes, alpha=2e-5, beta=1.2e-11)
        g_big, _, _ = solver.generate_groups_mgwfbp(
            names, tb, sizes, alpha=9e-4, beta=7.4e-10)
        assert len(g_small) >= len(g_big)

    def test_predicted_time_after_merge_not_worse(self):
        # the merged schedule's predicted total must never exceed the plain
        # WFBP prediction on the same inputs (solver only merges when it
        # helps or is free)
        names = _names(16)
        tb = [3e-5 + (i % 3) * 1e-5 for i in range(16)]
        sizes = [50000 + 10000 * (i % 7) for i in range(16)]
        alpha, beta = 5.2e-4, 8.6e-10
        _, _, stats = solver.generate_groups_mgwfbp(
            names, tb, sizes, alpha, beta)
        nbytes = 4
        # simulate WFBP (no merging) timeline
        L = len(names)
        tc = [solver.predict_allreduce_time(alpha, beta, s * nbytes)
              for s in sizes]
        taob = [0.0] * L
        for l in range(L - 2, -1, -1):
            taob[l] = taob[l + 1] + tb[l + 1]
        taoc = solver._comm_start_times(tc, tb, taob, L)
        wfbp_total = taoc[0] + tc[0]
        assert stats['predicted_total_time'] <= wfbp_total + 1e-12


class TestAlphaBetaFit:
    def test_exact_linear_recovered(self):
        alpha, beta = 3e-5, 2e-10
        sizes = [4096.0 * i for i in range(1, 64)]
        times = [alpha + beta * s for s in sizes]
        a, b = solver.fit_alpha_beta(sizes, times)
        assert math.isclose(a, alpha, rel_tol=1e-6)
        assert math.isclose(b, beta, rel_tol=1e-6)

    def test_noisy_fit_nonnegative(self):
        import random
        rng = random.Random(0)
        sizes = [8192.0 * i for i in range(1, 64)]
        times = [1e-6 + 1e-11 * s + rng.gauss(0, 2e-7) for s in sizes]
        a, b = solver.fit_alpha_beta(sizes, times)
        assert a >= 0.0
        assert b > 0.0

    def test_lookup_tables(self):
        a, b = solver.lookup_alpha_beta('xgmi', 8)
        assert 0 < a < 1e-3 and 0 < b < 1e-9
        a10, b10 = solver.lookup_alpha_beta('10GbE', 16)
        assert a10 > a  # ethernet launch latency far above xGMI
        # unknown world size falls back to nearest
        a2, _ = solver.lookup_alpha_beta('xgmi', 3)
        assert a2 > 0


class TestMeasuredCommTable:
    def test_interp_matches_linear_table(self):
        sizes = [1000.0 * i for i in range(1, 20)]
        times = [2e-5 + 1e-11 * s for s in sizes]
        for q in (1500.0, 7777.0, 19000.0, 50000.0):
            got = solver.predict_from_table(sizes, times, q)
            if q <= sizes[-1]:
                assert abs(got - (2e-5 + 1e-11 * q)) < 1e-12
        # below the table scales proportionally, zero -> zero
        assert solver.predict_from_table(sizes, times, 0) == 0.0
        assert solver.predict_from_table(sizes, times, 500.0) == \
            times[0] * 0.5

    def test_solver_with_table_equals_linear_when_table_linear(self):
        names = ['l%d' % i for i in range(12)]
        tb = [5e-5] * 12
        szs = [10000 + 1000 * i for i in range(12)]
        alpha, beta = 3e-5, 2e-10
        tsizes = [4.0 * 2 ** k for k in range(8, 30)]
        ttimes = [alpha + beta * s for s in tsizes]
        g1, _, s1 = solver.generate_groups_mgwfbp(names, tb, szs, alpha,
                                                  beta, 4)
        g2, _, s2 = solver.generate_groups_mgwfbp(
            names, tb, szs, alpha, beta, 4,
            size_commtime=(tsizes, ttimes))
        assert g1 == g2
        assert abs(s1['predicted_total_time']
                   - s2['predicted_total_time']) < 1e-9

    def test_nonlinear_table_respected(self):
        # a protocol-switch-like jump in the table must show in tc
        sizes = [1e3, 1e6, 1e6 + 1, 1e8]
        times = [1e-5, 2e-5, 2e-4, 1e-3]
        below = solver.predict_from_table(sizes, times, 5e5)
        above = solver.predict_from_table(sizes, times, 2e6)
        assert below < 2e-5
        assert above > 2e-4
