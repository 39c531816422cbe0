"""Property-based solver tests (hypothesis): invariants that must hold
for ANY profile the solver is fed."""
from hypothesis import given, settings as hsettings, strategies as st

from mgwfbp_amd import solver


profiles = st.integers(min_value=1, max_value=40).flatmap(
    lambda n: st.tuples(
        st.lists(st.floats(min_value=1e-7, max_value=1e-2,
                           allow_nan=False), min_size=n, max_size=n),
        st.lists(st.integers(min_value=1, max_value=10**7),
                 min_size=n, max_size=n),
        st.floats(min_value=0.0, max_value=1e-2, allow_nan=False),
        st.floats(min_value=1e-13, max_value=1e-8, allow_nan=False),
    ))


@given(profiles)
@hsettings(max_examples=200, deadline=None)
def test_mgwfbp_partitions_and_never_worse_than_wfbp(args):
    tb, sizes, alpha, beta = args
    names = ['l%04d' % i for i in range(len(tb))]
    groups, gmap, stats = solver.generate_groups_mgwfbp(
        names, tb, sizes, alpha, beta, 4)
    flat = [k for g in groups for k in g]
    # exact partition, in backward order
    assert flat == list(reversed(names))
    # map consistent
    for gi, g in enumerate(groups):
        for k in g:
            assert gmap[k] == gi
    # groups are contiguous runs in backward order (merging only joins
    # neighbours)
    assert 1 <= len(groups) <= len(names)
    # predicted schedule never worse than plain WFBP on the same inputs
    L = len(names)
    tc = [solver.predict_allreduce_time(alpha, beta, s * 4)
          for s in sizes]
    taob = [0.0] * L
    for l in range(L - 2, -1, -1):
        taob[l] = taob[l + 1] + tb[l + 1]
    taoc = solver._comm_start_times(tc, tb, taob, L)
    wfbp_total = taoc[0] + tc[0]
    assert stats['predicted_total_time'] <= wfbp_total + 1e-12


@given(profiles)
@hsettings(max_examples=100, deadline=None)
def test_threshold_partitions(args):
    tb, sizes, alpha, beta = args
    names = ['l%04d' % i for i in range(len(tb))]
    for threshold in (0, 1, sum(sizes) // 2 + 1, 1 << 40):
        groups, gmap = solver.generate_groups_with_threshold(
            names, sizes, threshold)
        flat = [k for g in groups for k in g]
        assert flat == list(reversed(names))


@given(st.lists(st.tuples(
    st.floats(min_value=1e3, max_value=1e9, allow_nan=False),
    st.floats(min_value=1e-7, max_value=1e-1, allow_nan=False)),
    min_size=2, max_size=50))
@hsettings(max_examples=100, deadline=None)
def test_fit_alpha_beta_nonnegative(points):
    sizes = [p[0] for p in points]
    times = [p[1] for p in points]
    a, b = solver.fit_alpha_beta(sizes, times)
    assert a >= 0.0 and b >= 0.0
