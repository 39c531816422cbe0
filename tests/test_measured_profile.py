"""Solver regression on the MEASURED resnet50 MI355X layer profile
(profiles/profile_resnet50.json, dumped on hardware)."""
import json
import os

import pytest

from mgwfbp_amd import solver

PROF = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), 'profiles', 'profile_resnet50.json')


@pytest.fixture(scope='module')
def prof():
    if not os.path.exists(PROF):
        pytest.skip('measured profile not committed')
    with open(PROF) as f:
        return json.load(f)


def test_profile_shape(prof):
    assert len(prof['seq_layernames']) == 161
    assert sum(prof['sizes']) == 25557032
    assert 0.005 < sum(prof['layerwise_times']) < 0.2


def test_solver_merges_on_xgmi_priors(prof):
    alpha, beta = solver.lookup_alpha_beta('xgmi', 8)
    groups, gmap, stats = solver.generate_groups_mgwfbp(
        prof['seq_layernames'], prof['layerwise_times'], prof['sizes'],
        alpha, beta, 4)
    # merges SOME tiny layers (fewer groups than layers) but keeps the
    # schedule fine-grained in the xGMI regime
    assert 40 < stats['num_groups'] < 161
    assert sorted(k for g in groups for k in g) == \
        sorted(prof['seq_layernames'])
    # predicted non-overlapped time is ~ the tail group's comm (tens of
    # microseconds), far below the 10GbE regime
    assert stats['predicted_nonoverlapped_time'] < 1e-3


def test_ethernet_regime_merges_harder(prof):
    a8, b8 = solver.lookup_alpha_beta('10GbE', 8)
    _, _, s_eth = solver.generate_groups_mgwfbp(
        prof['seq_layernames'], prof['layerwise_times'], prof['sizes'],
        a8, b8, 4)
    ax, bx = solver.lookup_alpha_beta('xgmi', 8)
    _, _, s_x = solver.generate_groups_mgwfbp(
        prof['seq_layernames'], prof['layerwise_times'], prof['sizes'],
        ax, bx, 4)
    # the reference's ethernet alpha forces much coarser groups
    assert s_eth['num_groups'] < s_x['num_groups']
