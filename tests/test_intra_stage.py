"""Intra-stage (dp, tp) escalation semantics."""

from metis_amd.planner.intra_stage import _next_strategy, _strategies_valid


def test_escalation_halves_least_headroom_stage():
    # stage 1 has the least memory headroom -> it escalates first
    out = _next_strategy([(4, 1), (4, 1)], memory_state=[5.0, 1.0])
    assert out == [(4, 1), (2, 2)]
    # then stage 0
    out = _next_strategy(out, memory_state=[1.0, 5.0])
    assert out == [(2, 2), (2, 2)]


def test_escalation_default_state_prefers_large_dp():
    # no memory state yet: default 1/dp ranks the largest dp first
    out = _next_strategy([(8, 1), (2, 1)], memory_state=None)
    assert out == [(4, 2), (2, 1)]


def test_escalation_exhausts_at_dp1():
    assert _next_strategy([(1, 4), (1, 4)], memory_state=[0.0, 0.0]) is None


def test_strategy_validity():
    # mbs = gbs // dp // batches must be in [1, max_bs]; tp <= max_tp
    assert _strategies_valid([(4, 1)], gbs=16, batches=4, max_tp=4, max_bs=4)
    assert not _strategies_valid([(4, 1)], gbs=16, batches=8, max_tp=4, max_bs=4)  # mbs 0
    assert not _strategies_valid([(1, 8)], gbs=16, batches=1, max_tp=4, max_bs=16)  # tp
    assert not _strategies_valid([(1, 1)], gbs=64, batches=1, max_tp=4, max_bs=16)  # mbs > max
