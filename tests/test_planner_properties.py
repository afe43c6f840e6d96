"""Property-based invariants of the planner core (hypothesis)."""

import math

from hypothesis import given, settings, strategies as st

from metis_amd.planner.groups import (
    multiset_permutations,
    power_of_two_shapes,
    stage_device_groups,
)
from metis_amd.planner.uniform import uniform_plans
from metis_amd.planner.volume import uniform_layer_split
from metis_amd.planner.balancer import pow2_slices


@settings(max_examples=60, deadline=None)
@given(st.integers(1, 5), st.sampled_from([4, 8, 16, 32]),
       st.sampled_from([0.5, 1.0]), st.sampled_from([2, 4, 6]))
def test_device_groups_invariants(num_stages, num_gpus, variance, mpl):
    shapes = power_of_two_shapes(num_gpus)
    for g in stage_device_groups(num_stages, num_gpus, shapes, variance, mpl):
        assert len(g) == num_stages
        assert sum(g) == num_gpus
        assert all(x in shapes for x in g)


@settings(max_examples=40, deadline=None)
@given(st.lists(st.integers(1, 4), min_size=1, max_size=6))
def test_multiset_permutation_count(items):
    perms = list(multiset_permutations(list(items)))
    n = math.factorial(len(items))
    for v in set(items):
        n //= math.factorial(items.count(v))
    assert len(perms) == n
    assert len(set(map(tuple, perms))) == len(perms)
    assert all(sorted(p) == sorted(items) for p in perms)


@settings(max_examples=40, deadline=None)
@given(st.sampled_from([2, 4, 8, 16]), st.sampled_from([2, 4, 8]),
       st.sampled_from([4, 8, 16, 32]))
def test_uniform_plans_valid(n_dev, max_tp, max_gbs):
    for p in uniform_plans(n_dev, max_tp, max_gbs):
        assert p.dp * p.pp * p.tp == n_dev
        assert p.gbs % p.mbs == 0
        assert p.tp <= max_tp
        # reference parity: gbs = dp is always visited first even when
        # dp > max_gbs (uniform.py:29-31)
        assert p.gbs <= max(max_gbs, p.dp)


@settings(max_examples=60, deadline=None)
@given(st.integers(1, 200))
def test_pow2_slices_sum(n):
    sl = pow2_slices(n)
    assert sum(sl) == n
    assert all(s & (s - 1) == 0 for s in sl)


@settings(max_examples=60, deadline=None)
@given(st.integers(3, 64), st.integers(1, 12))
def test_uniform_layer_split_covers(total, stages):
    if stages > total:
        return
    counts = uniform_layer_split(total, stages)
    assert len(counts) == stages
    assert sum(counts) == total
    assert all(c >= 0 for c in counts)
