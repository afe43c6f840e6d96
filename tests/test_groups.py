import math
from collections import Counter

from metis_amd.planner.groups import (
    compositions,
    merge_small_groups,
    multiset_permutations,
    power_of_two_shapes,
    stage_device_groups,
)


def test_power_of_two_shapes():
    assert power_of_two_shapes(16) == [1, 2, 4, 8, 16]
    assert power_of_two_shapes(6) == [1, 2, 4]
    assert power_of_two_shapes(1) == [1]


def test_compositions_properties():
    shapes = power_of_two_shapes(16)
    comps = list(compositions(3, 16, shapes))
    for c in comps:
        assert sum(c) == 16
        assert c == sorted(c)
        assert all(x in shapes for x in c)
    # distinct
    assert len({tuple(c) for c in comps}) == len(comps)
    # known: 16 = 1+1+... no; 3 parts of powers of two summing 16:
    # [4,4,8], [2,2,... no 2+2+12 invalid], [1,... 1+1+14 no], [2,... 2+6+8 no]
    # valid: 4+4+8, 2+... none, 1+... none, 8+8+... no (3 parts): also 16? no.
    assert {tuple(c) for c in comps} == {(4, 4, 8)}


def test_multiset_permutations_count():
    items = [1, 1, 2, 2, 2, 4]
    perms = list(multiset_permutations(items))
    expected = math.factorial(6) // (math.factorial(2) * math.factorial(3))
    assert len(perms) == expected
    assert len(set(perms)) == expected
    assert all(Counter(p) == Counter(items) for p in perms)


def test_merge_small_groups_bounds():
    merged = merge_small_groups([1, 1, 1, 1, 1, 1, 2], max_permute_len=6)
    # pairs of the smallest (1) merge: (1,1),(1,1),(1,1),(2)
    assert merged == [(1, 1), (1, 1), (1, 1), (2,)]
    # no merge needed when already short
    assert merge_small_groups([2, 2, 4], 4) == [(2,), (2,), (4,)]


def test_stage_device_groups_cover_cluster():
    shapes = power_of_two_shapes(16)
    for stages in (1, 2, 3, 4):
        for groups in stage_device_groups(stages, 16, shapes, 1.0, 4):
            assert sum(groups) == 16
            assert len(groups) == stages


def test_variance_prunes_small_groups():
    shapes = power_of_two_shapes(16)
    # variance=1: min group = 16//4 = 4 -> only [4,4,4,4] for 4 stages
    got = stage_device_groups(4, 16, shapes, 1.0, 4)
    assert got == [[4, 4, 4, 4]]
    # variance=0.5 admits group size 2 as well
    got_loose = stage_device_groups(4, 16, shapes, 0.5, 4)
    assert [4, 4, 4, 4] in got_loose and len(got_loose) > 1
    assert any(2 in g for g in got_loose)
