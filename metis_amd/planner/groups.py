"""Pipeline device-group composition and bounded permutation.

Behavior parity with reference search_space/device_group.py (the two "key
ideas" of the Metis paper) plus an original multiset-permutation generator
(the reference vendors ekg/multipermute; we use a counter-based recursion
producing the same distinct-permutation set).

* ``power_of_two_shapes(n)``          — candidate stage sizes 1,2,4,... <= n
* ``compositions(num_stages, n, shapes)`` — non-decreasing compositions of n
  into num_stages parts drawn from shapes (device_group.py:58-81)
* variance pruning                    — drop shapes below
  max(n // stages, stages // n) * variance ("Key idea 1", :93-98)
* ``merge_small_groups``              — merge pairs of equal smallest groups
  until at most ``max_permute_len`` permutation units remain
  ("Key idea 2", :7-55)
"""

from __future__ import annotations

from collections import Counter
from typing import Iterator, List, Sequence, Tuple


def power_of_two_shapes(num_gpus: int) -> List[int]:
    shapes, p = [], 1
    while p <= num_gpus:
        shapes.append(p)
        p *= 2
    return shapes


def compositions(num_stages: int, num_gpus: int, shapes: Sequence[int]) -> Iterator[List[int]]:
    """Non-decreasing compositions of num_gpus into num_stages parts from shapes."""
    shapes = sorted(shapes)
    if not shapes:
        return

    smallest, largest = shapes[0], shapes[-1]

    def rec(remaining: int, stages_left: int, min_idx: int, acc: List[int]) -> Iterator[List[int]]:
        # prune: even all-largest / all-smallest can't land on the target
        if largest * stages_left < remaining or smallest * stages_left > remaining:
            return
        if stages_left == 0:
            if remaining == 0:
                yield list(acc)
            return
        for i in range(min_idx, len(shapes)):
            s = shapes[i]
            if s > remaining:
                break
            acc.append(s)
            yield from rec(remaining - s, stages_left - 1, i, acc)
            acc.pop()

    yield from rec(num_gpus, num_stages, 0, [])


def multiset_permutations(items: Sequence) -> Iterator[Tuple]:
    """Each distinct permutation of ``items`` exactly once."""
    counts = Counter(items)
    keys = sorted(counts)
    n = len(items)
    acc: List = []

    def rec() -> Iterator[Tuple]:
        if len(acc) == n:
            yield tuple(acc)
            return
        for k in keys:
            if counts[k]:
                counts[k] -= 1
                acc.append(k)
                yield from rec()
                acc.pop()
                counts[k] += 1

    yield from rec()


def merge_small_groups(parts: Sequence[int], max_permute_len: int) -> List[Tuple[int, ...]]:
    """Merge adjacent pairs of equal smallest groups until the number of
    permutation units is <= max_permute_len (or no further merge applies).

    ``parts`` must be non-decreasing (as produced by :func:`compositions`).
    Returns a list of tuples; each tuple's members stay adjacent in every
    permutation. Behavior parity: device_group.py:7-55 including its
    "round num_reduce up to half the leading equal-group run" heuristic.
    """
    groups: List[Tuple[int, ...]] = [(p,) for p in parts]
    num_reduce = len(groups) - max_permute_len
    while num_reduce > 0:
        min_sum = sum(groups[0])
        # length of the leading run of groups equal to the first, +1 if a
        # different group follows (reference find_num_min quirk)
        leading = 0
        for g in groups:
            if g == groups[0]:
                leading += 1
            else:
                leading += 1
                break
        if leading // 2 > num_reduce:
            num_reduce = leading // 2

        merged: List[Tuple[int, ...]] = []
        i = 0
        while i < len(groups):
            if num_reduce <= i // 2:
                merged.extend(groups[i:])
                break
            if i + 1 >= len(groups):
                merged.append(groups[i])
                i += 2
                continue
            a, b = groups[i], groups[i + 1]
            if sum(a) == min_sum and sum(b) == min_sum:
                merged.append(a + b)
            else:
                merged.append(a)
                merged.append(b)
            i += 2

        groups = merged
        if num_reduce == len(groups) - max_permute_len:
            break  # no progress possible
        num_reduce = len(groups) - max_permute_len
    return groups


def stage_device_groups(
    num_stages: int,
    num_gpus: int,
    shapes: Sequence[int],
    variance: float,
    max_permute_len: int,
) -> List[List[int]]:
    """All ordered per-stage GPU-count assignments for a stage count.

    Composition -> variance pruning -> bounded multiset permutation ->
    flatten merged groups. Parity: device_group.py:93-107.
    """
    min_group = max(num_gpus // num_stages, num_stages // num_gpus) * variance
    pruned = [s for s in shapes if s >= min_group]

    out: List[List[int]] = []
    for comp in compositions(num_stages, num_gpus, pruned):
        merged = merge_small_groups(comp, max_permute_len)
        for perm in multiset_permutations(merged):
            flat: List[int] = []
            for grp in perm:
                flat.extend(grp)
            out.append(flat)
    return out
