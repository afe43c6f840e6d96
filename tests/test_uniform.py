from metis_amd.planner.uniform import uniform_plans


def test_uniform_plan_invariants():
    for n, max_tp, max_gbs in ((8, 4, 16), (4, 8, 32), (16, 2, 64), (1, 1, 4)):
        plans = list(uniform_plans(n, max_tp, max_gbs))
        assert plans, "enumeration must be non-empty"
        seen = set()
        for p in plans:
            assert p.dp * p.pp * p.tp == n
            assert p.tp <= max_tp
            assert p.gbs % p.mbs == 0
            assert p.mbs * p.dp <= p.gbs
            key = (p.dp, p.pp, p.tp, p.mbs, p.gbs)
            assert key not in seen, f"duplicate plan {key}"
            seen.add(key)


def test_uniform_plan_gbs_filter_complete():
    # at gbs == max_gbs every valid (strategy, mbs) combo must appear
    n, max_tp, G = 8, 4, 16
    got = {
        (p.dp, p.pp, p.tp, p.mbs)
        for p in uniform_plans(n, max_tp, G)
        if p.gbs == G
    }
    expected = set()
    for pp in range(1, n + 1):
        for tp in range(1, max_tp + 1):
            if n % (tp * pp):
                continue
            dp = n // tp // pp
            for mbs in range(1, G + 1):
                if G % mbs == 0 and mbs * dp <= G:
                    expected.add((dp, pp, tp, mbs))
    assert got == expected
