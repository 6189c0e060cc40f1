import numpy as np

from r2d2_amd.replay.priority_tree import PriorityTree


def test_root_equals_leaf_sum_after_random_interleaving():
    rng = np.random.default_rng(0)
    tree = PriorityTree(1000, prio_exponent=0.9, is_exponent=0.6,
                        rng=np.random.default_rng(1))
    for _ in range(50):
        n = rng.integers(1, 64)
        idx = rng.integers(0, 1000, size=n)
        td = rng.random(n).astype(np.float64) * 10
        tree.update(idx, td)
        if tree.total > 0:
            tree.sample(16)  # sampling must not mutate the tree
        assert abs(tree.total - tree.levels[-1].sum()) < 1e-9 * max(1, tree.total)
    # every internal node consistent
    for lvl in range(len(tree.levels) - 1):
        child = tree.levels[lvl + 1]
        np.testing.assert_allclose(tree.levels[lvl],
                                   child[0::2] + child[1::2], rtol=1e-12)


def test_sampled_frequency_proportional_to_priority():
    tree = PriorityTree(8, prio_exponent=1.0, is_exponent=0.6,
                        rng=np.random.default_rng(2))
    prios = np.array([1, 1, 2, 4, 8, 0, 0, 16], dtype=np.float64)
    tree.update(np.arange(8), prios)
    counts = np.zeros(8)
    n_draws = 2000
    for _ in range(n_draws):
        idx, _ = tree.sample(32)
        np.add.at(counts, idx, 1)
    freq = counts / counts.sum()
    expect = prios / prios.sum()
    assert counts[5] == 0 and counts[6] == 0
    np.testing.assert_allclose(freq, expect, atol=0.01)


def test_stratified_coverage():
    """Stratified sampling: with uniform priorities every draw covers distinct
    equal intervals -> all leaves hit with batch == capacity."""
    tree = PriorityTree(64, 1.0, 0.6, rng=np.random.default_rng(3))
    tree.update(np.arange(64), np.ones(64))
    idx, w = tree.sample(64)
    assert sorted(idx.tolist()) == list(range(64))
    np.testing.assert_allclose(w, np.ones(64))


def test_is_weights_formula():
    tree = PriorityTree(4, prio_exponent=0.9, is_exponent=0.6,
                        rng=np.random.default_rng(4))
    td = np.array([1.0, 2.0, 3.0, 4.0])
    tree.update(np.arange(4), td)
    idx, w = tree.sample(256)
    p = td[idx] ** 0.9
    expect = (p / p.min()) ** -0.6
    np.testing.assert_allclose(w, expect, rtol=1e-10)


def test_update_overwrite():
    tree = PriorityTree(16, 1.0, 0.6, rng=np.random.default_rng(5))
    tree.update(np.arange(16), np.ones(16))
    tree.update(np.array([3]), np.array([100.0]))
    assert abs(tree.total - (15 + 100)) < 1e-9
    idx, _ = tree.sample(64)
    assert (idx == 3).mean() > 0.5


def test_property_random_ops_keep_tree_consistent():
    """hypothesis: any interleaving of updates/samples keeps root == sum of
    leaves and samples in-range (SURVEY §4 sum-tree property tests)."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=30, deadline=None)
    @given(cap=st.integers(2, 200),
           ops=st.lists(st.tuples(st.integers(0, 1), st.integers(0, 10_000)),
                        min_size=1, max_size=20),
           seed=st.integers(0, 99999))
    def run(cap, ops, seed):
        rng = np.random.default_rng(seed)
        tree = PriorityTree(cap, prio_exponent=0.9, is_exponent=0.6,
                            rng=np.random.default_rng(seed + 1))
        touched = False
        for kind, s in ops:
            if kind == 0:
                k = int(rng.integers(1, min(cap, 16) + 1))
                idx = rng.choice(cap, size=k, replace=False)
                td = rng.random(k).astype(np.float64) + 1e-3
                tree.update(idx, td)
                touched = True
            elif touched:
                n = int(rng.integers(1, 17))
                idxes, w = tree.sample(n)
                assert ((idxes >= 0) & (idxes < cap)).all()
                assert np.isfinite(w).all() and (w > 0).all() and (w <= 1 + 1e-9).all()
            leaves = tree.levels[-1][:tree.num_leaves]
            np.testing.assert_allclose(tree.levels[0][0], leaves.sum(),
                                       rtol=1e-9)

    run()


def test_sample_never_returns_dead_slots():
    """Zero-priority leaves (partial-block slots, zero-padded capacity
    tail) must never be returned even under fp-edge descents — the
    consumer indexes the block ring with these (a dead slot crashed the
    full-scale host assembler before the snap-repair)."""
    rng = np.random.default_rng(0)
    tree = PriorityTree(50_000, 0.9, 0.6, rng=np.random.default_rng(1))
    # gappy occupancy like a ring of partial blocks: 10-slot groups with
    # random tails unwritten
    idx = []
    for g in range(0, 49_990, 10):
        n = int(rng.integers(1, 11))
        idx.extend(range(g, g + n))
    idx = np.array(idx, dtype=np.int64)
    tree.update(idx, rng.random(len(idx)).astype(np.float64) + 1e-3)
    occupied = np.zeros(tree.num_leaves, dtype=bool)
    occupied[idx] = True
    for _ in range(200):
        sampled, w = tree.sample(64)
        assert occupied[sampled].all()
        assert np.isfinite(w).all() and (w > 0).all()
