"""Device-side (torch ops) planner for the row-sharded engine:
make_grouped_plan_torch semantics vs the numpy planner, and the
GroupedPlanT execution path through pull/train/push with torch ids."""
import numpy as np
import pytest
import torch

from glint_word2vec_amd.models import sgns
from glint_word2vec_amd.vocab import build_unigram_table

pytest.importorskip("glint_word2vec_amd._cpu_native")


def _fixture(seed=0, vocab=50, n=500):
    rng = np.random.default_rng(seed)
    tokens = rng.integers(0, vocab, n).astype(np.int32)
    offsets = np.array([0, n // 4, n // 4, n // 2, n], dtype=np.int32)
    counts = np.bincount(tokens, minlength=vocab).astype(np.int64) + 1
    table = build_unigram_table(counts, 997)
    return tokens, offsets, counts, table


def test_planner_exact_match_when_deterministic():
    """window=1, n_neg=0, no subsample: the only random draw (shrunk
    window size b) is forced to 1, so numpy and torch planners must agree
    bit-for-bit — same enumeration order by construction."""
    tokens, offsets, _, table = _fixture()
    p_np = sgns.make_grouped_plan(tokens, offsets, None, table, 1, 0,
                                  np.random.default_rng(1))
    gen = torch.Generator().manual_seed(1)
    p_t = sgns.make_grouped_plan_torch(
        torch.from_numpy(tokens), torch.from_numpy(offsets), None,
        torch.from_numpy(table), 1, 0, gen)
    np.testing.assert_array_equal(p_np.group_center, p_t.group_center.numpy())
    np.testing.assert_array_equal(p_np.group_offsets,
                                  p_t.group_offsets.numpy())
    np.testing.assert_array_equal(p_np.pair_target, p_t.pair_target.numpy())
    np.testing.assert_array_equal(p_np.pair_label, p_t.pair_label.numpy())


@pytest.mark.parametrize("window_mode", ["canonical", "reference"])
def test_planner_structure(window_mode):
    tokens, offsets, _, table = _fixture(seed=3)
    gen = torch.Generator().manual_seed(2)
    p = sgns.make_grouped_plan_torch(
        torch.from_numpy(tokens), torch.from_numpy(offsets), None,
        torch.from_numpy(table), 3, 4, gen, window_mode)
    go = p.group_offsets.numpy()
    pl = p.pair_label.numpy()
    pt = p.pair_target.numpy()
    assert go[0] == 0 and go[-1] == p.num_pairs
    assert (np.diff(go) > 0).all()              # no empty groups
    assert (pl[go[:-1]] == 1.0).all()           # each group opens positive
    assert set(np.unique(pl)) <= {0.0, 1.0}
    if p.num_pairs:
        npos = int(pl.sum())
        # ~1/(1+4) positives modulo dropped negative collisions
        assert 0.15 < npos / p.num_pairs < 0.25
        assert set(pt[pl == 0].tolist()) <= set(table.tolist())
    assert p.group_center.dtype == torch.int32
    assert p.pair_target.dtype == torch.int32


def test_planner_subsample_reduces_pairs():
    tokens, offsets, _, table = _fixture(seed=4, n=2000)
    gen = torch.Generator().manual_seed(5)
    full = sgns.make_grouped_plan_torch(
        torch.from_numpy(tokens), torch.from_numpy(offsets), None,
        torch.from_numpy(table), 3, 2, gen)
    kp = torch.full((50,), 0.3)
    gen = torch.Generator().manual_seed(5)
    sub = sgns.make_grouped_plan_torch(
        torch.from_numpy(tokens), torch.from_numpy(offsets), kp,
        torch.from_numpy(table), 3, 2, gen)
    assert 0 < sub.num_pairs < full.num_pairs * 0.5


def test_planner_determinism():
    tokens, offsets, _, table = _fixture(seed=6)
    def plan(s):
        g = torch.Generator().manual_seed(s)
        return sgns.make_grouped_plan_torch(
            torch.from_numpy(tokens), torch.from_numpy(offsets), None,
            torch.from_numpy(table), 4, 3, g)
    a, b, c = plan(7), plan(7), plan(8)
    np.testing.assert_array_equal(a.pair_target.numpy(),
                                  b.pair_target.numpy())
    assert not np.array_equal(a.pair_target.numpy(), c.pair_target.numpy())


def test_row_engine_torch_plan_matches_numpy_plan():
    """Feeding the SAME plan through the GroupedPlanT path (torch unique +
    torch-routed pull/push) must produce exactly the host-plan result."""
    from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns
    tokens, offsets, counts, _ = _fixture(seed=9)
    p_np = None
    outs = []
    for variant in ("numpy", "torch"):
        eng = RowShardedSgns(50, 12, device="cpu", seed=3, counts=counts,
                             table_size=997)
        if p_np is None:
            p_np = eng.make_plan(tokens, offsets, 3, 4,
                                 np.random.default_rng(11))
        plan = (p_np if variant == "numpy" else sgns.GroupedPlanT(
            torch.from_numpy(p_np.group_center.astype(np.int32)),
            torch.from_numpy(p_np.group_offsets.astype(np.int64)),
            torch.from_numpy(p_np.pair_target.astype(np.int32)),
            torch.from_numpy(p_np.pair_label.astype(np.float32))))
        eng.train_step(tokens, offsets, 0.05, 3, 4,
                       np.random.default_rng(12), plan=plan)
        outs.append(eng.to_host())
    np.testing.assert_array_equal(outs[0][0], outs[1][0])
    np.testing.assert_array_equal(outs[0][1], outs[1][1])


def test_row_engine_device_plan_end_to_end():
    """make_plan_device on the engine (CPU device here) trains finite,
    learning updates through the full pull/train/push cycle."""
    from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns
    tokens, offsets, counts, _ = _fixture(seed=13)
    eng = RowShardedSgns(50, 12, device="cpu", seed=3, counts=counts,
                         table_size=997)
    before = eng.to_host()[0].copy()
    for s in range(3):
        plan = eng.make_plan_device(tokens, offsets, 3, 4, seed=100 + s)
        eng.train_step(tokens, offsets, 0.05, 3, 4,
                       np.random.default_rng(1), plan=plan)
    after, a1 = eng.to_host()
    assert np.isfinite(after).all() and np.isfinite(a1).all()
    assert not np.array_equal(before, after)
    st = eng.read_stats()
    assert st.pairs > 0 and st.positives > 0


@pytest.mark.parametrize("case", ["empty", "single_word", "one_sent_1tok",
                                  "window_bigger_than_sent"])
def test_planner_edge_cases(case):
    table = torch.from_numpy(
        build_unigram_table(np.ones(20, np.int64), 101))
    gen = torch.Generator().manual_seed(1)
    if case == "empty":
        tokens = torch.zeros(0, dtype=torch.int32)
        offsets = torch.zeros(1, dtype=torch.int32)
    elif case == "single_word":
        tokens = torch.tensor([7], dtype=torch.int32)
        offsets = torch.tensor([0, 1], dtype=torch.int32)
    elif case == "one_sent_1tok":
        tokens = torch.tensor([3, 7, 3], dtype=torch.int32)
        offsets = torch.tensor([0, 1, 2, 3], dtype=torch.int32)
    else:  # window 10 >> sentence length 3
        tokens = torch.tensor([1, 2, 3], dtype=torch.int32)
        offsets = torch.tensor([0, 3], dtype=torch.int32)
    p = sgns.make_grouped_plan_torch(tokens, offsets, None, table, 10, 3,
                                     gen)
    go = p.group_offsets.numpy()
    assert go[0] == 0 and go[-1] == p.num_pairs
    if p.num_pairs:
        pl = p.pair_label.numpy()
        assert (pl[go[:-1]] == 1.0).all()
        # targets only come from within the same sentence
        pt = p.pair_target.numpy()
        assert set(pt[pl == 1].tolist()) <= set(tokens.numpy().tolist())
    if case in ("empty", "single_word"):
        assert p.num_pairs == 0


def test_planner_full_subsample_drop():
    """keep_prob 0 drops everything: empty plan, no crash."""
    table = torch.from_numpy(
        build_unigram_table(np.ones(20, np.int64), 101))
    gen = torch.Generator().manual_seed(2)
    tokens = torch.from_numpy(
        np.random.default_rng(0).integers(0, 20, 100).astype(np.int32))
    offsets = torch.tensor([0, 100], dtype=torch.int32)
    kp = torch.zeros(20)
    p = sgns.make_grouped_plan_torch(tokens, offsets, kp, table, 3, 2, gen)
    assert p.num_pairs == 0 and p.num_groups == 0
