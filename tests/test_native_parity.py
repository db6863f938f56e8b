"""C++ native trainer vs Python oracle: identical RNG draws (exact pair
counts) and matching table updates (fp tolerance for reassociation)."""
import numpy as np
import pytest

from glint_word2vec_amd.models import sgns
from glint_word2vec_amd.ops import cpu_ref
from glint_word2vec_amd.vocab import build_unigram_table, keep_probabilities

native = pytest.importorskip("glint_word2vec_amd._cpu_native")


def _problem(vocab=30, dim=12, n_tokens=80, seed=3):
    rng = np.random.default_rng(seed)
    tokens = rng.integers(0, vocab, n_tokens).astype(np.int32)
    offsets = np.array([0, 25, 60, n_tokens], dtype=np.int32)
    counts = np.bincount(tokens, minlength=vocab).astype(np.int64) + 1
    table = build_unigram_table(counts, 997)
    syn0, syn1 = sgns.init_tables(vocab, dim, seed)
    return tokens, offsets, counts, table, syn0, syn1


@pytest.mark.parametrize("window_mode", ["canonical", "reference"])
@pytest.mark.parametrize("subsample", [False, True])
def test_cpp_matches_python_oracle(window_mode, subsample):
    tokens, offsets, counts, table, syn0, syn1 = _problem()
    kp = keep_probabilities(counts, int(counts.sum()), 0.2) if subsample else None
    a0, a1 = syn0.copy(), syn1.copy()
    b0, b1 = syn0.copy(), syn1.copy()
    st_py = cpu_ref.train_batch_oracle(a0, a1, tokens, offsets, kp, table,
                                       0.03, 3, 4, seed=99, sent_id_base=7,
                                       window_mode=window_mode)
    st_c = native.train_batch(b0, b1, tokens, offsets, kp, table,
                              0.03, 3, 4, 99, 7, window_mode, 1)
    assert st_c["pairs"] == st_py.pairs
    assert st_c["positives"] == st_py.positives
    assert st_c["words_trained"] == st_py.words_trained
    assert st_c["sum_fplus"] == pytest.approx(st_py.sum_fplus, rel=1e-4, abs=1e-4)
    np.testing.assert_allclose(a0, b0, rtol=1e-5, atol=1e-7)
    np.testing.assert_allclose(a1, b1, rtol=1e-5, atol=1e-7)


def test_lut_sigmoid_parity_and_semantics():
    """LUT mode (reference getSigmoid): C++ == Python oracle, and LUT
    differs measurably from exact sigmoid (it is a real mode)."""
    from glint_word2vec_amd.models.sgns import create_exp_table
    tokens, offsets, counts, table, syn0, syn1 = _problem()
    et = create_exp_table()
    assert et.shape == (1000,)
    assert abs(float(et[500]) - 0.5) < 0.01
    a0, a1 = syn0.copy(), syn1.copy()
    st_py = cpu_ref.train_batch_oracle(a0, a1, tokens, offsets, None, table,
                                       0.03, 3, 4, seed=99, exp_table=et)
    b0, b1 = syn0.copy(), syn1.copy()
    st_c = native.train_batch(b0, b1, tokens, offsets, None, table,
                              0.03, 3, 4, 99, 0, "canonical", 1, et)
    assert st_c["pairs"] == st_py.pairs
    np.testing.assert_allclose(a0, b0, rtol=1e-5, atol=1e-7)
    # differs from exact mode
    c0, c1 = syn0.copy(), syn1.copy()
    native.train_batch(c0, c1, tokens, offsets, None, table,
                       0.03, 3, 4, 99, 0, "canonical", 1)
    assert not np.allclose(a0, c0, rtol=1e-6, atol=1e-9)


def test_cpp_multithread_stats_close():
    """Hogwild threads race on rows but must process the same pair count."""
    tokens, offsets, counts, table, syn0, syn1 = _problem(n_tokens=200)
    offsets = np.arange(0, 201, 20, dtype=np.int32)
    st1 = native.train_batch(syn0.copy(), syn1.copy(), tokens, offsets, None,
                             table, 0.03, 3, 4, 5, 0, "canonical", 1)
    st4 = native.train_batch(syn0.copy(), syn1.copy(), tokens, offsets, None,
                             table, 0.03, 3, 4, 5, 0, "canonical", 4)
    assert st1["pairs"] == st4["pairs"]
    assert st1["positives"] == st4["positives"]


def test_shared_negatives_native_matches_oracle():
    """Shared-negative draw layout (rng.py): C++ trainer vs Python oracle,
    and distinct from the per-context layout."""
    import numpy as np
    from glint_word2vec_amd import _cpu_native
    from glint_word2vec_amd.models import sgns
    from glint_word2vec_amd.ops import cpu_ref
    from glint_word2vec_amd.vocab import build_unigram_table
    rng = np.random.default_rng(3)
    vocab, dim = 80, 16
    tokens = rng.integers(0, vocab, 300).astype(np.int32)
    offsets = np.array([0, 100, 220, 300], dtype=np.int32)
    counts = np.bincount(tokens, minlength=vocab).astype(np.int64) + 1
    table = build_unigram_table(counts, 1009)
    syn0, syn1 = sgns.init_tables(vocab, dim, 5)

    a0, a1 = syn0.copy(), syn1.copy()
    st_py = cpu_ref.train_batch_oracle(a0, a1, tokens, offsets, None, table,
                                       0.03, 3, 4, seed=7,
                                       shared_negatives=True)
    b0, b1 = syn0.copy(), syn1.copy()
    st_c = _cpu_native.train_batch(b0, b1, tokens, offsets, None, table,
                                   0.03, 3, 4, 7, 0, "canonical", 1, None, 1)
    assert st_c["pairs"] == st_py.pairs
    assert st_c["positives"] == st_py.positives
    np.testing.assert_allclose(b0, a0, rtol=1e-5, atol=1e-7)
    np.testing.assert_allclose(b1, a1, rtol=1e-5, atol=1e-7)

    # differs from the per-context layout (different negatives drawn)
    c0, c1 = syn0.copy(), syn1.copy()
    cpu_ref.train_batch_oracle(c0, c1, tokens, offsets, None, table,
                               0.03, 3, 4, seed=7, shared_negatives=False)
    assert not np.allclose(c1, a1)
