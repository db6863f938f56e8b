"""Edge cases across the CPU implementations (GPU twins share the walker
semantics; serial-parity tests transfer these guarantees)."""
import numpy as np
import pytest

from glint_word2vec_amd import GlintWord2Vec
from glint_word2vec_amd.models import sgns
from glint_word2vec_amd.ops import cpu_ref
from glint_word2vec_amd.vocab import build_unigram_table

native = pytest.importorskip("glint_word2vec_amd._cpu_native")


def _run_both(tokens, offsets, vocab, dim, window, n_neg, table_size=101):
    counts = np.bincount(tokens, minlength=vocab).astype(np.int64) + 1
    table = build_unigram_table(counts, table_size)
    syn0, syn1 = sgns.init_tables(vocab, dim, 3)
    a0, a1 = syn0.copy(), syn1.copy()
    st_py = cpu_ref.train_batch_oracle(a0, a1, tokens, offsets, None, table,
                                       0.03, window, n_neg, seed=5)
    b0, b1 = syn0.copy(), syn1.copy()
    st_c = native.train_batch(b0, b1, tokens, offsets, None, table, 0.03,
                              window, n_neg, 5, 0, "canonical", 1)
    assert st_c["pairs"] == st_py.pairs
    np.testing.assert_allclose(a0, b0, rtol=1e-5, atol=1e-7)
    return st_py


def test_zero_negatives():
    tokens = np.arange(20, dtype=np.int32) % 7
    offsets = np.array([0, 20], dtype=np.int32)
    st = _run_both(tokens, offsets, 7, 8, 3, 0)
    assert st.pairs == st.positives > 0


def test_single_token_sentences():
    tokens = np.zeros(5, dtype=np.int32)
    offsets = np.arange(6, dtype=np.int32)
    st = _run_both(tokens, offsets, 3, 4, 2, 2)
    assert st.pairs == 0          # no context possible
    assert st.words_trained == 0


def test_empty_sentences_interleaved():
    tokens = np.array([0, 1, 2, 1, 0], dtype=np.int32)
    offsets = np.array([0, 0, 3, 3, 5, 5], dtype=np.int32)
    st = _run_both(tokens, offsets, 3, 4, 2, 2)
    assert st.pairs > 0


def test_window_larger_than_sentence():
    tokens = np.array([0, 1, 2], dtype=np.int32)
    offsets = np.array([0, 3], dtype=np.int32)
    _run_both(tokens, offsets, 3, 4, 50, 2)


def test_vocab_one():
    tokens = np.zeros(10, dtype=np.int32)
    offsets = np.array([0, 10], dtype=np.int32)
    st = _run_both(tokens, offsets, 1, 4, 2, 3)
    # negatives always collide with the (only) target -> all skipped
    assert st.pairs == st.positives


def test_dim_one():
    tokens = np.arange(12, dtype=np.int32) % 5
    offsets = np.array([0, 12], dtype=np.int32)
    _run_both(tokens, offsets, 5, 1, 2, 2)


def test_empty_batch():
    tokens = np.zeros(0, dtype=np.int32)
    offsets = np.zeros(1, dtype=np.int32)
    st = _run_both(tokens, offsets, 2, 4, 2, 2)
    assert st.pairs == 0


def test_estimator_rejects_bad_config():
    with pytest.raises(ValueError):
        GlintWord2Vec(vector_size=0)
    with pytest.raises(ValueError):
        GlintWord2Vec(window_mode="bogus")
    with pytest.raises(ValueError):
        GlintWord2Vec(engine="bogus")
    with pytest.raises(ValueError):
        GlintWord2Vec(dtype="fp8")


def test_estimator_empty_corpus():
    est = GlintWord2Vec(min_count=5)
    est.config.device = "cpu"
    with pytest.raises(ValueError, match="empty vocabulary"):
        est.fit([["rare", "words", "only"]])


def test_row_cpu_shared_negatives_raises(tmp_path):
    """shared_negatives needs the counter-RNG planner; the CPU row engine
    must refuse loudly rather than silently train different semantics."""
    import numpy as np
    import pytest
    from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns
    eng = RowShardedSgns(50, 8, device="cpu", seed=1, table_size=101,
                         shared_negatives=True)
    with pytest.raises(NotImplementedError):
        eng.make_plan(np.zeros(10, dtype=np.int32),
                      np.array([0, 10], dtype=np.int32), 2, 3,
                      np.random.default_rng(0))


def test_sharded_load_missing_files(tmp_path):
    import pytest
    from glint_word2vec_amd.serving import ShardedWord2VecModel
    with pytest.raises(FileNotFoundError):
        ShardedWord2VecModel.load(str(tmp_path / "nope"), device="cpu")


def test_config_rejects_bad_values():
    import pytest
    from glint_word2vec_amd.config import Word2VecConfig
    for kw in ({"update_mode": "x"}, {"hybrid_hot_rows": -1},
               {"hybrid_skip_rows": -2}, {"engine": "zz"},
               {"vector_size": 0}):
        with pytest.raises(ValueError):
            Word2VecConfig(**kw)
