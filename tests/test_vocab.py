import numpy as np
import pytest

from glint_word2vec_amd.vocab import (build_unigram_table, build_vocab,
                                      encode_sentences, keep_probabilities,
                                      Vocabulary)


def test_build_vocab_counts_filter_sort():
    sents = [["a", "b", "a", "c"], ["a", "b"], ["d"]]
    v = build_vocab(sents, min_count=2)
    assert v.words == ["a", "b"]          # sorted desc by count: a=3, b=2
    assert v.counts.tolist() == [3, 2]
    assert v.index == {"a": 0, "b": 1}
    assert v.train_words_count == 5


def test_build_vocab_deterministic_ties():
    sents = [["z", "y", "x"], ["x", "y", "z"]]
    v = build_vocab(sents, min_count=1)
    assert v.words == ["x", "y", "z"]     # count ties broken by word


def test_encode_sentences_oov_and_chunking():
    v = build_vocab([["a", "b"] * 5], min_count=1)
    enc = list(encode_sentences([["a", "oov", "b"] * 4], v, max_sentence_length=5))
    assert len(enc) == 2                  # 8 tokens chunked at 5
    assert enc[0].tolist() == [0, 1, 0, 1, 0]
    assert enc[1].tolist() == [1, 0, 1]


def test_keep_probabilities_intended_math():
    counts = np.array([1000, 10], dtype=np.int64)
    kp = keep_probabilities(counts, 1010, subsample_ratio=0.05)
    # frequent word gets kp < 1; rare word (p < ratio) is clipped to 1
    p0 = 1000 / 1010
    expected0 = (np.sqrt(p0 / 0.05) + 1) * (0.05 / p0)
    assert kp[0] == pytest.approx(expected0, rel=1e-6)
    assert kp[1] == 1.0


def test_keep_probabilities_legacy_noop():
    counts = np.array([1000, 10], dtype=np.int64)
    kp = keep_probabilities(counts, 1010, 1e-3, legacy=True)
    assert np.all(kp == 1.0)


def test_unigram_table_proportions():
    counts = np.array([100, 10, 1], dtype=np.int64)
    table = build_unigram_table(counts, 100_000, power=0.75)
    frac = np.bincount(table, minlength=3) / len(table)
    w = counts.astype(float) ** 0.75
    expect = w / w.sum()
    assert np.allclose(frac, expect, atol=1e-3)
    assert table.dtype == np.int32


def test_words_roundtrip(tmp_path):
    v = build_vocab([["a", "b", "a"]], min_count=1)
    p = str(tmp_path / "words")
    v.save_words(p)
    v2 = Vocabulary.load_words(p, v.counts)
    assert v2.words == v.words
    assert v2.index == v.index


def test_update_mode_resolution():
    from glint_word2vec_amd.config import Word2VecConfig
    c = Word2VecConfig()
    assert c.update_mode == "hybrid"
    assert c.effective_atomic_below() == c.hybrid_hot_rows == 32768
    assert c.effective_atomic_floor() == 16
    assert c.effective_atomic_floor(1_000_000) == 16
    assert c.effective_atomic_floor(3611) == 3      # small-vocab scaling
    c2 = Word2VecConfig(update_mode="hogwild")
    assert c2.effective_atomic_below() == 0
    c3 = Word2VecConfig(update_mode="atomic")
    assert c3.effective_atomic_below() == 2 ** 31 - 1
    assert c3.effective_atomic_floor(100) == 0
    # deprecated alias wins, even assigned post-construction
    c4 = Word2VecConfig(atomic_updates=False)
    assert c4.resolved_update_mode() == "hogwild"
    c5 = Word2VecConfig()
    c5.atomic_updates = True
    assert c5.resolved_update_mode() == "atomic"


def test_choose_engine_policy():
    from glint_word2vec_amd.config import choose_engine, round_stride_py
    assert choose_engine(1_000_000, 300, 2, 1) == "fused"
    assert choose_engine(1_000_000, 300, 2, 8) == "dp"
    assert choose_engine(10_000_000, 300, 2, 8) == "row"
    assert round_stride_py(300) == 320
    assert round_stride_py(1024) == 1024
