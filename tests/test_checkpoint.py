import numpy as np

from glint_word2vec_amd.checkpoint import (load_model, save_model,
                                           save_word2vec_text)
from glint_word2vec_amd.config import Word2VecConfig
from glint_word2vec_amd.vocab import build_vocab


def _make(vocab_size=11, dim=6):
    sents = [[f"w{i}"] * (vocab_size - i + 1) for i in range(vocab_size)]
    v = build_vocab(sents, min_count=1)
    rng = np.random.default_rng(0)
    syn0 = rng.standard_normal((v.num_words, dim)).astype(np.float32)
    syn1 = rng.standard_normal((v.num_words, dim)).astype(np.float32)
    return v, syn0, syn1


def test_roundtrip_single_shard(tmp_path):
    v, syn0, syn1 = _make()
    cfg = Word2VecConfig(vector_size=6)
    save_model(str(tmp_path / "m"), cfg, v, syn0, syn1, num_shards=1)
    cfg2, v2, s0, s1 = load_model(str(tmp_path / "m"))
    assert cfg2.vector_size == 6
    assert v2.words == v.words
    assert np.array_equal(s0, syn0)
    assert np.array_equal(s1, syn1)


def test_roundtrip_multi_shard(tmp_path):
    """Shard count at save time must not change the reassembled matrix."""
    v, syn0, syn1 = _make(vocab_size=13, dim=5)
    cfg = Word2VecConfig(vector_size=5)
    save_model(str(tmp_path / "m4"), cfg, v, syn0, syn1, num_shards=4)
    _, _, s0, s1 = load_model(str(tmp_path / "m4"))
    assert np.array_equal(s0, syn0)
    assert np.array_equal(s1, syn1)


def test_no_syn1(tmp_path):
    v, syn0, _ = _make()
    cfg = Word2VecConfig(vector_size=6)
    save_model(str(tmp_path / "m"), cfg, v, syn0, None, num_shards=2)
    _, _, s0, s1 = load_model(str(tmp_path / "m"))
    assert s1 is None
    assert np.array_equal(s0, syn0)


def test_word2vec_text_format(tmp_path):
    p = str(tmp_path / "vecs.txt")
    save_word2vec_text(p, ["a", "b"], np.array([[1.0, 2.0], [3.0, 4.0]], dtype=np.float32))
    lines = open(p).read().strip().split("\n")
    assert lines[0] == "2 2"
    assert lines[1].startswith("a 1 2")
