import numpy as np
import pytest

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


def test_dim_engine_streaming_save(tmp_path):
    """DimShardedSgns.save_checkpoint streams row blocks (allgathered dim
    slices) without full host materialisation; the written checkpoint must
    equal to_host()."""
    import torch
    from glint_word2vec_amd.parallel.dim_sharded import DimShardedSgns

    rng = np.random.default_rng(7)
    v, _, _ = _make(vocab_size=23, dim=10)
    eng = DimShardedSgns(23, 10, device="cpu", seed=5,
                         counts=v.counts, table_size=997)
    tokens = torch.from_numpy(rng.integers(0, 23, 200).astype(np.int32))
    offsets = torch.from_numpy(np.array([0, 100, 200], dtype=np.int32))
    eng.train_step(tokens, offsets, 0.03, 3, 4, seed=2,
                   offsets_host=offsets.numpy())
    cfg = Word2VecConfig(vector_size=10)
    eng.save_checkpoint(str(tmp_path / "m"), cfg, v, num_shards=3,
                        block_rows=7)   # forces multiple blocks per shard
    _, v2, s0, s1 = load_model(str(tmp_path / "m"))
    h0, h1 = eng.to_host()
    assert v2.words == v.words
    np.testing.assert_array_equal(s0, h0)
    np.testing.assert_array_equal(s1, h1)


def test_dp_engine_streaming_save(tmp_path):
    """ReplicatedSgns.save_checkpoint streams from the fp32 master."""
    import torch
    from glint_word2vec_amd.parallel.replicated import ReplicatedSgns

    rng = np.random.default_rng(8)
    v, _, _ = _make(vocab_size=17, dim=8)
    eng = ReplicatedSgns(17, 8, device="cpu", seed=5, counts=v.counts,
                         table_size=997)
    tokens = rng.integers(0, 17, 150).astype(np.int32)
    offsets = np.array([0, 150], dtype=np.int32)
    eng.train_step(tokens, offsets, 0.03, 3, 4, seed=2)
    cfg = Word2VecConfig(vector_size=8)
    eng.save_checkpoint(str(tmp_path / "m"), cfg, v, num_shards=2,
                        block_rows=5)
    _, _, s0, s1 = load_model(str(tmp_path / "m"))
    h0, h1 = eng.to_host()
    np.testing.assert_array_equal(s0, h0)
    np.testing.assert_array_equal(s1, h1)


def test_mid_training_checkpoints(tmp_path):
    """checkpoint_every=N writes complete loadable models at
    <save_path>-step<k*N> during training (engine loops, engine.py)."""
    from glint_word2vec_amd import GlintWord2Vec, GlintWord2VecModel

    rng = np.random.default_rng(9)
    sents = [[f"w{rng.integers(0, 15)}" for _ in range(20)]
             for _ in range(30)]
    est = (GlintWord2Vec().setVectorSize(8).setMinCount(1).setSeed(3)
           .setNumIterations(1).setWindowSize(2).setN(3)
           .setUnigramTableSize(10000).setSubsampleRatio(0.0)
           .setCheckpointEvery(2))
    est.config.device = "cpu"
    est.config.engine = "dim"          # CPU-capable engine loop
    est.config.words_per_step = 150    # ~600 tokens -> 4 steps
    final = str(tmp_path / "model")
    model = est.fit(sents, save_path=final)
    mid = tmp_path / "model-step2"
    assert mid.is_dir(), "mid-training checkpoint missing"
    m2 = GlintWord2VecModel.load(str(mid))
    assert m2.num_words == model.num_words
    v2 = m2.to_local().vectors
    assert np.isfinite(v2).all()
    mfinal = GlintWord2VecModel.load(final)
    np.testing.assert_allclose(mfinal.to_local().vectors,
                               model.to_local().vectors,
                               rtol=1e-6, atol=1e-7)
    # the mid checkpoint differs from the final state (training continued)
    assert not np.array_equal(v2, mfinal.to_local().vectors)


def test_fit_resume_from_checkpoint(tmp_path):
    """fit(init_from=ckpt) continues training from a saved model: the
    checkpoint's vocab and matrices seed the run."""
    from glint_word2vec_amd import GlintWord2Vec

    rng = np.random.default_rng(12)
    sents = [[f"w{rng.integers(0, 12)}" for _ in range(15)]
             for _ in range(40)]

    def est():
        e = (GlintWord2Vec().setVectorSize(8).setMinCount(1).setSeed(3)
             .setNumIterations(1).setWindowSize(2).setN(3)
             .setUnigramTableSize(10000).setSubsampleRatio(0.0))
        e.config.device = "cpu"
        return e

    first = str(tmp_path / "first")
    m1 = est().fit(sents, save_path=first)
    v1 = m1.to_local().vectors
    # resume: one more iteration starting from the checkpoint
    m2 = est().fit(sents, init_from=first)
    v2 = m2.to_local().vectors
    assert m2.vocab.words == m1.vocab.words
    assert np.isfinite(v2).all()
    assert not np.array_equal(v1, v2)          # training continued
    # two-stage (1 iter + resume 1 iter) ~ training twice as long: the
    # resumed model must differ from scratch-1-iter more than noise
    assert np.abs(v2 - v1).max() > 1e-5
    # dimension mismatch is rejected
    bad = est()
    bad.config.vector_size = 16
    with pytest.raises(ValueError):
        bad.fit(sents, init_from=first)


def test_fit_resume_dim_engine(tmp_path):
    """Resume through the sharded-engine path (engine=dim, world 1 CPU)."""
    from glint_word2vec_amd import GlintWord2Vec

    rng = np.random.default_rng(13)
    sents = [[f"w{rng.integers(0, 10)}" for _ in range(12)]
             for _ in range(30)]

    def est():
        e = (GlintWord2Vec().setVectorSize(6).setMinCount(1).setSeed(3)
             .setNumIterations(1).setWindowSize(2).setN(2)
             .setUnigramTableSize(5000).setSubsampleRatio(0.0))
        e.config.device = "cpu"
        e.config.engine = "dim"
        return e

    first = str(tmp_path / "first")
    m1 = est().fit(sents, save_path=first)
    m2 = est().fit(sents, init_from=first)
    assert m2.vocab.words == m1.vocab.words
    v2 = m2.to_local().vectors
    assert np.isfinite(v2).all()
    assert not np.array_equal(m1.to_local().vectors, v2)


def test_dense_load_bf16_checkpoint(tmp_path):
    """load_model decodes bfloat16-on-disk shards (raw u16) to f32."""
    import json
    import os
    import numpy as np
    import torch
    from glint_word2vec_amd.checkpoint import load_model
    from glint_word2vec_amd.config import Word2VecConfig
    vocab, dim = 20, 8
    rng = np.random.default_rng(2)
    syn0_bf = torch.from_numpy(
        rng.standard_normal((vocab, dim)).astype(np.float32)).bfloat16()
    path = tmp_path / "bf"
    os.makedirs(path / "shards")
    with open(path / "metadata", "w") as f:
        json.dump({"numWords": vocab, "vectorSize": dim,
                   "paramMap": Word2VecConfig(vector_size=dim).to_dict()}, f)
    with open(path / "words", "w") as f:
        f.writelines(f"w{i}\n" for i in range(vocab))
    with open(path / "shards" / "index.json", "w") as f:
        json.dump({"num_shards": 1, "vocab": vocab, "dim": dim,
                   "dtype": "bfloat16", "layout": "row_mod",
                   "has_syn1": False}, f)
    syn0_bf.view(torch.uint16).numpy().tofile(
        path / "shards" / "syn0-00000.bin")
    _, _, s0, s1 = load_model(str(path))
    assert s1 is None and s0.dtype == np.float32
    np.testing.assert_allclose(s0, syn0_bf.float().numpy(), rtol=1e-6)
