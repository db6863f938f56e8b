"""Streaming / distributed checkpointing (the reference's per-PS parallel
shard save, mllib:493-498)."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from glint_word2vec_amd.checkpoint import (load_model, save_model,
                                           save_model_streaming)
from glint_word2vec_amd.config import Word2VecConfig
from glint_word2vec_amd.vocab import build_vocab

pytest.importorskip("glint_word2vec_amd._cpu_native")


def _vocab(n=23):
    return build_vocab([[f"w{i}"] * (n - i + 1) for i in range(n)], min_count=1)


def test_streaming_save_matches_host_save(tmp_path):
    v = _vocab()
    rng = np.random.default_rng(0)
    syn0 = rng.standard_normal((v.num_words, 12)).astype(np.float32)
    syn1 = rng.standard_normal((v.num_words, 12)).astype(np.float32)
    cfg = Word2VecConfig(vector_size=12)
    save_model(str(tmp_path / "host"), cfg, v, syn0, syn1, num_shards=3)

    def row_block(which, r0, r1):
        return (syn0 if which == 0 else syn1)[r0:r1]

    save_model_streaming(str(tmp_path / "stream"), cfg, v, row_block,
                         num_shards=3, block_rows=5)
    _, _, a0, a1 = load_model(str(tmp_path / "host"))
    _, _, b0, b1 = load_model(str(tmp_path / "stream"))
    np.testing.assert_array_equal(a0, b0)
    np.testing.assert_array_equal(a1, b1)


def _worker(rank, world, rdv, out_dir):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns
        v = _vocab(31)
        cfg = Word2VecConfig(vector_size=8)
        counts = np.ones(31, dtype=np.int64)
        eng = RowShardedSgns(31, 8, device="cpu", seed=3, counts=counts,
                             table_size=101)
        s0_full, s1_full = eng.to_host()
        p = os.path.join(out_dir, "ckpt")
        eng.save_checkpoint(p, cfg, v)
        # reload into a fresh engine and compare
        eng2 = RowShardedSgns(31, 8, device="cpu", seed=99, counts=counts,
                              table_size=101)
        eng2.load_checkpoint(p)
        r0, r1 = eng2.to_host()
        if rank == 0:
            np.save(os.path.join(out_dir, "orig.npy"), s0_full)
            np.save(os.path.join(out_dir, "reload.npy"), r0)
    finally:
        dist.destroy_process_group()


def test_fit_save_path_no_materialize(tmp_path):
    """fit(save_path=..., materialize=False): checkpoint written engine-side,
    nothing assembled on host (the 80M-vocab production path)."""
    from glint_word2vec_amd import GlintWord2Vec, GlintWord2VecModel
    sents = [["a", "b", "c", "a", "b"]] * 100
    est = (GlintWord2Vec().setVectorSize(8).setMinCount(1).setSeed(2)
           .setUnigramTableSize(1000).setNumIterations(2)
           .setSubsampleRatio(0.0))
    est.config.device = "cpu"
    est.config.engine = "row"     # has engine-side save_checkpoint
    p = str(tmp_path / "big_model")
    out = est.fit(sents, save_path=p, materialize=False)
    assert out is None
    m = GlintWord2VecModel.load(p)
    assert set(m.vocab.words) == {"a", "b", "c"}
    assert np.isfinite(m.syn0).all()


def test_row_sharded_distributed_checkpoint(tmp_path):
    rdv = str(tmp_path / "rdv")
    mp.spawn(_worker, args=(2, rdv, str(tmp_path)), nprocs=2, join=True)
    orig = np.load(tmp_path / "orig.npy")
    reload_ = np.load(tmp_path / "reload.npy")
    np.testing.assert_allclose(reload_, orig, rtol=1e-6)
    # and the host loader reads the distributed checkpoint directly
    from glint_word2vec_amd import GlintWord2VecModel
    m = GlintWord2VecModel.load(str(tmp_path / "ckpt"))
    np.testing.assert_allclose(m.syn0, orig, rtol=1e-6)


def test_num_shards_knob_controls_shard_count(tmp_path):
    """setNumParameterServers -> num_shards controls how many shard files
    the fit-time checkpoint writes (the reference's per-PS shard files)."""
    import os
    import numpy as np
    from glint_word2vec_amd import GlintWord2Vec
    rng = np.random.default_rng(0)
    sents = [[f"w{rng.integers(0, 30)}" for _ in range(8)]
             for _ in range(200)]
    est = (GlintWord2Vec().setVectorSize(8).setMinCount(1).setSeed(1)
           .setNumIterations(1).setUnigramTableSize(1000)
           .setNumParameterServers(3))
    est.config.device = "cpu"
    out = str(tmp_path / "m")
    est.fit(sents, save_path=out, materialize=False)
    files = sorted(os.listdir(os.path.join(out, "shards")))
    assert "syn0-00002.bin" in files and "syn0-00003.bin" not in files
