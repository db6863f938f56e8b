"""End-to-end estimator fit under a 2-process group (gloo), both engines.
This is the CPU stand-in for `torchrun --nproc-per-node N fit` on GPUs."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

pytest.importorskip("glint_word2vec_amd._cpu_native")


def _corpus():
    rng = np.random.default_rng(5)
    sents = []
    for _ in range(300):
        head = "aa" if rng.random() < 0.5 else "a2"
        filler = [f"x{rng.integers(0, 15)}" for _ in range(3)]
        sents.append([head, "bb"] + filler + [head, "bb"])
    return sents


def _fit(engine):
    from glint_word2vec_amd import GlintWord2Vec
    est = (GlintWord2Vec().setVectorSize(16).setMinCount(1).setSeed(4)
           .setNumIterations(4).setWindowSize(2).setN(5)
           .setUnigramTableSize(50000).setStepSize(0.05)
           .setSubsampleRatio(0.0))
    est.config.device = "cpu"
    est.config.engine = engine
    est.config.chunk_words = 256
    return est.fit(_corpus())


def _worker(rank, world, rdv, out_dir, engine):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        m = _fit(engine)
        np.save(os.path.join(out_dir, f"syn0_{rank}.npy"), m.syn0)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("engine", ["dim", "row"])
def test_distributed_fit_two_ranks(tmp_path, engine):
    rdv = str(tmp_path / f"rdv_{engine}")
    mp.spawn(_worker, args=(2, rdv, str(tmp_path), engine), nprocs=2,
             join=True)
    s0 = np.load(tmp_path / "syn0_0.npy")
    s1 = np.load(tmp_path / "syn0_1.npy")
    # both ranks hold the full assembled model
    np.testing.assert_allclose(s0, s1, rtol=1e-5, atol=1e-7)
    assert np.isfinite(s0).all()


def test_single_process_engine_dim_fit_quality():
    m = _fit("dim")
    syns = [w for w, _ in m.find_synonyms("aa", 3)]
    assert "a2" in syns


def test_single_process_engine_row_fit_quality():
    m = _fit("row")
    syns = [w for w, _ in m.find_synonyms("aa", 3)]
    assert "a2" in syns
