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


@pytest.mark.parametrize("engine", ["dim", "row", "dp"])
def test_distributed_fit_two_ranks(tmp_path, engine):
    rdv = str(tmp_path / f"rdv_{engine}")
    mp.spawn(_worker, args=(2, rdv, str(tmp_path), engine), nprocs=2,
             join=True)
    s0 = np.load(tmp_path / "syn0_0.npy")
    s1 = np.load(tmp_path / "syn0_1.npy")
    # both ranks hold the full assembled model
    np.testing.assert_allclose(s0, s1, rtol=1e-5, atol=1e-7)
    assert np.isfinite(s0).all()


def _worker_de(rank, world, rdv, out_dir):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        import os as _os
        from glint_word2vec_amd import GlintWord2Vec
        corpus = _os.path.join(_os.path.dirname(__file__), "fixtures",
                               "de_wikipedia_articles_country_capitals.txt")
        est = (GlintWord2Vec().setSeed(1).setStepSize(0.025)
               .setUnigramTableSize(1_000_000).setSubsampleRatio(0.0)
               .setNumIterations(5))
        est.config.device = "cpu"
        est.config.engine = "dp"
        est.config.sync_every = 2
        est.config.window_mode = "reference"
        m = est.fit(corpus)
        if rank == 0:
            syns = m.find_synonyms("österreich", 10)
            res = m.analogy(["wien", "deutschland"], ["österreich"], 10)
            with open(os.path.join(out_dir, "gate.txt"), "w") as f:
                f.write(repr(([w for w, _ in syns], dict(syns).get("wien", 0),
                              [w for w, _ in res])))
            ok = ("wien" in [w for w, _ in syns] and dict(syns)["wien"] > 0.85
                  and "berlin" in [w for w, _ in res])
            np.save(os.path.join(out_dir, "gate.npy"), np.array([int(ok)]))
    finally:
        dist.destroy_process_group()


def test_dp_engine_world2_quality_gate(tmp_path):
    """The dp (replicated + delta-allreduce) engine — the default multi-GPU
    strategy for HBM-resident tables — must preserve embedding quality with
    the corpus split across 2 workers (the reference gate, Spec:290-382)."""
    rdv = str(tmp_path / "rdv_de")
    mp.spawn(_worker_de, args=(2, rdv, str(tmp_path)), nprocs=2, join=True)
    ok = np.load(tmp_path / "gate.npy")
    assert ok[0] == 1, open(tmp_path / "gate.txt").read()


def test_single_process_engine_dim_fit_quality():
    m = _fit("dim")
    syns = [w for w, _ in m.find_synonyms("aa", 3)]
    assert "a2" in syns


def test_single_process_engine_row_fit_quality():
    m = _fit("row")
    syns = [w for w, _ in m.find_synonyms("aa", 3)]
    assert "a2" in syns


def _worker_save(rank, world, rdv, out_dir, engine):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        m = None
        from glint_word2vec_amd import GlintWord2Vec
        est = (GlintWord2Vec().setVectorSize(10).setMinCount(1).setSeed(4)
               .setNumIterations(2).setWindowSize(2).setN(3)
               .setUnigramTableSize(50000).setStepSize(0.05)
               .setSubsampleRatio(0.0))
        est.config.device = "cpu"
        est.config.engine = engine
        est.config.chunk_words = 256
        m = est.fit(_corpus(), save_path=os.path.join(out_dir, "ckpt"))
        if rank == 0:
            np.save(os.path.join(out_dir, "syn0_live.npy"), m.syn0)
            np.save(os.path.join(out_dir, "syn1_live.npy"), m.syn1)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("engine", ["dim", "dp"])
def test_distributed_streaming_save(tmp_path, engine):
    """dim: collective block-streamed save (rank 1 walks the same allgather
    schedule with write=False); dp: rank-0 master stream.  The checkpoint
    on disk must equal the live assembled model."""
    from glint_word2vec_amd.checkpoint import load_model
    rdv = str(tmp_path / f"rdv_save_{engine}")
    mp.spawn(_worker_save, args=(2, rdv, str(tmp_path), engine), nprocs=2,
             join=True)
    _, _, s0, s1 = load_model(str(tmp_path / "ckpt"))
    live0 = np.load(tmp_path / "syn0_live.npy")
    live1 = np.load(tmp_path / "syn1_live.npy")
    np.testing.assert_allclose(s0, live0, rtol=1e-6, atol=1e-7)
    np.testing.assert_allclose(s1, live1, rtol=1e-6, atol=1e-7)


def _dp_conservation_worker(rank, world, rdv, out_dir):
    """Controlled dp-sync check: each rank injects known constant deltas
    between syncs; after flush every update must have landed exactly once
    on every rank (master identical everywhere = init + sum of all
    injections)."""
    import torch
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        from glint_word2vec_amd.parallel.replicated import ReplicatedSgns
        eng = ReplicatedSgns(50, 8, device="cpu", seed=3, table_size=101,
                             sync_every=1)
        init0 = eng.master0.clone()
        total_injected = torch.zeros_like(eng.syn0)
        for r in range(5):
            inj = float((rank + 1) * (r + 1))   # distinct per rank/round
            eng.syn0 += inj
            total_injected += inj
            eng.sync()
        eng.sync(flush=True)
        # sum over ranks of each round's injection
        world_inj = torch.zeros_like(total_injected)
        for rk in range(world):
            for r in range(5):
                world_inj += float((rk + 1) * (r + 1))
        torch.testing.assert_close(eng.master0, init0 + world_inj)
        torch.testing.assert_close(eng.syn0, eng.master0)
        np.save(os.path.join(out_dir, f"dpcons_{rank}.npy"),
                eng.master0.numpy())
    finally:
        dist.destroy_process_group()


def test_dp_async_sync_applies_exactly_once(tmp_path):
    rdv = str(tmp_path / "rdv_dpc")
    mp.spawn(_dp_conservation_worker, args=(2, rdv, str(tmp_path)), nprocs=2,
             join=True)
    a = np.load(tmp_path / "dpcons_0.npy")
    b = np.load(tmp_path / "dpcons_1.npy")
    np.testing.assert_allclose(a, b)


def test_row_fit_world3(tmp_path):
    """Pipelined row engine at world 3 (odd world, uneven batch counts):
    all ranks converge to identical assembled models."""
    rdv = str(tmp_path / "rdv_row3")
    mp.spawn(_worker, args=(3, rdv, str(tmp_path), "row"), nprocs=3,
             join=True)
    s0 = np.load(tmp_path / "syn0_0.npy")
    for r in (1, 2):
        np.testing.assert_allclose(s0, np.load(tmp_path / f"syn0_{r}.npy"),
                                   rtol=1e-5, atol=1e-7)
    assert np.isfinite(s0).all()
