"""Sharded model load + serving (serving.ShardedWord2VecModel): streaming
load from any shard layout, findSynonyms/transform parity with the dense
model, world-2 gloo equivalence with world-1."""
import json
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

pytest.importorskip("glint_word2vec_amd._cpu_native")

from glint_word2vec_amd.checkpoint import save_model
from glint_word2vec_amd.config import Word2VecConfig
from glint_word2vec_amd.estimator import GlintWord2VecModel
from glint_word2vec_amd.vocab import Vocabulary


def _make_checkpoint(path, num_shards, vocab=60, dim=24, seed=1):
    rng = np.random.default_rng(seed)
    words = [f"w{i:03d}" for i in range(vocab)]
    counts = np.maximum((vocab - np.arange(vocab)) * 3, 1).astype(np.int64)
    voc = Vocabulary(words=words, counts=counts,
                     index={w: i for i, w in enumerate(words)},
                     train_words_count=int(counts.sum()))
    syn0 = rng.standard_normal((vocab, dim)).astype(np.float32)
    syn1 = rng.standard_normal((vocab, dim)).astype(np.float32) * 0.1
    save_model(path, Word2VecConfig(vector_size=dim), voc, syn0, syn1,
               num_shards=num_shards)
    return voc, syn0, syn1


@pytest.mark.parametrize("num_shards", [1, 3])
def test_sharded_load_world1_matches_dense(tmp_path, num_shards):
    path = str(tmp_path / "model")
    voc, syn0, _ = _make_checkpoint(path, num_shards)
    dense = GlintWord2VecModel.load(path)
    sharded = GlintWord2VecModel.load_sharded(path, device="cpu")
    assert sharded.num_words == dense.num_words
    assert sharded.vector_size == dense.vector_size
    # row pull
    np.testing.assert_allclose(sharded.get_vector("w007"), syn0[7],
                               rtol=1e-6)
    np.testing.assert_allclose(sharded.transform_words(["w003", "w042"]),
                               syn0[[3, 42]], rtol=1e-6)
    # findSynonyms parity (cosine values + word ranking)
    for q in ("w005", syn0[11] * 0.5):
        d = dense.find_synonyms(q, 7)
        s = sharded.find_synonyms(q, 7)
        assert [w for w, _ in d] == [w for w, _ in s]
        np.testing.assert_allclose([c for _, c in d], [c for _, c in s],
                                   rtol=1e-5, atol=1e-6)
    # batched == single (values up to GEMM reduction-order jitter)
    batch = sharded.find_synonyms_batch(["w005", "w010"], 5)
    for got, q in ((batch[0], "w005"), (batch[1], "w010")):
        ref = sharded.find_synonyms(q, 5)
        assert [w for w, _ in got] == [w for w, _ in ref]
        np.testing.assert_allclose([c for _, c in got],
                                   [c for _, c in ref], rtol=1e-5)
    # sentence-average transform
    sents = [["w001", "w002", "w003"], ["w010"], ["notinvocab"]]
    got = sharded.transform_sentences(sents)
    np.testing.assert_allclose(got[0], syn0[[1, 2, 3]].mean(0), rtol=1e-5,
                               atol=1e-6)
    np.testing.assert_allclose(got[1], syn0[10], rtol=1e-6)
    np.testing.assert_allclose(got[2], 0.0)
    # analogy runs
    assert len(sharded.analogy(["w001", "w002"], ["w003"], 3)) == 3
    df = sharded.find_synonyms_df("w005", 4)
    assert list(df.columns) == ["word", "similarity"] and len(df) == 4


def test_sharded_load_row_range_layout(tmp_path):
    """Streamed (row_range) checkpoints load too — the layout the training
    engines write from HBM."""
    from glint_word2vec_amd.checkpoint import save_model_streaming
    rng = np.random.default_rng(3)
    vocab, dim = 50, 16
    words = [f"t{i}" for i in range(vocab)]
    voc = Vocabulary(words=words, counts=np.ones(vocab, dtype=np.int64),
                     index={w: i for i, w in enumerate(words)},
                     train_words_count=vocab)
    syn0 = rng.standard_normal((vocab, dim)).astype(np.float32)
    path = str(tmp_path / "m2")
    save_model_streaming(path, Word2VecConfig(vector_size=dim), voc,
                         lambda which, r0, r1: (syn0 if which == 0
                                                else syn0 * 0)[r0:r1],
                         num_shards=4, block_rows=7)
    sharded = GlintWord2VecModel.load_sharded(path, device="cpu")
    np.testing.assert_allclose(sharded.get_vector("t13"), syn0[13],
                               rtol=1e-6)
    np.testing.assert_allclose(sharded.transform_words(words), syn0,
                               rtol=1e-6)


def _serve_worker(rank, world, rdv, path, out_dir):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        from glint_word2vec_amd.estimator import GlintWord2VecModel
        m = GlintWord2VecModel.load_sharded(path, device="cpu")
        syn = m.find_synonyms("w005", 7)
        vec = m.get_vector("w042")
        avg = m.transform_sentences([["w001", "w002"], ["w010", "w011"]])
        batch = m.find_synonyms_batch(["w002", "w020"], 5)
        with open(os.path.join(out_dir, f"serve_{rank}.json"), "w") as f:
            json.dump({"syn": syn, "vec": vec.tolist(), "avg": avg.tolist(),
                       "batch": batch}, f)
    finally:
        dist.destroy_process_group()


def test_sharded_serving_world2_matches_world1(tmp_path):
    path = str(tmp_path / "model")
    _make_checkpoint(path, num_shards=3)
    rdv = str(tmp_path / "rdv")
    mp.spawn(_serve_worker, args=(2, rdv, path, str(tmp_path)), nprocs=2,
             join=True)
    with open(tmp_path / "serve_0.json") as f:
        r0 = json.load(f)
    with open(tmp_path / "serve_1.json") as f:
        r1 = json.load(f)
    assert r0 == r1, "both ranks must return identical results"
    # and world-2 == world-1
    one = GlintWord2VecModel.load_sharded(path, device="cpu")
    syn1p = one.find_synonyms("w005", 7)
    assert [w for w, _ in syn1p] == [w for w, _ in r0["syn"]]
    np.testing.assert_allclose([c for _, c in syn1p],
                               [c for _, c in r0["syn"]], rtol=1e-5,
                               atol=1e-6)
    np.testing.assert_allclose(one.get_vector("w042"), r0["vec"], rtol=1e-5)
    np.testing.assert_allclose(
        one.transform_sentences([["w001", "w002"], ["w010", "w011"]]),
        r0["avg"], rtol=1e-5, atol=1e-6)


def test_row_engine_streaming_resume_any_shards(tmp_path):
    """RowShardedSgns.load_checkpoint streams from a checkpoint written
    with a different shard count/layout."""
    from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns
    path = str(tmp_path / "m3")
    voc, syn0, syn1 = _make_checkpoint(path, num_shards=4)
    eng = RowShardedSgns(60, 24, device="cpu", seed=9)
    eng.load_checkpoint(path)
    got0, got1 = eng.to_host()
    np.testing.assert_allclose(got0, syn0, rtol=1e-6)
    np.testing.assert_allclose(got1, syn1, rtol=1e-6)


def test_sharded_to_local_and_export(tmp_path):
    path = str(tmp_path / "model")
    voc, syn0, _ = _make_checkpoint(path, num_shards=2)
    m = GlintWord2VecModel.load_sharded(path, device="cpu")
    local = m.to_local()
    np.testing.assert_allclose(local.vectors, syn0, rtol=1e-6)
    assert local.words[:3] == ["w000", "w001", "w002"]
    with pytest.raises(ValueError):
        m.to_local(max_bytes=10)
    out = str(tmp_path / "vecs.txt")
    m.export_text(out, block=17)
    lines = open(out, encoding="utf-8").read().splitlines()
    assert lines[0] == "60 24"
    assert lines[1].split()[0] == "w000"
    got = np.array([float(x) for x in lines[8].split()[1:]])
    np.testing.assert_allclose(got, syn0[7], rtol=1e-4, atol=1e-5)


def test_sharded_model_resave_roundtrip(tmp_path):
    path = str(tmp_path / "model")
    _, syn0, _ = _make_checkpoint(path, num_shards=3)
    m = GlintWord2VecModel.load_sharded(path, device="cpu")
    out = str(tmp_path / "resaved")
    m.save(out, num_shards=2)
    m2 = GlintWord2VecModel.load_sharded(out, device="cpu")
    np.testing.assert_allclose(m2.get_vector("w007"), syn0[7], rtol=1e-6)
    assert m2.num_words == 60


def test_sharded_load_bf16_checkpoint(tmp_path):
    """bf16-on-disk checkpoints (raw u16) load through the sharded path."""
    import torch
    from glint_word2vec_amd.serving import ShardedWord2VecModel
    rng = np.random.default_rng(7)
    vocab, dim = 40, 16
    syn0 = rng.standard_normal((vocab, dim)).astype(np.float32)
    syn0_bf = torch.from_numpy(syn0).bfloat16()
    path = tmp_path / "mbf"
    os.makedirs(path / "shards")
    with open(path / "metadata", "w") as f:
        json.dump({"numWords": vocab, "vectorSize": dim,
                   "paramMap": Word2VecConfig(vector_size=dim).to_dict()}, f)
    with open(path / "words", "w") as f:
        f.writelines(f"w{i:03d}\n" for i in range(vocab))
    with open(path / "shards" / "index.json", "w") as f:
        json.dump({"num_shards": 1, "vocab": vocab, "dim": dim,
                   "dtype": "bfloat16", "layout": "row_mod",
                   "has_syn1": False}, f)
    syn0_bf.view(torch.uint16).numpy().tofile(path / "shards"
                                              / "syn0-00000.bin")
    m = ShardedWord2VecModel.load(str(path), device="cpu")
    np.testing.assert_allclose(m.get_vector("w007"),
                               syn0_bf[7].float().numpy(), rtol=1e-6)


def test_find_synonyms_batch_chunking_equivalence(tmp_path):
    """Tiny max_score_bytes forces internal query chunking; results must
    equal the unchunked path (both sharded and dense models)."""
    path = str(tmp_path / "model")
    _, syn0, _ = _make_checkpoint(path, num_shards=2)
    qs = [syn0[3] * 1.5, syn0[11], syn0[20] * 0.2, syn0[33]]
    sharded = GlintWord2VecModel.load_sharded(path, device="cpu")
    a = sharded.find_synonyms_batch(qs, 5)
    b = sharded.find_synonyms_batch(qs, 5, max_score_bytes=1)
    assert [[w for w, _ in r] for r in a] == [[w for w, _ in r] for r in b]
    np.testing.assert_allclose(
        [[c for _, c in r] for r in a],
        [[c for _, c in r] for r in b], rtol=1e-5)
    dense = GlintWord2VecModel.load(path)
    c = dense.find_synonyms_batch(qs, 5)
    d = dense.find_synonyms_batch(qs, 5, max_score_bytes=1)
    assert [[w for w, _ in r] for r in c] == [[w for w, _ in r] for r in d]


def test_sharded_transform_polymorphic(tmp_path):
    import pandas as pd
    path = str(tmp_path / "model")
    _, syn0, _ = _make_checkpoint(path, num_shards=2)
    import glint_word2vec_amd as g
    m = g.load_sharded(path, device="cpu")
    np.testing.assert_allclose(m.transform("w004"), syn0[4], rtol=1e-6)
    np.testing.assert_allclose(m.transform(["w001", "w002"]),
                               syn0[[1, 2]].mean(0), rtol=1e-5, atol=1e-6)
    df = pd.DataFrame({"sentence": [["w001", "w002"], ["w010"]],
                       "other": [1, 2]})
    out = m.transform(df)
    assert list(out.columns) == ["sentence", "other", "vector"]
    np.testing.assert_allclose(out["vector"][1], syn0[10], rtol=1e-6)


def _serve_worker_small(rank, world, rdv, path, out_dir):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"file://{rdv}")
    try:
        m = GlintWord2VecModel.load_sharded(path, device="cpu")
        out = {"syn": m.find_synonyms("w005", 5),
               "vec": m.get_vector("w010").tolist(),
               "avg": m.transform_sentences([["w001", "w002"]]).tolist()}
        with open(os.path.join(out_dir, f"serve3_{rank}.json"), "w") as f:
            json.dump(out, f)
    finally:
        dist.destroy_process_group()


def test_sharded_serving_world3_uneven(tmp_path):
    """World 3 over a 2-shard checkpoint: uneven per-rank row counts, all
    ranks agree and match world-1."""
    path = str(tmp_path / "model")
    _, syn0, _ = _make_checkpoint(path, num_shards=2, vocab=31)  # 31 % 3 != 0
    rdv = str(tmp_path / "rdv3")
    mp.spawn(_serve_worker_small, args=(3, rdv, path, str(tmp_path)),
             nprocs=3, join=True)
    outs = []
    for r in range(3):
        with open(tmp_path / f"serve3_{r}.json") as f:
            outs.append(json.load(f))
    assert outs[0] == outs[1] == outs[2]
    np.testing.assert_allclose(outs[0]["vec"], syn0[10], rtol=1e-5)
    one = GlintWord2VecModel.load_sharded(path, device="cpu")
    ref = one.find_synonyms("w005", 5)
    assert [w for w, _ in ref] == [w for w, _ in outs[0]["syn"]]
