"""End-to-end estimator/model API tests on a tiny synthetic corpus (fast) —
mirrors the reference's IT scenario surface minus Spark."""
import numpy as np
import pytest

from glint_word2vec_amd import GlintWord2Vec, GlintWord2VecModel


@pytest.fixture(scope="module")
def tiny_model():
    rng = np.random.default_rng(5)
    # "aa" and "a2" are interchangeable (identical context distribution) ->
    # they must come out as nearest syn0-cosine neighbours.  Same for the
    # (cc, c2) pair with a different context word.
    sents = []
    for _ in range(600):
        if rng.random() < 0.5:
            head = "aa" if rng.random() < 0.5 else "a2"
            tail = "bb"
        else:
            head = "cc" if rng.random() < 0.5 else "c2"
            tail = "dd"
        filler = [f"x{rng.integers(0, 20)}" for _ in range(3)]
        sents.append([head, tail] + filler + [head, tail])
    est = (GlintWord2Vec(input_col="s", output_col="v")
           .setVectorSize(24).setMinCount(1).setSeed(11)
           .setNumIterations(8).setWindowSize(2).setN(5)
           .setUnigramTableSize(100000).setStepSize(0.05)
           .setSubsampleRatio(0.0))
    est.config.device = "cpu"
    model = est.fit(sents)
    return model


def test_fit_produces_vectors(tiny_model):
    assert tiny_model.num_words > 6
    assert tiny_model.vector_size == 24
    v = tiny_model.transform("aa")
    assert v.shape == (24,)
    assert np.isfinite(v).all()


def test_find_synonyms_structure(tiny_model):
    syns = tiny_model.find_synonyms("aa", 3)
    assert len(syns) == 3
    words = [w for w, _ in syns]
    assert "aa" not in words
    # a2 is distributionally identical to aa -> nearest neighbour
    assert words[0] == "a2"


def test_find_synonyms_by_vector(tiny_model):
    vec = tiny_model.transform("aa")
    syns = tiny_model.find_synonyms(vec, 2)
    assert syns[0][0] == "aa"            # by-vector search keeps the word itself
    assert syns[0][1] == pytest.approx(1.0, abs=1e-5)


def test_transform_sentence_average(tiny_model):
    va = tiny_model.transform("aa")
    vb = tiny_model.transform("bb")
    vs = tiny_model.transform(["aa", "bb"])
    assert np.allclose(vs, (va + vb) / 2, atol=1e-6)


def test_transform_oov_only_sentence(tiny_model):
    v = tiny_model.transform(["zzznotaword"])
    assert np.all(v == 0)


def test_transform_dataframe(tiny_model):
    pd = pytest.importorskip("pandas")
    df = pd.DataFrame({"s": [["aa", "bb"], ["cc"]], "other": [1, 2]})
    out = tiny_model.transform(df)
    assert "v" in out.columns
    assert "other" in out.columns        # multi-column pass-through (Spec:260-288)
    assert out["v"][0].shape == (24,)


def test_get_vectors(tiny_model):
    vecs = tiny_model.get_vectors()
    assert len(vecs) == tiny_model.num_words   # Spec:384-398
    assert vecs["aa"].shape == (24,)


def test_save_load_roundtrip(tiny_model, tmp_path):
    p = str(tmp_path / "model")
    tiny_model.save(p, num_shards=3)
    m2 = GlintWord2VecModel.load(p)
    assert m2.vocab.words == tiny_model.vocab.words
    assert np.array_equal(m2.syn0, tiny_model.syn0)
    s1 = tiny_model.find_synonyms("aa", 3)
    s2 = m2.find_synonyms("aa", 3)
    assert [w for w, _ in s1] == [w for w, _ in s2]


def test_to_local(tiny_model, tmp_path):
    local = tiny_model.to_local()
    assert np.array_equal(local["aa"], tiny_model.transform("aa"))
    local.save(str(tmp_path / "vecs.txt"))
    head = open(tmp_path / "vecs.txt").readline().split()
    assert int(head[0]) == tiny_model.num_words


def test_analogy_api(tiny_model):
    res = tiny_model.analogy(["aa"], ["bb"], num=3)
    assert len(res) == 3
    assert all(w not in ("aa", "bb") for w, _ in res)


def test_fit_from_path(tmp_path):
    p = tmp_path / "corpus.txt"
    p.write_text("a b a b a b\n" * 50)
    est = GlintWord2Vec().setVectorSize(8).setMinCount(1).setSeed(1) \
        .setUnigramTableSize(1000).setNumIterations(2)
    est.config.device = "cpu"
    m = est.fit(str(p))
    assert set(m.vocab.words) == {"a", "b"}


def test_find_synonyms_df(tiny_model):
    pd = pytest.importorskip("pandas")
    df = tiny_model.find_synonyms_df("aa", 4)
    assert list(df.columns) == ["word", "similarity"]   # ml:390-420 shape
    assert len(df) == 4
    assert df["similarity"].is_monotonic_decreasing


def test_transform_words_batched(tiny_model):
    # mllib:529-543: iterator-of-words batched lookup
    out = tiny_model.transform_words(["aa", "bb", "aa"])
    assert out.shape == (3, 24)
    np.testing.assert_array_equal(out[0], out[2])


def test_metadata_roundtrips_all_knobs(tiny_model, tmp_path):
    """Every config knob (incl. semantics switches) survives save/load —
    the reference persists all params in metadata (ml:187-194, 504-560)."""
    tiny_model.config.window_mode = "reference"
    tiny_model.config.sigmoid_mode = "lut"
    p = str(tmp_path / "m")
    tiny_model.save(p)
    from glint_word2vec_amd import GlintWord2VecModel
    m2 = GlintWord2VecModel.load(p)
    assert m2.config.window_mode == "reference"
    assert m2.config.sigmoid_mode == "lut"
    assert m2.config.vector_size == tiny_model.config.vector_size
    assert m2.config.unigram_table_size == tiny_model.config.unigram_table_size
    tiny_model.config.window_mode = "canonical"
    tiny_model.config.sigmoid_mode = "exact"


def test_fit_reproducible_same_seed():
    """Same seed, single worker -> bitwise-identical model (counter-based
    RNG; no hidden global state)."""
    sents = [["p", "q", "r", "p", "q"]] * 80
    def go():
        est = (GlintWord2Vec().setVectorSize(12).setMinCount(1).setSeed(9)
               .setUnigramTableSize(500).setNumIterations(2)
               .setSubsampleRatio(0.0))
        est.config.device = "cpu"
        est.config.num_partitions = 1
        return est.fit(sents)
    m1, m2 = go(), go()
    np.testing.assert_array_equal(m1.syn0, m2.syn0)
    np.testing.assert_array_equal(m1.syn1, m2.syn1)


def test_stop_noop(tiny_model):
    tiny_model.stop()   # must not raise without dist initialised


def test_find_synonyms_batch_matches_single():
    import numpy as np
    from glint_word2vec_amd.config import Word2VecConfig
    from glint_word2vec_amd.estimator import GlintWord2VecModel
    from glint_word2vec_amd.vocab import Vocabulary
    rng = np.random.default_rng(2)
    vocab, dim = 40, 12
    words = [f"w{i}" for i in range(vocab)]
    voc = Vocabulary(words=words, counts=np.ones(vocab, dtype=np.int64),
                     index={w: i for i, w in enumerate(words)},
                     train_words_count=vocab)
    syn0 = rng.standard_normal((vocab, dim)).astype(np.float32)
    m = GlintWord2VecModel(Word2VecConfig(vector_size=dim), voc, syn0)
    batch = m.find_synonyms_batch(["w3", syn0[7] * 2.0], 5)
    for got, q in zip(batch, ["w3", syn0[7] * 2.0]):
        ref = m.find_synonyms(q, 5)
        assert [w for w, _ in got] == [w for w, _ in ref]
        np.testing.assert_allclose([c for _, c in got],
                                   [c for _, c in ref], rtol=1e-5)
