"""HTTP serving tier (server.build_app) over dense and sharded models."""
import numpy as np
import pytest

pytest.importorskip("fastapi")
pytest.importorskip("glint_word2vec_amd._cpu_native")

from glint_word2vec_amd.checkpoint import save_model
from glint_word2vec_amd.config import Word2VecConfig
from glint_word2vec_amd.estimator import GlintWord2VecModel
from glint_word2vec_amd.server import build_app
from glint_word2vec_amd.vocab import Vocabulary


@pytest.fixture(scope="module")
def model_dir(tmp_path_factory):
    rng = np.random.default_rng(1)
    vocab, dim = 50, 12
    words = [f"w{i:02d}" for i in range(vocab)]
    voc = Vocabulary(words=words, counts=np.ones(vocab, dtype=np.int64),
                     index={w: i for i, w in enumerate(words)},
                     train_words_count=vocab)
    syn0 = rng.standard_normal((vocab, dim)).astype(np.float32)
    path = str(tmp_path_factory.mktemp("srv") / "m")
    save_model(path, Word2VecConfig(vector_size=dim), voc, syn0,
               num_shards=2)
    return path, syn0


@pytest.mark.parametrize("sharded", [False, True])
def test_http_endpoints(model_dir, sharded):
    from fastapi.testclient import TestClient
    path, syn0 = model_dir
    model = (GlintWord2VecModel.load_sharded(path, device="cpu") if sharded
             else GlintWord2VecModel.load(path))
    client = TestClient(build_app(model))
    h = client.get("/health").json()
    assert h["status"] == "ok" and h["vocab"] == 50 and h["dim"] == 12
    v = client.get("/vector", params={"word": "w07"}).json()
    np.testing.assert_allclose(v["vector"], syn0[7], rtol=1e-6)
    assert client.get("/vector", params={"word": "nope"}).status_code == 404
    r = client.post("/synonyms", json={"query": "w05", "num": 3}).json()
    assert len(r) == 1 and len(r[0]) == 3 and r[0][0][0] != "w05"
    rb = client.post("/synonyms",
                     json={"query": ["w05", "w09"], "num": 3}).json()
    assert len(rb) == 2
    assert [w for w, _ in rb[0]] == [w for w, _ in r[0]]
    t = client.post("/transform",
                    json={"sentences": [["w01", "w02"], ["nope"]]}).json()
    np.testing.assert_allclose(t["vectors"][0], syn0[[1, 2]].mean(0),
                               rtol=1e-5, atol=1e-6)
    a = client.post("/analogy",
                    json={"pos": ["w01", "w02"], "neg": ["w03"],
                          "num": 4}).json()
    assert len(a) == 4
