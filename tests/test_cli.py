"""CLI end-to-end: train -> similar -> analogy -> export -> info."""
import subprocess
import sys

import pytest


def _run(args, **kw):
    return subprocess.run([sys.executable, "-m", "glint_word2vec_amd"] + args,
                          capture_output=True, text=True, timeout=300, **kw)


def test_cli_roundtrip(tmp_path):
    corpus = tmp_path / "corpus.txt"
    lines = []
    import numpy as np
    rng = np.random.default_rng(3)
    for _ in range(300):
        head = "aa" if rng.random() < 0.5 else "a2"
        lines.append(" ".join([head, "bb"] +
                              [f"x{rng.integers(0, 15)}" for _ in range(3)] +
                              [head, "bb"]))
    corpus.write_text("\n".join(lines))
    model_dir = str(tmp_path / "model")
    r = _run(["train", str(corpus), model_dir, "--vector-size", "16",
              "--min-count", "1", "--iterations", "4", "--window", "2",
              "--learning-rate", "0.05", "--unigram-table-size", "50000",
              "--subsample", "0", "--seed", "7", "--device", "cpu",
              "--workers", "2", "--shards", "2"])
    assert r.returncode == 0, r.stderr
    assert "model saved" in r.stdout

    r = _run(["similar", model_dir, "aa", "-n", "3"])
    assert r.returncode == 0, r.stderr
    assert "a2" in r.stdout

    r = _run(["analogy", model_dir, "aa", "bb", "-m", "a2", "-n", "3"])
    assert r.returncode == 0, r.stderr

    out = tmp_path / "vecs.txt"
    r = _run(["export", model_dir, str(out)])
    assert r.returncode == 0, r.stderr
    head = out.read_text().splitlines()[0].split()
    assert int(head[1]) == 16

    r = _run(["info", model_dir])
    assert r.returncode == 0
    assert "vectorSize" in r.stdout


def test_cli_resume_and_mid_checkpoints(tmp_path):
    corpus = tmp_path / "c.txt"
    import numpy as np
    rng = np.random.default_rng(5)
    corpus.write_text("\n".join(
        " ".join(f"w{rng.integers(0, 10)}" for _ in range(15))
        for _ in range(40)))
    m1 = str(tmp_path / "m1")
    r = _run(["train", str(corpus), m1, "--vector-size", "8",
              "--min-count", "1", "--iterations", "1", "--window", "2",
              "--subsample", "0", "--seed", "3", "--device", "cpu",
              "--unigram-table-size", "10000"])
    assert r.returncode == 0, r.stderr
    m2 = str(tmp_path / "m2")
    r = _run(["train", str(corpus), m2, "--vector-size", "8",
              "--min-count", "1", "--iterations", "1", "--window", "2",
              "--subsample", "0", "--seed", "3", "--device", "cpu",
              "--unigram-table-size", "10000", "--resume-from", m1])
    assert r.returncode == 0, r.stderr
    r = _run(["info", m2])
    assert r.returncode == 0


def test_cli_sharded_similar_and_export(tmp_path):
    import numpy as np
    from glint_word2vec_amd.__main__ import main
    from glint_word2vec_amd.checkpoint import save_model
    from glint_word2vec_amd.config import Word2VecConfig
    from glint_word2vec_amd.vocab import Vocabulary
    rng = np.random.default_rng(1)
    words = [f"w{i}" for i in range(30)]
    voc = Vocabulary(words=words, counts=np.ones(30, dtype=np.int64),
                     index={w: i for i, w in enumerate(words)},
                     train_words_count=30)
    syn0 = rng.standard_normal((30, 8)).astype(np.float32)
    path = str(tmp_path / "m")
    save_model(path, Word2VecConfig(vector_size=8), voc, syn0, num_shards=2)
    assert main(["similar", path, "w3", "-n", "3", "--sharded"]) == 0
    out = str(tmp_path / "v.txt")
    assert main(["export", path, out, "--sharded"]) == 0
    lines = open(out).read().splitlines()
    assert lines[0] == "30 8" and len(lines) == 31
