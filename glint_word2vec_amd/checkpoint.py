"""Checkpoint layout — compatible with the reference's structure
(SURVEY.md §3.4): ``path/metadata`` (params JSON), ``path/words`` (one word
per line, line number == row index), plus binary shard files of both
matrices (one per shard, concatenable by row interleave).

Reference: save at mllib:493-498 / ml:499-560, load at mllib:683-725.
"""
from __future__ import annotations

import json
import os
import time
from typing import List, Tuple

import numpy as np

from .config import Word2VecConfig
from .vocab import Vocabulary

_DTYPES = {"float32": np.float32, "bfloat16": np.uint16}  # bf16 stored as raw u16


def _shard_rows(vocab_size: int, shard: int, num_shards: int) -> np.ndarray:
    """Row r lives in shard r % num_shards (round-robin row interleave —
    balances Zipf-hot rows across shards/GPUs)."""
    return np.arange(shard, vocab_size, num_shards, dtype=np.int64)


def save_model(path: str, config: Word2VecConfig, vocab: Vocabulary,
               syn0: np.ndarray, syn1: np.ndarray | None = None,
               num_shards: int = 1) -> None:
    os.makedirs(path, exist_ok=True)
    os.makedirs(os.path.join(path, "shards"), exist_ok=True)
    meta = {
        "class": "glint_word2vec_amd.GlintWord2VecModel",
        "timestamp": int(time.time() * 1000),
        "numWords": vocab.num_words,
        "vectorSize": int(syn0.shape[1]),
        "paramMap": config.to_dict(),
    }
    with open(os.path.join(path, "metadata"), "w") as f:
        json.dump(meta, f, indent=2, sort_keys=True)
    vocab.save_words(os.path.join(path, "words"))
    if vocab.counts is not None:
        np.save(os.path.join(path, "counts.npy"), vocab.counts)
    dt = syn0.dtype
    index = {
        "num_shards": num_shards,
        "vocab": int(syn0.shape[0]),
        "dim": int(syn0.shape[1]),
        "dtype": str(dt),
        "layout": "row_mod",
        "has_syn1": syn1 is not None,
    }
    with open(os.path.join(path, "shards", "index.json"), "w") as f:
        json.dump(index, f, indent=2)
    for s in range(num_shards):
        rows = _shard_rows(syn0.shape[0], s, num_shards)
        syn0[rows].tofile(os.path.join(path, "shards", f"syn0-{s:05d}.bin"))
        if syn1 is not None:
            syn1[rows].tofile(os.path.join(path, "shards", f"syn1-{s:05d}.bin"))


def save_model_streaming(path: str, config: Word2VecConfig, vocab: Vocabulary,
                         row_block_fn, num_shards: int = 8,
                         block_rows: int = 1 << 20,
                         has_syn1: bool = True, write: bool = True) -> None:
    """Checkpoint without materialising the full matrices on host (80M-vocab
    models: 96 GB per table).  ``row_block_fn(which, r0, r1)`` returns the
    f32 [r1-r0, dim] block of syn0 (which=0) / syn1 (which=1).  Shards are
    contiguous row ranges ("row_range" layout; the reference's per-PS shard
    files are the analog, mllib:493-498).

    ``write=False`` performs the exact same sequence of ``row_block_fn``
    calls but no file IO: non-zero ranks of an engine whose row_block_fn is
    collective (dim-sharded allgather) call with write=False so every rank
    walks the identical block schedule."""
    if write:
        os.makedirs(os.path.join(path, "shards"), exist_ok=True)
    V = vocab.num_words
    bounds = [min(V, s * ((V + num_shards - 1) // num_shards))
              for s in range(num_shards + 1)]
    bounds[-1] = V
    probe = row_block_fn(0, 0, min(1, V))
    dim = probe.shape[1]
    if write:
        meta = {
            "class": "glint_word2vec_amd.GlintWord2VecModel",
            "timestamp": int(time.time() * 1000),
            "numWords": V,
            "vectorSize": dim,
            "paramMap": config.to_dict(),
        }
        with open(os.path.join(path, "metadata"), "w") as f:
            json.dump(meta, f, indent=2, sort_keys=True)
        vocab.save_words(os.path.join(path, "words"))
        if getattr(vocab, "counts", None) is not None:
            np.save(os.path.join(path, "counts.npy"), vocab.counts)
        index = {"num_shards": num_shards, "vocab": V, "dim": dim,
                 "dtype": "float32", "layout": "row_range", "bounds": bounds,
                 "has_syn1": has_syn1}
        with open(os.path.join(path, "shards", "index.json"), "w") as f:
            json.dump(index, f, indent=2)
    for which, name in ((0, "syn0"), (1, "syn1"))[:2 if has_syn1 else 1]:
        for s in range(num_shards):
            f = (open(os.path.join(path, "shards", f"{name}-{s:05d}.bin"),
                      "wb") if write else None)
            try:
                for r0 in range(bounds[s], bounds[s + 1], block_rows):
                    r1 = min(bounds[s + 1], r0 + block_rows)
                    block = np.ascontiguousarray(
                        row_block_fn(which, r0, r1), dtype=np.float32)
                    if f is not None:
                        f.write(block.tobytes())
            finally:
                if f is not None:
                    f.close()


def load_model(path: str) -> Tuple[Word2VecConfig, Vocabulary, np.ndarray, np.ndarray | None]:
    with open(os.path.join(path, "metadata")) as f:
        meta = json.load(f)
    config = Word2VecConfig.from_dict(meta["paramMap"])
    counts_p = os.path.join(path, "counts.npy")
    counts = np.load(counts_p) if os.path.exists(counts_p) else None
    vocab = Vocabulary.load_words(os.path.join(path, "words"), counts)
    with open(os.path.join(path, "shards", "index.json")) as f:
        index = json.load(f)
    V, D, k = index["vocab"], index["dim"], index["num_shards"]
    # "bfloat16" shards are raw u16 (half the disk of f32); decode to f32
    is_bf16 = index["dtype"] == "bfloat16"
    dt = np.uint16 if is_bf16 else np.dtype(index["dtype"])
    out_dt = np.float32 if is_bf16 else dt
    layout = index.get("layout", "row_mod")
    syn0 = np.empty((V, D), dtype=out_dt)
    syn1 = np.empty((V, D), dtype=out_dt) if index.get("has_syn1") else None

    def _decode(buf, rows):
        block = buf.reshape(len(rows), D)
        if is_bf16:
            import torch
            block = torch.from_numpy(block).view(torch.bfloat16) \
                .float().numpy()
        return block

    for s in range(k):
        if layout == "row_mod":
            rows = _shard_rows(V, s, k)
        else:  # row_range
            b = index["bounds"]
            rows = np.arange(b[s], b[s + 1], dtype=np.int64)
        buf = np.fromfile(os.path.join(path, "shards", f"syn0-{s:05d}.bin"), dtype=dt)
        syn0[rows] = _decode(buf, rows)
        if syn1 is not None:
            buf = np.fromfile(os.path.join(path, "shards", f"syn1-{s:05d}.bin"), dtype=dt)
            syn1[rows] = _decode(buf, rows)
    return config, vocab, syn0, syn1


def save_word2vec_text(path: str, words: List[str], vectors: np.ndarray) -> None:
    """Classic word2vec/gensim text format (the `toLocal` export analog,
    mllib:651-659)."""
    with open(path, "w", encoding="utf-8") as f:
        f.write(f"{len(words)} {vectors.shape[1]}\n")
        for w, v in zip(words, vectors):
            f.write(w + " " + " ".join(f"{x:.6g}" for x in v) + "\n")
