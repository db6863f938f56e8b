#!/usr/bin/env python3
"""Quality-at-scale probe: planted-synonym retrieval accuracy.

Corpus: a Zipf stream over V/2 *concepts*; every concept occurrence is
randomly emitted as word 2c or 2c+1, so the two words of each pair are
perfectly interchangeable.  After training, word 2c's nearest syn0-cosine
neighbour should be 2c+1.  Accuracy over mid-frequency concepts is a
scale-appropriate embedding-quality score (the German-corpus gates cover
small-scale; this covers the benchmark regime).

Usage:
  python benchmarks/quality_probe.py --vocab 200000 --dim 128 \
      --words 50000000 --mode hogwild|atomic [--device cuda|cpu]
"""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def planted_corpus(vocab, num_tokens, sentence_len, seed, zipf_a=1.05,
                   locality=5):
    """Concept-line corpus: each sentence is anchored at a Zipf-drawn
    concept and draws its words from the anchor's +-locality neighbourhood,
    so every concept has a distinct context distribution (its neighbours)
    while the two words of a pair share it exactly."""
    rng = np.random.default_rng(seed)
    concepts = vocab // 2
    ranks = np.arange(1, concepts + 1, dtype=np.float64)
    w = ranks ** (-zipf_a)
    cdf = np.cumsum(w)
    cdf /= cdf[-1]
    n_sent = max(1, num_tokens // sentence_len)
    anchors = np.searchsorted(cdf, rng.random(n_sent)).astype(np.int64)
    jitter = rng.integers(-locality, locality + 1,
                          size=(n_sent, sentence_len))
    concept = np.clip(anchors[:, None] + jitter, 0, concepts - 1)
    bit = rng.integers(0, 2, size=(n_sent, sentence_len))
    tokens = (2 * concept + bit).astype(np.int32).ravel()
    offsets = np.arange(0, n_sent + 1, dtype=np.int64) * sentence_len
    return tokens, offsets.astype(np.int32)


def nn_accuracy(syn0, eval_concepts, block=512):
    """Top-1 neighbour accuracy: is 2c+1 the NN of 2c (excluding itself)?"""
    f = syn0 / (np.linalg.norm(syn0, axis=1, keepdims=True) + 1e-12)
    hits = 0
    ids = np.asarray([2 * c for c in eval_concepts])
    for s in range(0, len(ids), block):
        q = f[ids[s:s + block]]
        cos = q @ f.T
        for row, wid in enumerate(ids[s:s + block]):
            cos[row, wid] = -2
        nn = np.argmax(cos, axis=1)
        hits += int(np.sum(nn == ids[s:s + block] + 1))
    return hits / len(ids)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--vocab", type=int, default=200_000)
    ap.add_argument("--dim", type=int, default=128)
    ap.add_argument("--words", type=int, default=50_000_000)
    ap.add_argument("--sentence-len", type=int, default=100)
    ap.add_argument("--mode",
                    choices=["hogwild", "atomic", "hybrid", "positives"],
                    default="hogwild")
    ap.add_argument("--shared-negatives", action="store_true")
    ap.add_argument("--hot-floor", type=int, default=16,
                    help="hybrid: rows < F stay hogwild (contention escape)")
    ap.add_argument("--hot-rows", type=int, default=32768,
                    help="hybrid: atomics for rows < K (word id ~ 2*Zipf "
                         "rank in this corpus, so K covers the K/2 hottest "
                         "concepts)")
    ap.add_argument("--device", choices=["cuda", "cpu"], default="cuda")
    ap.add_argument("--lr", type=float, default=0.025)
    ap.add_argument("--neg", type=int, default=5)
    ap.add_argument("--window", type=int, default=5)
    ap.add_argument("--subsample", type=float, default=1e-4)
    ap.add_argument("--eval-lo", type=int, default=100,
                    help="eval concepts: frequency ranks [lo, hi)")
    ap.add_argument("--eval-hi", type=int, default=2100)
    args = ap.parse_args()

    from glint_word2vec_amd.vocab import build_unigram_table

    tokens, offsets = planted_corpus(args.vocab, args.words,
                                     args.sentence_len, seed=7)
    counts = np.bincount(tokens, minlength=args.vocab).astype(np.int64) + 1
    t0 = time.time()
    if args.device == "cuda":
        import torch
        from glint_word2vec_amd.ops.gpu import GpuSgns
        gs = GpuSgns(args.vocab, args.dim, dtype="bfloat16", device="cuda",
                     seed=3)
        gs.set_table(build_unigram_table(counts, 50_000_000))
        if args.subsample > 0:
            gs.set_subsample(counts, int(counts.sum()), args.subsample)
        tok = torch.from_numpy(tokens).cuda()
        off = torch.from_numpy(offsets).cuda()
        abelow = {"hogwild": None, "atomic": None, "positives": -1,
                  "hybrid": args.hot_rows}[args.mode]
        gs.train_batch(tok, off, args.lr, args.window, args.neg, 42,
                       atomic=(args.mode != "hogwild"), atomic_below=abelow,
                       atomic_floor=(args.hot_floor
                                     if args.mode == "hybrid" else 0),
                       shared_negatives=args.shared_negatives)
        torch.cuda.synchronize()
        st = gs.read_stats()
        syn0, _ = gs.to_host()
    else:
        from glint_word2vec_amd import _cpu_native
        from glint_word2vec_amd.models import sgns
        syn0, syn1 = sgns.init_tables(args.vocab, args.dim, 3)
        table = build_unigram_table(counts, 10_000_000)
        st = _cpu_native.train_batch(syn0, syn1, tokens, offsets, None, table,
                                     args.lr, args.window, args.neg, 42, 0,
                                     "canonical", os.cpu_count() or 8)
        st = type("S", (), st)
    dt = time.time() - t0
    # eval on mid-frequency concepts (Zipf rank = concept id)
    acc = nn_accuracy(syn0, range(args.eval_lo, args.eval_hi))
    mode = (f"hybrid[{args.hot_floor}..{args.hot_rows}]"
            if args.mode == "hybrid" else args.mode)
    if args.shared_negatives:
        mode += "+sharedneg"
    print(f"mode={mode} device={args.device} vocab={args.vocab} "
          f"words={args.words}: planted-NN acc={acc:.3f} "
          f"({args.words / dt / 1e6:.1f}M words/s incl. setup)")


if __name__ == "__main__":
    main()
