#!/usr/bin/env python3
"""BASELINE.json config 1: the CPU plumbing benchmark (text8-shaped —
vocab~10k, dim=50, neg=5, window=5 — on synthetic data; there is no network
to fetch text8).  Measures the native C++ hogwild trainer."""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--vocab", type=int, default=10_000)
    ap.add_argument("--dim", type=int, default=50)
    ap.add_argument("--neg", type=int, default=5)
    ap.add_argument("--window", type=int, default=5)
    ap.add_argument("--words", type=int, default=2_000_000)
    ap.add_argument("--threads", type=int, default=os.cpu_count() or 8)
    args = ap.parse_args()

    from glint_word2vec_amd import _cpu_native as nat
    from glint_word2vec_amd.data import synthetic_corpus
    from glint_word2vec_amd.models import sgns
    from glint_word2vec_amd.vocab import build_unigram_table

    batch = synthetic_corpus(args.vocab, args.words, sentence_len=100, seed=7)
    counts = np.bincount(batch.tokens, minlength=args.vocab).astype(np.int64) + 1
    table = build_unigram_table(counts, 10_000_000)
    syn0, syn1 = sgns.init_tables(args.vocab, args.dim, 1)
    # warmup
    nat.train_batch(syn0, syn1, batch.tokens[:5000],
                    np.array([0, 5000], dtype=np.int32), None, table, 0.025,
                    args.window, args.neg, 1, 0, "canonical", args.threads)
    t0 = time.time()
    st = nat.train_batch(syn0, syn1, batch.tokens, batch.offsets, None, table,
                         0.025, args.window, args.neg, 1, 0, "canonical",
                         args.threads)
    dt = time.time() - t0
    wps = args.words / dt
    print(f"CPU config-1 bench: vocab={args.vocab} dim={args.dim} "
          f"neg={args.neg} threads={args.threads}: "
          f"{wps/1e6:.2f}M words/s ({st['pairs']} pairs, "
          f"mean_fplus={st['sum_fplus']/max(st['positives'],1):.4f})")


if __name__ == "__main__":
    main()
