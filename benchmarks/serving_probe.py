"""Serving micro-benchmark: model-op throughput on one MI355X at the
headline shape (vocab 1M, dim 300) — the reference's server-side model
ops (pullAverage / multiply+norms behind findSynonyms).

Run: python benchmarks/serving_probe.py
"""
from __future__ import annotations

import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from glint_word2vec_amd.ops.gpu import GpuSgns  # noqa: E402


def main():
    vocab, dim = 1_000_000, 300
    gs = GpuSgns(vocab, dim, dtype="bfloat16", device="cuda", seed=1)

    # --- sentence-average transform (pullAverage): batches of 10k
    # sentences x 20 words, mirroring the reference's 10k-sentence batch
    # cap (ml:449)
    rng = np.random.default_rng(0)
    n_sent, sent_len = 10_000, 20
    tokens = torch.from_numpy(
        rng.integers(0, vocab, n_sent * sent_len).astype(np.int32)).cuda()
    offsets = torch.from_numpy(
        np.arange(0, n_sent * sent_len + 1, sent_len,
                  dtype=np.int32)).cuda()
    for _ in range(3):
        gs.pull_average(tokens, offsets)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 50
    for _ in range(reps):
        gs.pull_average(tokens, offsets)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"pullAverage: {reps * n_sent / dt / 1e6:.2f}M sentences/s "
          f"({dt / reps * 1e3:.2f} ms per 10k-sentence batch)")

    # --- findSynonyms core: norms (cached) + GEMV scores over all rows
    norms = gs.norms()
    torch.cuda.synchronize()
    q = torch.randn(dim).cuda()
    for _ in range(3):
        scores = gs.multiply(q) / norms.clamp_min(1e-12)
        torch.topk(scores, 10)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 200
    for _ in range(reps):
        scores = gs.multiply(q) / norms.clamp_min(1e-12)
        torch.topk(scores, 10)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"findSynonyms (GEMV over 1M rows + top-10): "
          f"{reps / dt:.0f} queries/s ({dt / reps * 1e3:.3f} ms/query)")



    # --- graph-replayed single query (hipGraph: one launch per query)
    qv = torch.randn(dim).cuda()
    gs.synonyms_query(qv, 11)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 2000
    for _ in range(reps):
        gs.synonyms_query(qv, 11)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"findSynonyms graphed single query: {reps / dt:.0f} queries/s "
          f"({dt / reps * 1e3:.3f} ms/query)")

    # --- batched findSynonyms: one GEMM over Q queries + one topk
    norms_r = norms.clamp_min(1e-12)[None, :]
    for Q in (256, 1024, 4096):
        qb = torch.randn(Q, dim)
        for _ in range(3):
            cos = gs.multiply_batch(qb) / norms_r
            torch.topk(cos, 10, dim=1)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 20
        for _ in range(reps):
            cos = gs.multiply_batch(qb) / norms_r
            torch.topk(cos, 10, dim=1)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"findSynonyms batched Q={Q}: "
              f"{reps * Q / dt / 1e3:.1f}k queries/s "
              f"({dt / reps * 1e3:.2f} ms per batch)")


if __name__ == "__main__":
    main()
