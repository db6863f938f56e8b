"""Exact-semantics CPU oracle for the fused SGNS kernel.

This is the single source of truth for the training math.  It is slow
(pure-Python loops) and exists for correctness: the HIP kernel launched with
a serial grid must reproduce it element-wise (same RNG draws — see rng.py).

Math per positive pair (center c, target t, label):
  f      = dot(syn0_row_c_cached, syn1[t])            # rows read pre-update
  sigma  = 1 if f > MAX_EXP else 0 if f < -MAX_EXP else 1/(1+exp(-f))
  g      = (label - sigma) * alpha
  grad_c += g * syn1[t]        # with the *pre-update* syn1 row
  syn1[t] += g * syn0_row_c_cached
  ... after all targets of the position: syn0[c] += grad_c

This mirrors the server-side dotprod/adjust contract of the reference
(mllib:419-429 client math at 423-424; server ops per SURVEY.md §2.2) and
canonical word2vec: the center row is cached at position start, updates land
after the window is processed.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import numpy as np

from ..config import MAX_EXP
from ..rng import NEG_BASE, WIN_BASE, draw_u32, keep_threshold, sentence_base


def sigmoid_clipped(f: float) -> float:
    if f > MAX_EXP:
        return 1.0
    if f < -MAX_EXP:
        return 0.0
    return 1.0 / (1.0 + math.exp(-f))


def sigmoid_lut(f: float, table: "np.ndarray") -> float:
    """Reference getSigmoid semantics (mllib:292-302): clip at +-MAX_EXP,
    else floor-indexed table lookup.  Index math in float32 to match the
    C++/HIP implementations bit-for-bit."""
    if f > MAX_EXP:
        return 1.0
    if f < -MAX_EXP:
        return 0.0
    scale = np.float32(len(table)) / np.float32(2.0 * MAX_EXP)
    ind = int(np.float32(np.float32(f) + np.float32(MAX_EXP)) * scale)
    if ind >= len(table):
        ind = len(table) - 1
    if ind < 0:
        ind = 0
    return float(table[ind])


@dataclass
class TrainStats:
    pairs: int = 0
    positives: int = 0
    sum_fplus: float = 0.0
    words_trained: int = 0   # kept positions actually processed


def train_batch_oracle(syn0: np.ndarray, syn1: np.ndarray,
                       tokens: np.ndarray, offsets: np.ndarray,
                       keep_prob: np.ndarray | None,
                       table: np.ndarray,
                       alpha: float, window: int, n_neg: int,
                       seed: int, sent_id_base: int = 0,
                       window_mode: str = "canonical",
                       exp_table: np.ndarray | None = None,
                       shared_negatives: bool = False) -> TrainStats:
    """In-place SGNS update over one batch.  syn0/syn1 float32 [vocab, dim].

    ``keep_prob`` None => subsampling off (no RNG draws for it).
    ``sent_id_base`` + local sentence index seeds each sentence's RNG stream.
    """
    stats = TrainStats()
    sig = (sigmoid_clipped if exp_table is None
           else (lambda f: sigmoid_lut(f, exp_table)))
    table_size = len(table)
    num_sentences = len(offsets) - 1
    do_subsample = keep_prob is not None
    for s in range(num_sentences):
        sent = tokens[offsets[s]:offsets[s + 1]]
        base = sentence_base(seed, sent_id_base + s)
        # 1. subsample
        if do_subsample:
            kept = []
            for p, w in enumerate(sent):
                u = draw_u32(base, p)
                if u <= keep_threshold(float(keep_prob[w])):
                    kept.append(int(w))
        else:
            kept = [int(w) for w in sent]
        L = len(kept)
        # 2./3. windows and training
        for i in range(L):
            c = kept[i]
            u = draw_u32(base, WIN_BASE + i)
            if window_mode == "canonical":
                b = 1 + (u % window)              # symmetric +-b
                lo, hi = max(0, i - b), min(L - 1, i + b)
            else:  # "reference": B2 semantics (mllib:385-387)
                b = u % window                    # left b, right b-1
                lo, hi = max(0, i - b), min(L - 1, i + b - 1)
                if b == 0:
                    lo, hi = i, i                 # empty context
            ctx = [j for j in range(lo, hi + 1) if j != i]
            if not ctx:
                continue
            c_row = syn0[c].copy()
            grad = np.zeros_like(c_row)
            if shared_negatives:
                # one negative set per POSITION (rng.py shared layout):
                # positives for every context first, then n_neg negatives
                # applied once, discarded when the draw equals the center
                for j in ctx:
                    t = kept[j]
                    f = float(np.dot(c_row, syn1[t]))
                    g = (1.0 - sig(f)) * alpha
                    grad += g * syn1[t]
                    syn1[t] += g * c_row
                    stats.pairs += 1
                    stats.positives += 1
                    stats.sum_fplus += f
                for k in range(n_neg):
                    u = draw_u32(base, NEG_BASE + i * n_neg + k)
                    neg = int(table[u % table_size])
                    if neg == c:
                        continue
                    f = float(np.dot(c_row, syn1[neg]))
                    g = (0.0 - sig(f)) * alpha
                    grad += g * syn1[neg]
                    syn1[neg] += g * c_row
                    stats.pairs += 1
                syn0[c] += grad
                stats.words_trained += 1
                continue
            for j in ctx:
                t = kept[j]
                # positive
                f = float(np.dot(c_row, syn1[t]))
                g = (1.0 - sig(f)) * alpha
                grad += g * syn1[t]
                syn1[t] += g * c_row
                stats.pairs += 1
                stats.positives += 1
                stats.sum_fplus += f
                # negatives
                kbase = NEG_BASE + (i * (2 * window + 1)
                                    + (j - i + window)) * n_neg
                for k in range(n_neg):
                    u = draw_u32(base, kbase + k)
                    neg = int(table[u % table_size])
                    if neg == t:
                        continue
                    fn = float(np.dot(c_row, syn1[neg]))
                    gn = (0.0 - sig(fn)) * alpha
                    grad += gn * syn1[neg]
                    syn1[neg] += gn * c_row
                    stats.pairs += 1
            syn0[c] += grad
            stats.words_trained += 1
    return stats


# ---------------------------------------------------------------------------
# Model-op oracles (reference op semantics per SURVEY.md §2.2 table)
# ---------------------------------------------------------------------------

def pull(syn0: np.ndarray, indices: np.ndarray) -> np.ndarray:
    """Row gather (Glint `pull`, mllib:514,539,639,652)."""
    return syn0[indices]


def pull_average(syn0: np.ndarray, sentences: list[np.ndarray]) -> np.ndarray:
    """Per-sentence mean of word vectors (Glint `pullAverage`, ml:453)."""
    dim = syn0.shape[1]
    out = np.zeros((len(sentences), dim), dtype=syn0.dtype)
    for i, s in enumerate(sentences):
        if len(s):
            out[i] = syn0[s].mean(axis=0)
    return out


def multiply(syn0: np.ndarray, vec: np.ndarray) -> np.ndarray:
    """Whole-table matrix-vector product (Glint `multiply`, mllib:598)."""
    return syn0 @ vec


def norms(syn0: np.ndarray) -> np.ndarray:
    """Euclidean norm of every row (Glint `norms`, mllib:486)."""
    return np.linalg.norm(syn0, axis=1)
