"""Corpus reading and step batching.

The reference streams RDD partitions of encoded sentences into mini-batches
(mllib:392-429).  Here the unit of work is a *step*: a packed flat buffer of
encoded sentences totalling ~``words_per_step`` tokens, which is what one
fused-kernel launch consumes.  Sentence boundaries travel as an offsets
array (CSR layout): tokens int32 [total], offsets int32 [num_sentences+1].
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Iterable, Iterator, List

import numpy as np


@dataclass
class SentenceBatch:
    """One training step's worth of encoded sentences (CSR)."""
    tokens: np.ndarray    # int32 [total_tokens]
    offsets: np.ndarray   # int32 [num_sentences + 1]

    @property
    def num_sentences(self) -> int:
        return len(self.offsets) - 1

    @property
    def num_tokens(self) -> int:
        return int(self.offsets[-1])

    def sentence(self, i: int) -> np.ndarray:
        return self.tokens[self.offsets[i]:self.offsets[i + 1]]


def read_text_corpus(path: str) -> Iterator[List[str]]:
    """One sentence per line, whitespace-tokenized.  (The reference takes
    already-tokenized Seq[String] rows from Spark; plain-text reading is the
    equivalent entry point here.)"""
    with open(path, "r", encoding="utf-8", errors="replace") as f:
        for line in f:
            toks = line.split()
            if toks:
                yield toks


def batch_sentences(encoded: Iterable[np.ndarray], words_per_step: int) -> Iterator[SentenceBatch]:
    """Pack encoded sentences into step-sized CSR batches."""
    tok_chunks: List[np.ndarray] = []
    offsets: List[int] = [0]
    total = 0
    for sent in encoded:
        tok_chunks.append(sent)
        total += len(sent)
        offsets.append(total)
        if total >= words_per_step:
            yield SentenceBatch(tokens=np.concatenate(tok_chunks).astype(np.int32),
                                offsets=np.asarray(offsets, dtype=np.int32))
            tok_chunks, offsets, total = [], [0], 0
    if tok_chunks:
        yield SentenceBatch(tokens=np.concatenate(tok_chunks).astype(np.int32),
                            offsets=np.asarray(offsets, dtype=np.int32))


def batches_from_arrays(tokens: np.ndarray, offsets: np.ndarray,
                        words_per_step: int) -> Iterator[SentenceBatch]:
    """Slice a pre-encoded CSR corpus (e.g. from the native encoder) into
    step-sized batches along sentence boundaries."""
    num_sent = len(offsets) - 1
    s0 = 0
    while s0 < num_sent:
        s1 = s0
        start_tok = offsets[s0]
        while s1 < num_sent and offsets[s1 + 1] - start_tok < words_per_step:
            s1 += 1
        s1 = min(max(s1 + 1, s0 + 1), num_sent)
        yield SentenceBatch(
            tokens=np.ascontiguousarray(tokens[offsets[s0]:offsets[s1]]),
            offsets=np.ascontiguousarray(offsets[s0:s1 + 1] - offsets[s0]))
        s0 = s1


def partition_round_robin(items: Iterable, rank: int, world: int) -> Iterator:
    """Deterministic corpus partitioning across ranks (the reference's
    repartition(numPartitions), mllib:345)."""
    return (x for i, x in enumerate(items) if i % world == rank)


def synthetic_corpus(vocab_size: int, num_tokens: int, sentence_len: int = 200,
                     seed: int = 1234, zipf_a: float = 1.05) -> SentenceBatch:
    """Synthetic Zipf-distributed token stream for benchmarking (no network,
    no datasets on the box — BASELINE.json configs use synthetic data).

    Token ids follow an approximate Zipf law so the unigram table and the
    hot-row access pattern (frequent words) are realistic.
    """
    rng = np.random.default_rng(seed)
    # Zipf via inverse-CDF on ranks 0..vocab_size-1 with weight (r+1)^-a.
    ranks = np.arange(1, vocab_size + 1, dtype=np.float64)
    w = ranks ** (-zipf_a)
    cdf = np.cumsum(w)
    cdf /= cdf[-1]
    u = rng.random(num_tokens)
    tokens = np.searchsorted(cdf, u).astype(np.int32)
    n_sent = max(1, num_tokens // sentence_len)
    bounds = np.linspace(0, num_tokens, n_sent + 1).astype(np.int32)
    return SentenceBatch(tokens=tokens, offsets=bounds)
