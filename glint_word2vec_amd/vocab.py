"""Vocabulary construction.

Semantics follow the reference's ``learnVocab`` (mllib
ServerSideGlintWord2Vec.scala:258-279): count words over the corpus, drop
words below ``min_count``, sort descending by count (ties broken by insertion
order of the sort, stable on word for determinism), assign row index =
position in the sorted order.  The word -> index map plays the role of the
reference's broadcast vocabHash (mllib:269-275); here it is a plain host dict
(no 8 GB Spark-broadcast ceiling, cf. README.md:71-73 of the reference).
"""
from __future__ import annotations

from collections import Counter
from dataclasses import dataclass
from typing import Iterable, Iterator, List, Sequence

import numpy as np


@dataclass
class Vocabulary:
    words: List[str]          # row index -> word, sorted by count desc
    counts: np.ndarray        # int64 [num_words], counts aligned with words
    index: dict               # word -> row index
    train_words_count: int    # total count of retained words over the corpus

    @property
    def num_words(self) -> int:
        return len(self.words)

    def __contains__(self, word: str) -> bool:
        return word in self.index

    def __getitem__(self, word: str) -> int:
        return self.index[word]

    def get(self, word: str, default: int = -1) -> int:
        return self.index.get(word, default)

    # --- persistence (checkpoint "words" file; one word per line, line
    #     number == row index — mllib:493-498, 714-715) -------------------
    def save_words(self, path: str) -> None:
        with open(path, "w", encoding="utf-8") as f:
            for w in self.words:
                f.write(w)
                f.write("\n")

    @classmethod
    def load_words(cls, path: str, counts: np.ndarray | None = None) -> "Vocabulary":
        with open(path, "r", encoding="utf-8") as f:
            words = [line.rstrip("\n") for line in f]
        if words and words[-1] == "":
            words.pop()
        if counts is None:
            counts = np.zeros(len(words), dtype=np.int64)
        index = {w: i for i, w in enumerate(words)}
        return cls(words=words, counts=counts, index=index,
                   train_words_count=int(counts.sum()))


def build_vocab(sentences: Iterable[Sequence[str]], min_count: int = 5) -> Vocabulary:
    """Count -> filter(min_count) -> sort desc by count (stable by word for
    determinism) -> hash.  Reference: mllib:258-279."""
    counter: Counter = Counter()
    for sent in sentences:
        counter.update(sent)
    items = [(w, c) for w, c in counter.items() if c >= min_count]
    # Descending count; secondary key = word, so order is deterministic
    # regardless of hash-iteration order.
    items.sort(key=lambda wc: (-wc[1], wc[0]))
    words = [w for w, _ in items]
    counts = np.asarray([c for _, c in items], dtype=np.int64)
    index = {w: i for i, w in enumerate(words)}
    return Vocabulary(words=words, counts=counts, index=index,
                      train_words_count=int(counts.sum()))


def keep_probabilities(counts: np.ndarray, train_words_count: int,
                       subsample_ratio: float, legacy: bool = False) -> np.ndarray:
    """Per-word keep probability for frequency subsampling.

    Intended math of the reference (mllib:371-379, with the B1 integer-
    division bug fixed — SURVEY.md §3.6): with word frequency p = count/total
    and ratio r, keep probability = (sqrt(p/r) + 1) * (r/p), clipped to 1.
    ``legacy=True`` reproduces the reference's de-facto behaviour (keep all).
    """
    if legacy or subsample_ratio <= 0 or train_words_count <= 0:
        return np.ones(len(counts), dtype=np.float32)
    p = counts.astype(np.float64) / float(train_words_count)
    with np.errstate(divide="ignore", invalid="ignore"):
        kp = (np.sqrt(p / subsample_ratio) + 1.0) * (subsample_ratio / p)
    kp = np.where(p > 0, kp, 1.0)
    return np.minimum(kp, 1.0).astype(np.float32)


def build_unigram_table(counts: np.ndarray, table_size: int,
                        power: float = 0.75) -> np.ndarray:
    """Classic word2vec negative-sampling table: word i occupies a slice of
    the table proportional to count_i^power.  The reference builds this
    server-side from the broadcast counts (Word2VecArguments.unigramTableSize,
    mllib:85,351); here it is built once on host and uploaded per GPU/shard.

    Returns int32 [table_size].
    """
    n = len(counts)
    if n == 0:
        return np.zeros(0, dtype=np.int32)
    weights = counts.astype(np.float64) ** power
    cum = np.cumsum(weights)
    total = cum[-1]
    # Boundary positions: word i covers table slots [cum[i-1], cum[i]) / total.
    bounds = np.floor(cum / total * table_size).astype(np.int64)
    table = np.zeros(table_size, dtype=np.int32)
    prev = 0
    for i in range(n):
        hi = min(int(bounds[i]), table_size)
        if hi > prev:
            table[prev:hi] = i
        prev = hi
    if prev < table_size:
        table[prev:] = n - 1
    return table


def encode_sentences(sentences: Iterable[Sequence[str]], vocab: Vocabulary,
                     max_sentence_length: int = 1000) -> Iterator[np.ndarray]:
    """words -> row indices, dropping OOV, chunking at max_sentence_length.
    Reference: mllib:335-343."""
    for sent in sentences:
        idx = [vocab.index[w] for w in sent if w in vocab.index]
        for start in range(0, len(idx), max_sentence_length):
            chunk = idx[start:start + max_sentence_length]
            if chunk:
                yield np.asarray(chunk, dtype=np.int32)
