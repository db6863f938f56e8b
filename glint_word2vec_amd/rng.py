"""Deterministic counter-based RNG shared bit-for-bit between the CPU
oracle, the C++ native trainer (csrc/cpu_sgns.cpp) and the HIP kernel
(csrc/hip/sgns_hip.hip).

All randomness is drawn from a per-sentence splitmix64 stream addressed by a
*draw index* — counter-based, so the GPU can evaluate any draw at any lane
without sequential state:

    base    = splitmix64(seed ^ (sentence_id * GOLDEN))
    draw(k) = high 32 bits of splitmix64(base + k * GOLDEN)

Draw-index layout per sentence (normative — every implementation must match):
  * subsample test for original position p:       k = p
    keep iff u <= thr[w], thr[w] = min(floor(keep_prob[w] * 2^32), 2^32-1);
    no draws at all when subsampling is disabled.
  * window draw for kept (compacted) position i:  k = WIN_BASE + i
  * negative slot s of pair (center i, context j):
        k = NEG_BASE + (i * (2*window+1) + (j - i + window)) * n + s
    a draw whose table entry equals the positive target is discarded (that
    negative slot is skipped, as in canonical word2vec.c).
  * SHARED-NEGATIVES mode (config.shared_negatives=True, opt-in — the
    HogBatch-style trick that reuses one negative set across all contexts
    of a position, cutting target-row traffic per position from
    ~2b*(1+n) to ~2b+n rows):
        k = NEG_BASE + i * n + s          (independent of j)
    The same collision-discard rule applies per context (the shared draw
    is discarded only against that context's positive target), so pair
    counts can differ between contexts.

Sentence ids must stay < 2^63.  The three draw-index streams never
collide for any window/negative count: subsample uses k < 1024, window
draws use [WIN_BASE, WIN_BASE + 1024), negatives use [NEG_BASE, inf) and
NEG_BASE > WIN_BASE + 1024.
"""
from __future__ import annotations

_M64 = (1 << 64) - 1
GOLDEN = 0x9E3779B97F4A7C15
WIN_BASE = 1 << 20
NEG_BASE = 1 << 21


def splitmix64(x: int) -> int:
    z = (x + GOLDEN) & _M64
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & _M64
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & _M64
    return z ^ (z >> 31)


def sentence_base(seed: int, sentence_id: int) -> int:
    return splitmix64((seed ^ (sentence_id * GOLDEN)) & _M64)


def draw_u32(base: int, k: int) -> int:
    """The k-th u32 of the sentence stream."""
    return splitmix64((base + k * GOLDEN) & _M64) >> 32


def keep_threshold(keep_prob: float) -> int:
    return min(int(keep_prob * 4294967296.0), (1 << 32) - 1)
