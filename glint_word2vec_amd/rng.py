"""Deterministic RNG shared bit-for-bit between the CPU oracle, the C++
native trainer (csrc/cpu_sgns.cpp) and the HIP kernel.

The fused kernel draws all randomness (subsample keep tests, window shrink,
negative-table indices) from a per-sentence xorshift64* stream seeded by
splitmix64(seed ^ (sentence_id * GOLDEN)).  The oracle implements the
identical sequence so a serial GPU launch can be compared element-wise
against the Python reference (the reference repo has no such test —
SURVEY.md §4 "rebuild implication": kernel unit tests vs an oracle).

Draw order per sentence (normative — every implementation must match):
  1. one u32 per token, in order, for the subsample keep test
     (u32 < keep_prob * 2^32); skipped entirely when subsampling is off.
  2. one u32 per *kept* position, in compacted order, for the window draw.
  3. per (position, context) pair in ascending context order, ``n`` u32 draws
     for negative-table indices; a draw whose table entry equals the
     positive target is discarded (that negative slot is skipped, as in
     canonical word2vec.c).
"""
from __future__ import annotations

_M64 = (1 << 64) - 1
_GOLDEN = 0x9E3779B97F4A7C15


def splitmix64(x: int) -> int:
    z = (x + _GOLDEN) & _M64
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & _M64
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & _M64
    return z ^ (z >> 31)


class XorShift64Star:
    """xorshift64* — 3 shifts + multiply; trivially implementable per-lane in
    the kernel.  State must be nonzero."""

    __slots__ = ("state",)

    def __init__(self, seed: int, sentence_id: int):
        s = splitmix64((seed ^ (sentence_id * _GOLDEN)) & _M64)
        self.state = s if s != 0 else 1

    def next_u64(self) -> int:
        x = self.state
        x ^= x >> 12
        x = (x ^ (x << 25)) & _M64
        x ^= x >> 27
        self.state = x
        return (x * 0x2545F4914F6CDD1D) & _M64

    def next_u32(self) -> int:
        return self.next_u64() >> 32
