"""SGNS model: initialisation, vectorized pair-plan generation, and a fast
vectorized CPU trainer.

Two training paths exist:
  * the fused HIP kernel (ops/gpu.py) — sampling + training in one kernel,
    exact-oracle-matched (ops/cpu_ref.py);
  * this module's vectorized path — generates an explicit pair plan
    (subsample -> window -> negatives) then applies mini-batched updates.
    It is the CPU production path (quality gates, plumbing config 1 of
    BASELINE.json) and the plan generator also feeds the sharded multi-GPU
    path, where remote rows must be known before the kernel runs.

The plan's RNG is numpy's (performance); exact-RNG parity with the kernel is
the oracle's job, not this path's.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import numpy as np

from ..config import MAX_EXP


def create_exp_table(size: int = 1000, max_exp: float = MAX_EXP) -> np.ndarray:
    """The reference's 1000-entry sigmoid lookup table (createExpTable,
    mllib:281-290): entry i = sigma((i/size*2-1)*max_exp)."""
    x = np.exp((np.arange(size, dtype=np.float64) / size * 2.0 - 1.0) * max_exp)
    return (x / (x + 1.0)).astype(np.float32)


def init_tables(vocab_size: int, dim: int, seed: int) -> Tuple[np.ndarray, np.ndarray]:
    """Canonical word2vec init: syn0 ~ U(-0.5/dim, 0.5/dim), syn1 = 0."""
    rng = np.random.default_rng(seed)
    syn0 = ((rng.random((vocab_size, dim), dtype=np.float32) - 0.5) / dim).astype(np.float32)
    syn1 = np.zeros((vocab_size, dim), dtype=np.float32)
    return syn0, syn1


@dataclass
class PairPlan:
    """Flat training work for one step: positives and negatives interleaved
    per positive (targets of one center are contiguous)."""
    center: np.ndarray   # int32 [P]
    target: np.ndarray   # int32 [P]
    label: np.ndarray    # float32 [P] (1.0 positive, 0.0 negative)

    @property
    def num_pairs(self) -> int:
        return len(self.center)


def subsample_batch(tokens: np.ndarray, offsets: np.ndarray,
                    keep_prob: np.ndarray | None,
                    rng: np.random.Generator) -> Tuple[np.ndarray, np.ndarray]:
    """Apply frequency subsampling, preserving sentence boundaries.
    Returns (kept_tokens, new_offsets)."""
    if keep_prob is None:
        return tokens, offsets
    u = rng.random(len(tokens))
    keep = u < keep_prob[tokens]
    kept = tokens[keep]
    # new offsets: count kept per sentence
    counts = np.add.reduceat(keep.astype(np.int64), offsets[:-1]) if len(tokens) else \
        np.zeros(len(offsets) - 1, dtype=np.int64)
    # reduceat quirk: empty sentences at the end still index; guard zero-len
    sent_len = np.diff(offsets)
    counts = np.where(sent_len > 0, counts, 0)
    new_offsets = np.zeros(len(offsets), dtype=np.int32)
    np.cumsum(counts, out=new_offsets[1:])
    return kept.astype(np.int32), new_offsets


def make_plan(tokens: np.ndarray, offsets: np.ndarray,
              keep_prob: np.ndarray | None, table: np.ndarray,
              window: int, n_neg: int, rng: np.random.Generator,
              window_mode: str = "canonical") -> PairPlan:
    """Vectorized subsample -> shrunk-window pair generation -> negative
    draws.  Negatives colliding with their positive target are dropped
    (matching the oracle's skip rule)."""
    toks, offs = subsample_batch(tokens, offsets, keep_prob, rng)
    T = len(toks)
    if T == 0:
        z = np.zeros(0, dtype=np.int32)
        return PairPlan(z, z, np.zeros(0, dtype=np.float32))
    # sentence id and in-sentence bounds per position
    sent_id = np.repeat(np.arange(len(offs) - 1, dtype=np.int64), np.diff(offs))
    sent_lo = offs[:-1].astype(np.int64)[sent_id]
    sent_hi = offs[1:].astype(np.int64)[sent_id]          # exclusive
    pos = np.arange(T, dtype=np.int64)
    if window_mode == "canonical":
        b = rng.integers(1, window + 1, size=T)
        left, right = b, b
    else:  # reference-mode: left b, right b-1, b in [0, window-1]
        b = rng.integers(0, window, size=T)
        left, right = b, np.maximum(b - 1, 0) * (b > 0)
    centers_l, targets_l = [], []
    for o in range(1, window + 1):
        m = (o <= left) & (pos - o >= sent_lo)
        centers_l.append(pos[m]); targets_l.append(pos[m] - o)
        m = (o <= right) & (pos + o < sent_hi)
        centers_l.append(pos[m]); targets_l.append(pos[m] + o)
    cpos = np.concatenate(centers_l) if centers_l else np.zeros(0, dtype=np.int64)
    tpos = np.concatenate(targets_l) if targets_l else np.zeros(0, dtype=np.int64)
    # keep targets of one center contiguous & deterministic: sort by (center, target)
    order = np.lexsort((tpos, cpos))
    cpos, tpos = cpos[order], tpos[order]
    pc = toks[cpos].astype(np.int32)     # positive centers
    pt = toks[tpos].astype(np.int32)     # positive targets
    P = len(pc)
    if n_neg > 0 and P > 0:
        nidx = rng.integers(0, len(table), size=(P, n_neg))
        negs = table[nidx].astype(np.int32)               # [P, n]
        valid = negs != pt[:, None]
        # interleave: for each positive, its negatives follow it
        ctr = np.repeat(pc, n_neg).reshape(P, n_neg)
        all_center = np.concatenate([pc[:, None], ctr], axis=1)      # [P, 1+n]
        all_target = np.concatenate([pt[:, None], negs], axis=1)
        all_label = np.concatenate(
            [np.ones((P, 1), np.float32), np.zeros((P, n_neg), np.float32)], axis=1)
        all_valid = np.concatenate([np.ones((P, 1), bool), valid], axis=1)
        flat = all_valid.ravel()
        return PairPlan(all_center.ravel()[flat].astype(np.int32),
                        all_target.ravel()[flat].astype(np.int32),
                        all_label.ravel()[flat])
    return PairPlan(pc, pt, np.ones(P, dtype=np.float32))


@dataclass
class GroupedPlan:
    """Pair plan grouped by center position (the row-sharded engine's work
    unit: one group = one center position's window + negatives, trained
    against a pulled row cache — cf. the Glint dotprod/adjust batch)."""
    group_center: np.ndarray    # int32 [G] center WORD ids
    group_offsets: np.ndarray   # int64 [G+1] into pair arrays
    pair_target: np.ndarray     # int32 [P] target WORD ids
    pair_label: np.ndarray      # float32 [P]

    @property
    def num_groups(self) -> int:
        return len(self.group_center)

    @property
    def num_pairs(self) -> int:
        return len(self.pair_target)


def make_grouped_plan(tokens: np.ndarray, offsets: np.ndarray,
                      keep_prob: np.ndarray | None, table: np.ndarray,
                      window: int, n_neg: int, rng: np.random.Generator,
                      window_mode: str = "canonical") -> GroupedPlan:
    """Like make_plan but with center-position group structure."""
    toks, offs = subsample_batch(tokens, offsets, keep_prob, rng)
    T = len(toks)
    empty = GroupedPlan(np.zeros(0, np.int32), np.zeros(1, np.int64),
                        np.zeros(0, np.int32), np.zeros(0, np.float32))
    if T == 0:
        return empty
    sent_id = np.repeat(np.arange(len(offs) - 1, dtype=np.int64), np.diff(offs))
    sent_lo = offs[:-1].astype(np.int64)[sent_id]
    sent_hi = offs[1:].astype(np.int64)[sent_id]
    pos = np.arange(T, dtype=np.int64)
    if window_mode == "canonical":
        b = rng.integers(1, window + 1, size=T)
        left, right = b, b
    else:
        b = rng.integers(0, window, size=T)
        left, right = b, np.maximum(b - 1, 0) * (b > 0)
    centers_l, targets_l = [], []
    for o in range(1, window + 1):
        m = (o <= left) & (pos - o >= sent_lo)
        centers_l.append(pos[m]); targets_l.append(pos[m] - o)
        m = (o <= right) & (pos + o < sent_hi)
        centers_l.append(pos[m]); targets_l.append(pos[m] + o)
    if not centers_l:
        return empty
    cpos = np.concatenate(centers_l)
    tpos = np.concatenate(targets_l)
    if len(cpos) == 0:
        return empty
    order = np.lexsort((tpos, cpos))
    cpos, tpos = cpos[order], tpos[order]
    pc, pt = toks[cpos], toks[tpos]
    P = len(pc)
    if n_neg > 0:
        negs = table[rng.integers(0, len(table), size=(P, n_neg))].astype(np.int32)
        valid = negs != pt[:, None]
        all_target = np.concatenate([pt[:, None], negs], axis=1)
        all_label = np.concatenate(
            [np.ones((P, 1), np.float32), np.zeros((P, n_neg), np.float32)], axis=1)
        all_valid = np.concatenate([np.ones((P, 1), bool), valid], axis=1)
        all_cpos = np.repeat(cpos, 1 + n_neg).reshape(P, 1 + n_neg)
        flat = all_valid.ravel()
        pair_target = all_target.ravel()[flat].astype(np.int32)
        pair_label = all_label.ravel()[flat]
        pair_cpos = all_cpos.ravel()[flat]
    else:
        pair_target, pair_label, pair_cpos = pt.astype(np.int32), \
            np.ones(P, np.float32), cpos
    # group boundaries: runs of equal center position
    change = np.nonzero(np.diff(pair_cpos))[0] + 1
    starts = np.concatenate([[0], change])
    group_offsets = np.concatenate([starts, [len(pair_cpos)]]).astype(np.int64)
    group_center = toks[pair_cpos[starts]].astype(np.int32)
    return GroupedPlan(group_center, group_offsets, pair_target, pair_label)


def train_plan_minibatched(syn0: np.ndarray, syn1: np.ndarray, plan: PairPlan,
                           alpha: float, minibatch: int = 16384) -> Tuple[int, float]:
    """Fast vectorized CPU SGD over a pair plan (torch index_add under the
    hood).  Updates within a mini-batch see pre-batch rows (mini-batched
    hogwild — same asynchrony class as the reference's concurrent workers,
    mllib:392-433).  Returns (num_positive_pairs, sum_f_plus) for the
    divergence canary (mllib:411-412)."""
    import torch
    # intra-op thread-pool sync costs ~ms per op under contention — a 500x
    # slowdown for tiny tables.  Single-thread below a work threshold.
    prev_nt = torch.get_num_threads()
    if syn0.size < (1 << 18):
        torch.set_num_threads(1)
    try:
        return _train_plan_minibatched_impl(syn0, syn1, plan, alpha,
                                            minibatch)
    finally:
        torch.set_num_threads(prev_nt)


def _train_plan_minibatched_impl(syn0, syn1, plan, alpha, minibatch):
    import torch
    s0 = torch.from_numpy(syn0)
    s1 = torch.from_numpy(syn1)
    c_all = torch.from_numpy(plan.center.astype(np.int64))
    t_all = torch.from_numpy(plan.target.astype(np.int64))
    l_all = torch.from_numpy(plan.label)
    n_pos = 0
    sum_fplus = 0.0
    for s in range(0, plan.num_pairs, minibatch):
        c = c_all[s:s + minibatch]
        t = t_all[s:s + minibatch]
        lab = l_all[s:s + minibatch]
        c_rows = s0.index_select(0, c)
        t_rows = s1.index_select(0, t)
        f = (c_rows * t_rows).sum(dim=1)
        sig = torch.sigmoid(f.clamp(-MAX_EXP, MAX_EXP))
        sig = torch.where(f > MAX_EXP, torch.ones_like(sig), sig)
        sig = torch.where(f < -MAX_EXP, torch.zeros_like(sig), sig)
        g = (lab - sig) * alpha
        s1.index_add_(0, t, g[:, None] * c_rows)
        s0.index_add_(0, c, g[:, None] * t_rows)
        pos_mask = lab > 0.5
        n_pos += int(pos_mask.sum())
        sum_fplus += float(f[pos_mask].sum())
    return n_pos, sum_fplus


# ---------------------------------------------------------------------------
# Device-side planning (torch ops — GPU-resident in production, CPU in tests)
# ---------------------------------------------------------------------------

@dataclass
class GroupedPlanT:
    """GroupedPlan twin with torch tensors, produced entirely with tensor
    ops on the training device.  The numpy planner costs ~45 s of host time
    per 2M-word batch (benchmarks/results.md); this one runs in
    milliseconds on the GPU and feeds the row-sharded engine without any
    host round-trip."""
    group_center: "object"      # int32 [G] center WORD ids (torch.Tensor)
    group_offsets: "object"     # int64 [G+1] into pair arrays
    pair_target: "object"       # int32 [P] target WORD ids
    pair_label: "object"        # float32 [P]

    @property
    def num_groups(self) -> int:
        return int(self.group_center.numel())

    @property
    def num_pairs(self) -> int:
        return int(self.pair_target.numel())


def make_grouped_plan_torch(tokens, offsets, keep_prob, table,
                            window: int, n_neg: int, gen,
                            window_mode: str = "canonical") -> GroupedPlanT:
    """Torch twin of make_grouped_plan: identical pair-enumeration
    semantics (subsample -> compacted sentences -> shrunk window ->
    negatives with positive-collision drop; targets of one center
    contiguous, ascending).  Randomness comes from ``gen``
    (torch.Generator on the same device), so draws differ from the numpy
    planner's — the *structure* is equivalent (tested with randomness
    pinned) but streams are not bit-identical.

    tokens int32 [T0], offsets int32/int64 [S+1], keep_prob float32
    [vocab] or None, table int32 [table_size] — all on the same device.
    """
    import torch
    dev = tokens.device
    S = offsets.numel() - 1
    lengths = (offsets[1:] - offsets[:-1]).long()
    sent_id = torch.repeat_interleave(torch.arange(S, device=dev), lengths)
    if keep_prob is not None and tokens.numel():
        u = torch.rand(tokens.numel(), generator=gen, device=dev)
        keep = u < keep_prob[tokens.long()]
        toks = tokens[keep]
        sid = sent_id[keep]
    else:
        toks, sid = tokens, sent_id
    T = toks.numel()
    ei32 = torch.zeros(0, dtype=torch.int32, device=dev)
    empty = GroupedPlanT(ei32, torch.zeros(1, dtype=torch.int64, device=dev),
                         ei32.clone(), torch.zeros(0, device=dev))
    if T == 0:
        return empty
    cnt = torch.bincount(sid, minlength=S)
    offs = torch.zeros(S + 1, dtype=torch.int64, device=dev)
    torch.cumsum(cnt, 0, out=offs[1:])
    sent_lo = offs[:-1][sid]
    sent_hi = offs[1:][sid]
    pos = torch.arange(T, dtype=torch.int64, device=dev)
    if window_mode == "canonical":
        b = torch.randint(1, window + 1, (T,), generator=gen, device=dev)
        left, right = b, b
    else:   # reference mode (B2): left b, right b-1, b in [0, window-1]
        b = torch.randint(0, window, (T,), generator=gen, device=dev)
        left = b
        right = torch.clamp(b - 1, min=0) * (b > 0)
    # candidate grid [T, 2W] laid out center-major with ascending target
    # offsets — flattening row-major yields the numpy planner's
    # (center, target)-sorted order without a sort
    d = torch.cat([torch.arange(-window, 0, device=dev),
                   torch.arange(1, window + 1, device=dev)])
    tpos_grid = pos[:, None] + d[None, :]
    is_left = d < 0
    reach = torch.where(is_left[None, :], left[:, None], right[:, None])
    valid = (d.abs()[None, :] <= reach) & \
        torch.where(is_left[None, :], tpos_grid >= sent_lo[:, None],
                    tpos_grid < sent_hi[:, None])
    flat = valid.reshape(-1)
    cpos = pos[:, None].expand(-1, 2 * window).reshape(-1)[flat]
    tpos = tpos_grid.reshape(-1)[flat]
    pt = toks[tpos].to(torch.int32)
    P = int(pt.numel())
    if P == 0:
        return empty
    if n_neg > 0:
        nidx = torch.randint(0, table.numel(), (P, n_neg), generator=gen,
                             device=dev, dtype=torch.int32)
        negs = table[nidx.long()].to(torch.int32)
        nvalid = negs != pt[:, None]
        all_target = torch.cat([pt[:, None], negs], dim=1)
        all_label = torch.cat(
            [torch.ones(P, 1, device=dev),
             torch.zeros(P, n_neg, device=dev)], dim=1)
        all_valid = torch.cat(
            [torch.ones(P, 1, dtype=torch.bool, device=dev), nvalid], dim=1)
        all_cpos = cpos.to(torch.int32)[:, None].expand(-1, 1 + n_neg)
        fmask = all_valid.reshape(-1)
        pair_target = all_target.reshape(-1)[fmask]
        pair_label = all_label.reshape(-1)[fmask]
        pair_cpos = all_cpos.reshape(-1)[fmask]
    else:
        pair_target, pair_label, pair_cpos = pt, \
            torch.ones(P, device=dev), cpos
    # group boundaries: runs of equal center position
    Np = pair_cpos.numel()
    change = torch.nonzero(pair_cpos[1:] != pair_cpos[:-1]).reshape(-1) + 1
    starts = torch.cat([torch.zeros(1, dtype=torch.int64, device=dev),
                        change])
    group_offsets = torch.cat(
        [starts, torch.tensor([Np], dtype=torch.int64, device=dev)])
    group_center = toks[pair_cpos[starts].long()].to(torch.int32)
    return GroupedPlanT(group_center, group_offsets, pair_target.contiguous(),
                        pair_label.contiguous())
