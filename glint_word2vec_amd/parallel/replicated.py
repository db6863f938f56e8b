"""Replicated data-parallel SGNS engine: each GPU trains the fused kernel
hogwild on its own corpus partition against a full table replica; every
``sync_every`` steps the table *deltas* are summed across ranks with one
RCCL allreduce and folded into an fp32 master.

This is the bounded-staleness analog of the reference's async workers
streaming against shared parameter-server state (mllib:392-433, with
`adjust` fire-and-forget): between syncs each replica plays both worker
and server for itself; the merge applies every rank's updates exactly once
(sum of deltas), like the PS would have, just ``sync_every * words_per_step``
words late.  288 GB HBM3E makes full replication viable far beyond the
reference's per-server memory budget (README.md:69) — up to multi-million
vocabularies; beyond that the dim-/row-sharded engines take over
(DESIGN.md).

The fp32 master accumulates merges exactly even when replicas store bf16
(small deltas never vanish against bf16 ulps).
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..ops.gpu import GpuStats
from ..vocab import build_unigram_table, keep_probabilities
from . import comm


class ReplicatedSgns:
    def __init__(self, vocab_size: int, dim: int, dtype: str = "float32",
                 device: str = "cuda", seed: int = 1,
                 counts: Optional[np.ndarray] = None,
                 table_size: int = 1_000_000, subsample: float = 0.0,
                 window_mode: str = "canonical", sync_every: int = 4,
                 atomic: bool = False, atomic_below: "int | None" = None,
                 atomic_floor: int = 0, shared_negatives: bool = False):
        self.rank, self.world = comm.init_from_env()
        self.vocab_size = vocab_size
        self.dim = dim
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        self.window_mode = window_mode
        self.sync_every = max(1, sync_every)
        self.atomic = atomic
        self.atomic_below = ((2 ** 31 - 1 if atomic else 0)
                             if atomic_below is None else int(atomic_below))
        self.atomic_floor = int(atomic_floor)
        self.shared_neg = int(shared_negatives)
        self._steps_since_sync = 0
        counts = (np.ones(vocab_size, dtype=np.int64) if counts is None
                  else counts)
        if self.is_cuda:
            from ..ops.gpu import GpuSgns
            self.gs = GpuSgns(vocab_size, dim, dtype=dtype,
                              device=str(self.device), seed=seed)
            self.gs.set_table(build_unigram_table(counts, table_size))
            if subsample > 0:
                self.gs.set_subsample(counts, int(counts.sum()), subsample)
            self.syn0, self.syn1 = self.gs.syn0, self.gs.syn1
        else:
            from .. import _cpu_native
            self.native = _cpu_native
            from ..models import sgns as sgns_mod
            s0, s1 = sgns_mod.init_tables(vocab_size, dim, seed)
            self.syn0 = torch.from_numpy(s0)
            self.syn1 = torch.from_numpy(s1)
            self.table = build_unigram_table(counts, table_size)
            self.keep_prob = (keep_probabilities(counts, int(counts.sum()),
                                                 subsample)
                              if subsample > 0 else None)
            self._cpu_stats = dict(pairs=0, positives=0, words_trained=0,
                                   sum_fplus=0.0)
        # fp32 master = the agreed cross-rank state at the last sync
        self.master0 = self.syn0.float().clone()
        self.master1 = self.syn1.float().clone()

    # ------------------------------------------------------------------
    def train_step(self, tokens, offsets, alpha, window, n_neg, seed,
                   sent_id_base=0) -> None:
        """One local step on THIS rank's batch (tokens/offsets differ per
        rank — data parallel).  Triggers a sync every sync_every steps; all
        ranks must call train_step in lockstep counts."""
        if self.is_cuda:
            self.gs.train_batch(tokens, offsets, alpha, window, n_neg, seed,
                                sent_id_base=sent_id_base,
                                window_mode=self.window_mode,
                                atomic=self.atomic,
                                atomic_below=(None if self.atomic_below
                                              >= 2 ** 31 - 1
                                              else self.atomic_below),
                                atomic_floor=self.atomic_floor,
                                shared_negatives=bool(self.shared_neg))
        else:
            st = self.native.train_batch(
                self.syn0.numpy(), self.syn1.numpy(),
                np.ascontiguousarray(tokens), np.ascontiguousarray(offsets),
                self.keep_prob, self.table, float(alpha), int(window),
                int(n_neg), seed & 0xFFFFFFFFFFFFFFFF, int(sent_id_base),
                self.window_mode, 1,
                shared_negatives=self.shared_neg)
            for k in ("pairs", "positives", "words_trained"):
                self._cpu_stats[k] += st[k]
            self._cpu_stats["sum_fplus"] += st["sum_fplus"]
        self._steps_since_sync += 1
        if self._steps_since_sync >= self.sync_every:
            self.sync()

    def sync(self, flush: bool = False) -> None:
        """Merge table deltas across ranks, applied ONE sync period late so
        the allreduce overlaps training (round-2 overlap design):

          round r:  capture delta_r = replica - master; start
                    allreduce(delta_r) on the comm stream; training of the
                    next sync_every steps proceeds concurrently.
          round r+1: wait allreduce(delta_{r}); master += sum(delta_r);
                    replica += sum(delta_r) - own(delta_r)  (the replica
                    already contains its own delta plus newer local
                    updates — every update still lands exactly once).

        This is the reference's fire-and-forget adjust at the engine
        level: cross-rank state is bounded-stale by <= 2*sync_every steps
        instead of sync blocking the step loop (~15 ms per sync at world 8
        for vocab-1M tables).  ``flush=True`` drains the pipeline AND runs
        one synchronous round — callers that need the agreed state
        (save/to_host) get it."""
        self._steps_since_sync = 0
        if self.world == 1:
            self.master0.copy_(self.syn0.float())
            self.master1.copy_(self.syn1.float())
            return
        self._apply_pending()
        self._begin_round()
        if flush:
            self._apply_pending()

    # -- async round machinery ------------------------------------------
    def _comm_ctx(self):
        if not self.is_cuda:
            import contextlib
            return contextlib.nullcontext()
        if not hasattr(self, "_comm_stream"):
            self._comm_stream = torch.cuda.Stream(self.device)
        return torch.cuda.stream(self._comm_stream)

    def _begin_round(self) -> None:
        """Capture deltas and launch their allreduce on the comm stream."""
        pend = []
        for syn, master in ((self.syn0, self.master0),
                            (self.syn1, self.master1)):
            own = syn.float() - master          # on compute stream
            total = own.clone()
            pend.append((own, total))
        if self.is_cuda:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream(self.device))
        with self._comm_ctx():
            if self.is_cuda:
                torch.cuda.current_stream(self.device).wait_event(ev)
            for own, total in pend:
                comm.all_reduce_sum_compressed(total)
            if self.is_cuda:
                done = torch.cuda.Event()
                done.record(torch.cuda.current_stream(self.device))
                self._round_done = done
        self._pending = pend

    def _apply_pending(self) -> None:
        """Fold a completed round into master + replica (compute stream)."""
        pend = getattr(self, "_pending", None)
        if pend is None:
            return
        self._pending = None
        if self.is_cuda:
            torch.cuda.current_stream(self.device).wait_event(
                self._round_done)
        for (own, total), (syn, master) in zip(
                pend, ((self.syn0, self.master0),
                       (self.syn1, self.master1))):
            master += total
            # replica already holds `own` (and newer local updates):
            # add only the other ranks' contribution
            total -= own
            syn += total.to(syn.dtype)

    def load_host(self, syn0, syn1) -> None:
        """Initialise from full host f32 matrices (training resume)."""
        import torch as _t
        s0 = _t.from_numpy(np.ascontiguousarray(syn0, dtype=np.float32))
        s1 = _t.from_numpy(np.ascontiguousarray(syn1, dtype=np.float32))
        d = self.dim
        self.master0[:, :d] = s0.to(self.master0.device)
        self.master1[:, :d] = s1.to(self.master1.device)
        self.syn0.copy_(self.master0.to(self.syn0.dtype))
        self.syn1.copy_(self.master1.to(self.syn1.dtype))
        self._steps_since_sync = 0

    # ------------------------------------------------------------------
    def read_stats(self, reset: bool = True) -> GpuStats:
        if self.is_cuda:
            return self.gs.read_stats(reset)
        st = self._cpu_stats
        out = GpuStats(st["pairs"], st["positives"], st["words_trained"],
                       st["sum_fplus"])
        if reset:
            self._cpu_stats = dict(pairs=0, positives=0, words_trained=0,
                                   sum_fplus=0.0)
        return out

    def save_checkpoint(self, path: str, config, vocab,
                        num_shards: int = 8,
                        block_rows: int = 1 << 20) -> None:
        """Streamed checkpoint from the fp32 masters (replicated — rank 0
        writes, host memory O(block_rows * dim)).  Collective: sync() flushes
        pending deltas on every rank first."""
        self.sync(flush=True)
        if self.rank == 0:
            from ..checkpoint import save_model_streaming

            def row_block(which, r0, r1):
                m = self.master0 if which == 0 else self.master1
                blk = m[r0:r1, :self.dim]
                return (blk.cpu().numpy() if self.is_cuda
                        else blk.numpy())

            save_model_streaming(path, config, vocab, row_block,
                                 num_shards=num_shards,
                                 block_rows=block_rows)
        comm.barrier()

    def to_host(self):
        self.sync(flush=True)
        d = self.dim
        if self.is_cuda:
            return (self.master0[:, :d].cpu().numpy().copy(),
                    self.master1[:, :d].cpu().numpy().copy())
        return (self.master0[:, :d].numpy().copy(),
                self.master1[:, :d].numpy().copy())
