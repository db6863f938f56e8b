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
                 atomic_floor: int = 0):
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
                                atomic_floor=self.atomic_floor)
        else:
            st = self.native.train_batch(
                self.syn0.numpy(), self.syn1.numpy(),
                np.ascontiguousarray(tokens), np.ascontiguousarray(offsets),
                self.keep_prob, self.table, float(alpha), int(window),
                int(n_neg), seed & 0xFFFFFFFFFFFFFFFF, int(sent_id_base),
                self.window_mode, 1)
            for k in ("pairs", "positives", "words_trained"):
                self._cpu_stats[k] += st[k]
            self._cpu_stats["sum_fplus"] += st["sum_fplus"]
        self._steps_since_sync += 1
        if self._steps_since_sync >= self.sync_every:
            self.sync()

    def sync(self) -> None:
        """delta = replica - master; allreduce(sum); master += delta;
        replica <- master.  Applies every rank's updates exactly once.

        On CUDA the tables are processed in row chunks with the allreduce
        on a separate stream, so chunk k's xGMI transfer overlaps chunk
        k+1's delta compute — same result, about half the sync wall time.
        """
        self._steps_since_sync = 0
        if self.world == 1:
            self.master0.copy_(self.syn0.float())
            self.master1.copy_(self.syn1.float())
            return
        if self.is_cuda:
            self._sync_cuda_pipelined()
            return
        for syn, master in ((self.syn0, self.master0),
                            (self.syn1, self.master1)):
            delta = syn.float() - master
            comm.all_reduce_sum_compressed(delta)
            master += delta
            syn.copy_(master.to(syn.dtype))

    def _sync_cuda_pipelined(self, chunk_rows: int = 1 << 19) -> None:
        if not hasattr(self, "_comm_stream"):
            self._comm_stream = torch.cuda.Stream(self.device)
        cs = self._comm_stream
        comp = torch.cuda.current_stream(self.device)
        prev = None   # (syn_c, master_c, delta, ar_event)

        def apply(entry):
            syn_c, master_c, delta, ev = entry
            comp.wait_event(ev)
            master_c += delta
            syn_c.copy_(master_c.to(syn_c.dtype))

        for syn, master in ((self.syn0, self.master0),
                            (self.syn1, self.master1)):
            for r0 in range(0, syn.shape[0], chunk_rows):
                syn_c = syn[r0:r0 + chunk_rows]
                master_c = master[r0:r0 + chunk_rows]
                delta = syn_c.float() - master_c
                ev = torch.cuda.Event()
                ev.record(comp)
                with torch.cuda.stream(cs):
                    cs.wait_event(ev)
                    comm.all_reduce_sum_compressed(delta)
                    ar_ev = torch.cuda.Event()
                    ar_ev.record(cs)
                if prev is not None:
                    apply(prev)
                prev = (syn_c, master_c, delta, ar_ev)
        if prev is not None:
            apply(prev)

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
        self.sync()
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
        self.sync()
        d = self.dim
        if self.is_cuda:
            return (self.master0[:, :d].cpu().numpy().copy(),
                    self.master1[:, :d].cpu().numpy().copy())
        return (self.master0[:, :d].numpy().copy(),
                self.master1[:, :d].numpy().copy())
