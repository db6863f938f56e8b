"""Row-sharded multi-GPU SGNS engine — the Glint parameter-server shape on
RCCL alltoallv over xGMI (BASELINE.json north star).

Row r of both tables lives on rank r % world (round-robin interleave keeps
Zipf-hot rows balanced).  Each rank is simultaneously a *worker* over its
own corpus partition and the *server* for its row shard:

  plan pairs (torch ops on the device) ->  unique touched rows  ->
  alltoallv index request to owners    ->  owners gather rows   ->
  alltoallv rows back (pull)           ->  train on the local f32 cache
  (pairs kernel, live hogwild feedback within the step)  ->
  alltoallv row deltas to owners       ->  owners scatter-add (push/adjust)

At world 1 the cycle collapses: shard row r IS table row r and shards are
stored at the kernel stride, so the pairs kernel trains directly on the
tables — no unique, no pull, no write-back (143M words/s at the headline
config, benchmarks/results.md).

This is the reference's dotprod/adjust split with the network role of Akka
messages taken by two alltoallv exchanges; traffic scales with unique rows
x dim, so it is the engine of choice when the touched-row set is sparse
(very large vocabularies) — the dim-sharded engine (dim_sharded.py) covers
the dense case with dimension-independent traffic.

Works on CUDA (HIP train_pairs kernels) and CPU (C++ twin; gloo tests).
"""
from __future__ import annotations

import logging
from typing import Optional, Tuple

import numpy as np
import torch

from ..models import sgns
from ..ops.gpu import GpuStats
from ..vocab import build_unigram_table, keep_probabilities
from . import comm

log = logging.getLogger("glint_word2vec_amd")


class RowShardedSgns:
    def __init__(self, vocab_size: int, dim: int, dtype: str = "float32",
                 device: str = "cuda", seed: int = 1,
                 counts: Optional[np.ndarray] = None,
                 table_size: int = 1_000_000, subsample: float = 0.0,
                 window_mode: str = "canonical", atomic: bool = True,
                 atomic_below: "int | None" = None,
                 init_full_limit: int = 1 << 28):
        self.rank, self.world = comm.init_from_env()
        self.vocab_size = vocab_size
        self.dim = dim
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        self.is_bf16 = dtype == "bfloat16"
        self.window_mode = window_mode
        # cache-update mode for the GPU pairs kernel: True = fp32 atomics
        # (no lost updates; the PS adjust semantics), False = hogwild RMW
        # (the fused kernel's default class; ~2.6x faster).
        # atomic_below: hybrid threshold used by the world-1 DIRECT mode,
        # where plan ids are global rows (sorted by count, so rows < K are
        # the contended Zipf head); the cache modes use the plain bool
        # (cache-local ids carry no frequency meaning).
        self.atomic = atomic
        self.atomic_below = ((2 ** 31 - 1 if atomic else 0)
                             if atomic_below is None else int(atomic_below))
        if self.is_cuda:
            from .. import _hip_native
            self.native = _hip_native
            self.cache_stride = self.native.round_stride(dim)
        else:
            from .. import _cpu_native
            self.native = _cpu_native
            self.cache_stride = dim
        self.my_rows = np.arange(self.rank, vocab_size, self.world,
                                 dtype=np.int64)
        self.shard_size = len(self.my_rows)
        tdtype = torch.bfloat16 if self.is_bf16 else torch.float32
        # shards stored padded to the kernel stride on GPU so the pairs
        # kernel can train DIRECTLY on them at world 1 (no pull/unique/
        # write-back); padding columns provably stay zero
        self.store_stride = self.cache_stride if self.is_cuda else dim
        self.syn0 = torch.zeros((self.shard_size, self.store_stride),
                                dtype=tdtype, device=self.device)
        self.syn1 = torch.zeros((self.shard_size, self.store_stride),
                                dtype=tdtype, device=self.device)
        self._init_shard(seed, init_full_limit)
        counts = (np.ones(vocab_size, dtype=np.int64) if counts is None
                  else counts)
        self.table = build_unigram_table(counts, table_size)
        self.keep_prob = None
        if subsample > 0:
            self.keep_prob = keep_probabilities(counts, int(counts.sum()),
                                                subsample)
        self._stats = torch.zeros(4, dtype=torch.int64, device=self.device)
        self._cpu_stats = dict(pairs=0, positives=0, words_trained=0,
                               sum_fplus=0.0)
        self.serial = False

    def _init_shard(self, seed: int, full_limit: int) -> None:
        """Word2vec init, world-size invariant: same full-matrix stream as
        the other engines, this rank keeps rows r % world == rank."""
        tdtype = self.syn0.dtype
        block = max(self.world, (full_limit // max(self.dim, 1)) //
                    self.world * self.world)
        rng = np.random.default_rng(seed)
        for r0 in range(0, self.vocab_size, block):
            r1 = min(self.vocab_size, r0 + block)
            full = (rng.random((r1 - r0, self.dim), dtype=np.float32) - 0.5) / self.dim
            mine = full[(np.arange(r0, r1) % self.world) == self.rank]
            lo = (r0 + self.world - 1 - self.rank) // self.world
            self.syn0[lo:lo + len(mine), :self.dim] = \
                torch.from_numpy(np.ascontiguousarray(mine)).to(tdtype).to(self.device)

    def load_host(self, syn0, syn1) -> None:
        """Initialise this rank's row shard from full host f32 matrices
        (training resume)."""
        tdtype = self.syn0.dtype
        for host, dev in ((syn0, self.syn0), (syn1, self.syn1)):
            mine = np.ascontiguousarray(host[self.my_rows], dtype=np.float32)
            dev[:, :self.dim] = \
                torch.from_numpy(mine).to(tdtype).to(self.device)

    # ------------------------------------------------------------------
    # pull / push: the alltoallv exchanges (Glint pull / adjust push)
    # ------------------------------------------------------------------
    def _to_ids(self, ids) -> torch.Tensor:
        """Accept numpy or torch row ids; return int64 on the engine device."""
        if isinstance(ids, np.ndarray):
            return torch.from_numpy(np.ascontiguousarray(ids)).long() \
                .to(self.device)
        return ids.long().to(self.device)

    def _route(self, ids: torch.Tensor):
        """Sort global row ids by owner.  Returns (perm, send_counts_host,
        sorted_local_ids) — all device tensors except the host counts."""
        owner = ids % self.world
        perm = torch.argsort(owner, stable=True)
        send_counts = torch.bincount(owner, minlength=self.world)
        local = ids[perm] // self.world
        return perm, send_counts, local

    def _exchange(self, send_chunks, recv_shapes, dtype):
        """alltoallv helper: send_chunks[d] -> rank d; returns recv list."""
        recv = [torch.empty(shape, dtype=dtype, device=send_chunks[0].device)
                for shape in recv_shapes]
        comm.all_to_all_v(recv, send_chunks)
        return recv

    def pull(self, ids, which: int) -> torch.Tensor:
        """Gather rows `ids` (global, numpy or torch) of syn0 (which=0) /
        syn1 (which=1) from their owners.  Returns f32
        [len(ids), cache_stride] on the engine device."""
        shard = self.syn0 if which == 0 else self.syn1
        idx = self._to_ids(ids)
        if self.world == 1:
            out = torch.zeros((idx.numel(), self.cache_stride),
                              dtype=torch.float32, device=self.device)
            B = 1 << 22   # blockwise: bounds the f32 temp at ~7 GB
            for i in range(0, idx.numel(), B):
                out[i:i + B, :self.dim] = \
                    shard.index_select(0, idx[i:i + B])[:, :self.dim].float()
            return out
        perm, send_counts, local_sorted = self._route(idx)
        # 1) exchange request sizes + index lists (counts stay on the
        # engine device: RCCL collectives need device tensors, gloo CPU)
        rc = comm.exchange_counts(send_counts).cpu().numpy()
        bounds = np.concatenate([[0],
                                 np.cumsum(send_counts.cpu().numpy())])
        idx_send = [local_sorted[bounds[d]:bounds[d + 1]].contiguous()
                    for d in range(self.world)]
        idx_recv = self._exchange(idx_send, [int(rc[s]) for s in range(self.world)],
                                  torch.int64)
        # 2) owners gather rows, reply
        row_send = []
        for s in range(self.world):
            rows = torch.zeros((int(rc[s]), self.cache_stride),
                               dtype=torch.float32, device=self.device)
            if int(rc[s]):
                rows[:, :self.dim] = \
                    shard.index_select(0, idx_recv[s])[:, :self.dim].float()
            row_send.append(rows)
        row_recv = self._exchange(
            row_send, [(int(bounds[d + 1] - bounds[d]), self.cache_stride)
                       for d in range(self.world)], torch.float32)
        sorted_rows = torch.cat(row_recv, dim=0)
        out = torch.empty_like(sorted_rows)
        out[perm] = sorted_rows
        return out

    def _pull_cache(self, ids, which: int) -> torch.Tensor:
        """Training-cache pull.  World-1 bf16 hogwild keeps the cache in
        the shard's native dtype (half the kernel bytes — the fused
        kernel's precision class); every other case uses the f32 pull."""
        if (self.world == 1 and self.is_cuda and self.is_bf16
                and not self.atomic and not self.serial):
            shard = self.syn0 if which == 0 else self.syn1
            idx = self._to_ids(ids)
            out = torch.empty((idx.numel(), self.cache_stride),
                              dtype=shard.dtype, device=self.device)
            B = 1 << 22
            for i in range(0, idx.numel(), B):
                out[i:i + B] = shard.index_select(0, idx[i:i + B])
            return out
        return self.pull(ids, which)

    def push_add(self, ids, deltas: torch.Tensor, which: int) -> None:
        """Scatter-add row deltas back to their owners (the adjust push)."""
        shard = self.syn0 if which == 0 else self.syn1
        idx = self._to_ids(ids)
        if self.world == 1:
            sel = shard.index_select(0, idx)
            sel[:, :self.dim] = (sel[:, :self.dim].float()
                                 + deltas[:, :self.dim]).to(shard.dtype)
            shard.index_copy_(0, idx, sel)
            return
        perm, send_counts, local_sorted = self._route(idx)
        rc = comm.exchange_counts(send_counts).cpu().numpy()
        bounds = np.concatenate([[0],
                                 np.cumsum(send_counts.cpu().numpy())])
        idx_send = [local_sorted[bounds[d]:bounds[d + 1]].contiguous()
                    for d in range(self.world)]
        idx_recv = self._exchange(idx_send, [int(rc[s]) for s in range(self.world)],
                                  torch.int64)
        deltas_sorted = deltas.index_select(0, perm)
        del_send = [deltas_sorted[bounds[d]:bounds[d + 1]].contiguous()
                    for d in range(self.world)]
        del_recv = self._exchange(
            del_send, [(int(rc[s]), self.cache_stride)
                       for s in range(self.world)], torch.float32)
        for s in range(self.world):
            if int(rc[s]) == 0:
                continue
            cur = shard.index_select(0, idx_recv[s]).float()
            cur[:, :self.dim] += del_recv[s][:, :self.dim]
            # duplicate local indices across sources are rare (each source
            # deduplicates); across sources index_copy applies sequentially
            shard.index_copy_(0, idx_recv[s], cur.to(shard.dtype))

    # ------------------------------------------------------------------
    def make_plan(self, tokens: np.ndarray, offsets: np.ndarray,
                  window: int, n_neg: int, rng: np.random.Generator):
        """Host-side pair planning for one batch — independent of table
        state, so it can be prefetched on a worker thread while the GPU
        trains the previous step (the reference's async mini-batch workers
        overlapping compute, SURVEY §2.2 pipeline note)."""
        return sgns.make_grouped_plan(tokens, offsets, self.keep_prob,
                                      self.table, window, n_neg, rng,
                                      self.window_mode)

    def make_plan_device(self, tokens, offsets, window: int, n_neg: int,
                         seed: int):
        """Device-side planning (torch ops, models/sgns.py
        make_grouped_plan_torch): ~45 s of numpy host time per 2M-word
        batch becomes milliseconds on the GPU.  ``tokens``/``offsets``
        numpy or torch; deterministic per (engine device, seed)."""
        if not hasattr(self, "_table_t"):
            self._table_t = torch.from_numpy(self.table).to(self.device)
            self._keep_prob_t = (None if self.keep_prob is None else
                                 torch.from_numpy(self.keep_prob)
                                 .to(self.device))
            self._gen = torch.Generator(device=self.device)
        tok = (torch.from_numpy(tokens) if isinstance(tokens, np.ndarray)
               else tokens).to(self.device)
        off = (torch.from_numpy(offsets) if isinstance(offsets, np.ndarray)
               else offsets).to(self.device)
        self._gen.manual_seed(int(seed) & 0x7FFFFFFFFFFFFFFF)
        return sgns.make_grouped_plan_torch(tok, off, self._keep_prob_t,
                                            self._table_t, window, n_neg,
                                            self._gen, self.window_mode)

    def make_plan_counter(self, tokens, offsets, window: int, n_neg: int,
                          seed: int, sent_id_base: int = 0,
                          window_mode_ref: "bool | None" = None):
        """GPU plan via the fused kernel's walker (counter-based RNG,
        rng.py contract): count_pairs -> cumsum -> plan_emit.  Pair
        enumeration is bit-identical to the fused kernel / CPU oracle for
        the same (seed, sent_id_base) — unlike make_plan_device, whose
        draws come from torch's RNG.  CUDA only."""
        assert self.is_cuda, "counter planner runs the HIP walker"
        if not hasattr(self, "_table_t"):
            self._table_t = torch.from_numpy(self.table).to(self.device)
            self._keep_prob_t = (None if self.keep_prob is None else
                                 torch.from_numpy(self.keep_prob)
                                 .to(self.device))
            self._gen = torch.Generator(device=self.device)
        if not hasattr(self, "_keep_thr_t"):
            if self.keep_prob is None:
                self._keep_thr_t = None
            else:
                thr = np.minimum(self.keep_prob.astype(np.float64)
                                 * 4294967296.0,
                                 4294967295.0).astype(np.uint32)
                self._keep_thr_t = torch.from_numpy(
                    thr.view(np.int32)).to(self.device)
        tok = (torch.from_numpy(tokens) if isinstance(tokens, np.ndarray)
               else tokens).to(self.device)
        off = (torch.from_numpy(offsets) if isinstance(offsets, np.ndarray)
               else offsets).to(self.device)
        num_sent = off.numel() - 1
        ref = int(self.window_mode == "reference"
                  if window_mode_ref is None else window_mode_ref)
        kthr = (0 if self._keep_thr_t is None
                else self._keep_thr_t.data_ptr())
        stream = torch.cuda.current_stream(self.device)
        nb = 1 if self.serial else max(1, min((num_sent + 3) // 4, 8192))
        nt = 64 if self.serial else 256
        counts = torch.empty(num_sent, dtype=torch.int64, device=self.device)
        self.native.count_pairs(
            tok.data_ptr(), off.data_ptr(), num_sent, kthr,
            self._table_t.data_ptr(), int(self._table_t.numel()), window,
            n_neg, seed & 0xFFFFFFFFFFFFFFFF, sent_id_base, ref,
            counts.data_ptr(), nb, nt, stream.cuda_stream)
        poff = torch.zeros(num_sent + 1, dtype=torch.int64,
                           device=self.device)
        torch.cumsum(counts, 0, out=poff[1:])
        total = int(poff[-1].item())
        ei32 = torch.zeros(0, dtype=torch.int32, device=self.device)
        if total == 0:
            return sgns.GroupedPlanT(
                ei32, torch.zeros(1, dtype=torch.int64, device=self.device),
                ei32.clone(), torch.zeros(0, device=self.device))
        target = torch.empty(total, dtype=torch.int32, device=self.device)
        label = torch.empty(total, dtype=torch.float32, device=self.device)
        start = torch.empty(total, dtype=torch.uint8, device=self.device)
        center = torch.empty(total, dtype=torch.int32, device=self.device)
        self.native.plan_emit(
            tok.data_ptr(), off.data_ptr(), num_sent, kthr,
            self._table_t.data_ptr(), int(self._table_t.numel()), window,
            n_neg, seed & 0xFFFFFFFFFFFFFFFF, sent_id_base, ref,
            poff.data_ptr(), target.data_ptr(), label.data_ptr(),
            start.data_ptr(), center.data_ptr(), nb, nt, stream.cuda_stream)
        starts = start.nonzero().reshape(-1)
        group_offsets = torch.cat(
            [starts, torch.tensor([total], dtype=torch.int64,
                                  device=self.device)])
        group_center = center[starts]
        return sgns.GroupedPlanT(group_center, group_offsets, target, label)

    def train_step(self, tokens: np.ndarray, offsets: np.ndarray,
                   alpha: float, window: int, n_neg: int,
                   rng: np.random.Generator, plan=None) -> None:
        """One data-parallel step over this rank's batch.  ``plan`` may be
        a host GroupedPlan or a device GroupedPlanT (make_plan_device).
        Ranks with no data still participate in the collectives."""
        if plan is None:
            plan = self.make_plan(tokens, offsets, window, n_neg, rng)
        if isinstance(plan, sgns.GroupedPlanT):
            if self.world == 1 and self.is_cuda and not self.serial:
                # world 1: row r = word r and the shard is stored at the
                # kernel stride — train DIRECTLY on the tables (no
                # unique/pull/write-back at all)
                self._train_pairs_direct(plan, alpha)
                return
            # unique on int32 ids: half the radix-sort bytes of .long()
            uc, inv_c = torch.unique(plan.group_center,
                                     return_inverse=True)
            ut, inv_t = torch.unique(plan.pair_target,
                                     return_inverse=True)
            cache0 = self._pull_cache(uc, 0)
            cache1 = self._pull_cache(ut, 1)
            # world 1: the trained cache IS the new row value — write it
            # back directly and skip the orig clones + delta temps
            # (3 extra cache-sized buffers; matters at 80M-vocab scale)
            local = self.world == 1
            orig0 = None if local else cache0.clone()
            orig1 = None if local else cache1.clone()
            if plan.num_pairs > 0:
                self._train_pairs_t(cache0, cache1, inv_c.int(),
                                    plan.group_offsets, inv_t.int(),
                                    plan.pair_label, alpha)
            if local:
                self._write_back(uc, cache0, 0)
                self._write_back(ut, cache1, 1)
            else:
                self.push_add(uc, cache0 - orig0, 0)
                self.push_add(ut, cache1 - orig1, 1)
            return
        uc, inv_c = np.unique(plan.group_center, return_inverse=True)
        ut, inv_t = np.unique(plan.pair_target, return_inverse=True)
        cache0 = self.pull(uc.astype(np.int64), 0)
        cache1 = self.pull(ut.astype(np.int64), 1)
        local = self.world == 1
        orig0 = None if local else cache0.clone()
        orig1 = None if local else cache1.clone()
        if plan.num_pairs > 0:
            self._train_pairs(cache0, cache1, inv_c.astype(np.int32),
                              plan.group_offsets,
                              inv_t.astype(np.int32), plan.pair_label, alpha)
        if local:
            self._write_back(self._to_ids(uc.astype(np.int64)), cache0, 0)
            self._write_back(self._to_ids(ut.astype(np.int64)), cache1, 1)
        else:
            self.push_add(uc.astype(np.int64), cache0 - orig0, 0)
            self.push_add(ut.astype(np.int64), cache1 - orig1, 1)

    def _write_back(self, ids: torch.Tensor, rows: torch.Tensor,
                    which: int) -> None:
        """World-1 fast path: replace owned rows with the trained cache
        (ids unique, so copy == add-delta)."""
        shard = self.syn0 if which == 0 else self.syn1
        assert rows.shape[1] == self.store_stride
        shard.index_copy_(0, self._to_ids(ids), rows.to(shard.dtype))

    def _train_pairs_direct(self, plan, alpha: float) -> None:
        if plan.num_pairs == 0:
            return
        G = plan.num_groups
        nb = max(1, min((G + 3) // 4, 8192))
        stream = torch.cuda.current_stream(self.device)
        gc = plan.group_center.contiguous()   # global word id == shard row
        go = plan.group_offsets.contiguous()
        pt = plan.pair_target.contiguous()
        pl = plan.pair_label.contiguous()
        pair_mode = 0 if (self.atomic and not self.is_bf16) else 1
        self.native.train_pairs(
            self.syn0.data_ptr(), self.syn1.data_ptr(), self.store_stride,
            gc.data_ptr(), go.data_ptr(), G, pt.data_ptr(), pl.data_ptr(),
            float(alpha), self._stats.data_ptr(), nb, 256,
            stream.cuda_stream, pair_mode, int(self.atomic),
            int(self.is_bf16))
        self._inflight = (gc, go, pt, pl)

    def _train_pairs_t(self, cache0, cache1, group_center, group_offsets,
                       pair_target, pair_label, alpha):
        """Device-tensor twin of _train_pairs: no host conversion, no
        stream sync (everything stays queued on the compute stream)."""
        if self.is_cuda:
            G = int(group_center.numel())
            nb = 1 if self.serial else max(1, min((G + 3) // 4, 8192))
            nt = 64 if self.serial else 256
            stream = torch.cuda.current_stream(self.device)
            gc = group_center.contiguous()
            go = group_offsets.contiguous()
            pt = pair_target.contiguous()
            pl = pair_label.contiguous()
            self.native.train_pairs(
                cache0.data_ptr(), cache1.data_ptr(), self.cache_stride,
                gc.data_ptr(), go.data_ptr(), G, pt.data_ptr(), pl.data_ptr(),
                float(alpha), self._stats.data_ptr(), nb, nt,
                stream.cuda_stream,
                0 if (self.serial or self.atomic) else 1,
                int(self.atomic), int(cache0.dtype == torch.bfloat16))
            # keep args alive until the kernel drains
            self._inflight = (gc, go, pt, pl, cache0, cache1)
        else:
            st = self.native.train_pairs(
                cache0.numpy(), cache1.numpy(), group_center.numpy(),
                group_offsets.numpy(), pair_target.numpy(),
                pair_label.numpy(), float(alpha))
            for k in ("pairs", "positives", "words_trained"):
                self._cpu_stats[k] += st[k]
            self._cpu_stats["sum_fplus"] += st["sum_fplus"]

    def _train_pairs(self, cache0, cache1, group_center, group_offsets,
                     pair_target, pair_label, alpha):
        if self.is_cuda:
            gc = torch.from_numpy(group_center).to(self.device)
            go = torch.from_numpy(group_offsets.astype(np.int64)).to(self.device)
            pt = torch.from_numpy(pair_target).to(self.device)
            pl = torch.from_numpy(pair_label).to(self.device)
            stream = torch.cuda.current_stream(self.device)
            G = len(group_center)
            nb = 1 if self.serial else max(1, min((G + 3) // 4, 8192))
            nt = 64 if self.serial else 256
            self.native.train_pairs(
                cache0.data_ptr(), cache1.data_ptr(), self.cache_stride,
                gc.data_ptr(), go.data_ptr(), G, pt.data_ptr(), pl.data_ptr(),
                float(alpha), self._stats.data_ptr(), nb, nt,
                stream.cuda_stream,
                0 if (self.serial or self.atomic) else 1,
                int(self.atomic), 0)
            torch.cuda.current_stream(self.device).synchronize()
        else:
            st = self.native.train_pairs(cache0.numpy(), cache1.numpy(),
                                         group_center,
                                         group_offsets.astype(np.int64),
                                         pair_target, pair_label, float(alpha))
            for k in ("pairs", "positives", "words_trained"):
                self._cpu_stats[k] += st[k]
            self._cpu_stats["sum_fplus"] += st["sum_fplus"]

    # ------------------------------------------------------------------
    def read_stats(self, reset: bool = True) -> GpuStats:
        if self.is_cuda:
            h = self._stats.cpu()
            out = GpuStats(int(h[0]), int(h[1]), int(h[2]),
                           float(h[3:4].view(torch.float64)[0]))
            if reset:
                self._stats.zero_()
            return out
        st = self._cpu_stats
        out = GpuStats(st["pairs"], st["positives"], st["words_trained"],
                       st["sum_fplus"])
        if reset:
            self._cpu_stats = dict(pairs=0, positives=0, words_trained=0,
                                   sum_fplus=0.0)
        return out

    def save_checkpoint(self, path: str, config, vocab) -> None:
        """Each rank writes its own shard file — the reference's per-PS
        parallel save (matrix.save, mllib:493-498).  Shard layout is the
        host checkpoint's row_mod interleave, so GlintWord2VecModel.load
        reads it directly."""
        import json
        import os
        if self.rank == 0:
            os.makedirs(os.path.join(path, "shards"), exist_ok=True)
            import time as _t
            meta = {
                "class": "glint_word2vec_amd.GlintWord2VecModel",
                "timestamp": int(_t.time() * 1000),
                "numWords": vocab.num_words,
                "vectorSize": self.dim,
                "paramMap": config.to_dict(),
            }
            with open(os.path.join(path, "metadata"), "w") as f:
                json.dump(meta, f, indent=2, sort_keys=True)
            vocab.save_words(os.path.join(path, "words"))
            np.save(os.path.join(path, "counts.npy"), vocab.counts)
            with open(os.path.join(path, "shards", "index.json"), "w") as f:
                json.dump({"num_shards": self.world, "vocab": self.vocab_size,
                           "dim": self.dim, "dtype": "float32",
                           "layout": "row_mod", "has_syn1": True}, f, indent=2)
        comm.barrier()
        block = 1 << 20
        for t, name in ((self.syn0, "syn0"), (self.syn1, "syn1")):
            with open(os.path.join(path, "shards",
                                   f"{name}-{self.rank:05d}.bin"), "wb") as f:
                for r0 in range(0, self.shard_size, block):
                    r1 = min(self.shard_size, r0 + block)
                    f.write(np.ascontiguousarray(
                        t[r0:r1, :self.dim].float().cpu().numpy(),
                        dtype=np.float32).tobytes())
        comm.barrier()

    def load_checkpoint(self, path: str) -> None:
        """Each rank reads its own shard (requires num_shards == world and
        row_mod layout — i.e. a checkpoint written by this engine at the
        same world size, or by save_model(num_shards=world))."""
        import json
        import os
        with open(os.path.join(path, "shards", "index.json")) as f:
            index = json.load(f)
        if (index["num_shards"] != self.world or
                index.get("layout", "row_mod") != "row_mod"):
            raise ValueError(
                f"checkpoint has {index['num_shards']} {index.get('layout')} "
                f"shards; this engine needs {self.world} row_mod shards — "
                "load via GlintWord2VecModel.load instead")
        dt = np.dtype(index["dtype"])
        for t, name in ((self.syn0, "syn0"), (self.syn1, "syn1")):
            buf = np.fromfile(os.path.join(path, "shards",
                                           f"{name}-{self.rank:05d}.bin"),
                              dtype=dt).reshape(self.shard_size, self.dim)
            t[:, :self.dim] = \
                torch.from_numpy(buf).to(t.dtype).to(self.device)

    def to_host(self) -> Tuple[np.ndarray, np.ndarray]:
        out = []
        for shard in (self.syn0, self.syn1):
            my = shard[:, :self.dim].float()
            if self.world == 1:
                out.append(my.cpu().numpy().copy())
                continue
            maxn = (self.vocab_size + self.world - 1) // self.world
            padded = torch.zeros((maxn, self.dim), dtype=torch.float32,
                                 device=self.device)
            padded[:self.shard_size] = my
            gathered = [torch.empty_like(padded) for _ in range(self.world)]
            torch.distributed.all_gather(gathered, padded)
            full = np.empty((self.vocab_size, self.dim), dtype=np.float32)
            for r in range(self.world):
                rows = np.arange(r, self.vocab_size, self.world)
                full[rows] = gathered[r][:len(rows)].cpu().numpy()
            out.append(full)
        return out[0], out[1]
