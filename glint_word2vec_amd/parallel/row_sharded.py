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

Memory: the pull cycle stages ~4 cache-sized buffers per in-flight step
(cache+orig per table) of unique_rows x stride elements, and the
pipelined loop keeps two steps in flight — at very large vocabularies
size words_per_step so 8 x unique_rows x stride x dtype fits HBM
(e.g. 80M vocab / 1M-word steps ~= 90 GB; bench warns beyond 200 GB).
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
                 atomic_below: "int | None" = None, atomic_floor: int = 0,
                 shared_negatives: bool = False,
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
        self.atomic_floor = int(atomic_floor)
        self.shared_neg = int(shared_negatives)
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
        if self.shared_neg:
            raise NotImplementedError(
                "shared_negatives needs the counter-RNG planner "
                "(make_plan_counter, CUDA) or the fused world-1 path")
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
            counts.data_ptr(), nb, nt, stream.cuda_stream, self.shared_neg)
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
            start.data_ptr(), center.data_ptr(), nb, nt, stream.cuda_stream,
            self.shared_neg)
        starts = start.nonzero().reshape(-1)
        group_offsets = torch.cat(
            [starts, torch.tensor([total], dtype=torch.int64,
                                  device=self.device)])
        group_center = center[starts]
        return sgns.GroupedPlanT(group_center, group_offsets, target, label)

    # ------------------------------------------------------------------
    # Pipelined pull/train/push cycle (the Glint dotprod/adjust message
    # pipeline, mllib:419-429: the next batch's dotprod overlaps the
    # previous batch's adjust).  pull_begin(k+1) may be issued before
    # train_push(k): its row reads then predate step k's push — one step
    # of hogwild staleness, exactly the reference's fire-and-forget
    # semantics.  All collectives are enqueued in a deterministic order
    # (pull k+1 before push k on every rank), on a dedicated stream so
    # xGMI transfers overlap the compute-stream train kernel.
    # ------------------------------------------------------------------
    def _route32(self, ids: torch.Tensor):
        """Sort unique int32 ids by owner (id % world).  Returns
        (pos, send_counts, sorted_local_ids_i32): pos maps an id's position
        in `ids` to its position in the owner-sorted order."""
        owner = torch.remainder(ids, self.world)
        perm = torch.argsort(owner, stable=True)
        counts = torch.bincount(owner, minlength=self.world)
        local = torch.div(ids.index_select(0, perm), self.world,
                          rounding_mode="floor").int().contiguous()
        pos = torch.empty(ids.numel(), dtype=torch.int32, device=ids.device)
        pos.index_copy_(0, perm, torch.arange(ids.numel(), dtype=torch.int32,
                                              device=ids.device))
        return pos, counts, local

    def _exchange_counts2(self, c0: torch.Tensor, c1: torch.Tensor):
        """One allgather for both tables' request counts.  Returns
        (recv0, recv1) as host numpy arrays."""
        both = torch.stack([c0, c1]).to(self.device)
        if self.world == 1 or not torch.distributed.is_initialized():
            h = both.cpu().numpy()
            return h[0], h[1]
        gathered = [torch.empty_like(both) for _ in range(self.world)]
        torch.distributed.all_gather(gathered, both)
        h = [g.cpu().numpy() for g in gathered]
        r0 = np.array([h[s][0][self.rank] for s in range(self.world)])
        r1 = np.array([h[s][1][self.rank] for s in range(self.world)])
        return r0, r1

    def _gather_native(self, shard: torch.Tensor, idx_i32: torch.Tensor,
                       out: torch.Tensor) -> None:
        """out[i] = shard[idx[i]] at native dtype (fused HIP gather on GPU:
        one pass, no f32 inflation, padding columns travel as-is)."""
        n = int(idx_i32.numel())
        if n == 0:
            return
        if self.is_cuda:
            s = torch.cuda.current_stream(self.device)
            self.native.gather_rows(shard.data_ptr(), int(self.is_bf16),
                                    self.store_stride, idx_i32.data_ptr(), n,
                                    out.data_ptr(), s.cuda_stream)
        else:
            out.copy_(shard.index_select(0, idx_i32.long()))

    def _scatter_add_native(self, shard: torch.Tensor, idx_i32: torch.Tensor,
                            deltas: torch.Tensor) -> None:
        """shard[idx[i]] += deltas[i] (atomic: duplicate ids across source
        ranks sum exactly — the adjust semantics)."""
        n = int(idx_i32.numel())
        if n == 0:
            return
        if self.is_cuda:
            s = torch.cuda.current_stream(self.device)
            self.native.scatter_add_rows(shard.data_ptr(), int(self.is_bf16),
                                         self.store_stride,
                                         idx_i32.data_ptr(), n,
                                         deltas.data_ptr(), s.cuda_stream)
        else:
            shard.index_add_(0, idx_i32.long(), deltas)

    def _sub_native_into(self, a: torch.Tensor, b: torch.Tensor,
                         out: torch.Tensor) -> None:
        """out = a - b at native dtype, one fused pass (the push delta);
        out may alias b (elementwise)."""
        if a.numel() == 0:
            return
        if self.is_cuda:
            s = torch.cuda.current_stream(self.device)
            self.native.sub_rows(a.data_ptr(), b.data_ptr(),
                                 int(self.is_bf16), a.numel(),
                                 out.data_ptr(), s.cuda_stream)
        else:
            torch.sub(a, b, out=out)

    def _comm_ctx(self):
        """Stream the pull/push collectives run on (CUDA: dedicated stream
        overlapping the compute stream's train kernel)."""
        if not self.is_cuda:
            import contextlib
            return contextlib.nullcontext()
        if not hasattr(self, "_pull_stream"):
            self._pull_stream = torch.cuda.Stream(self.device)
        return torch.cuda.stream(self._pull_stream)

    def _ws(self, name: str, rows: int, cols: int,
            dtype: torch.dtype) -> torch.Tensor:
        """Grow-only workspace: big per-step buffers (row caches, wire
        buffers) whose sizes jitter step to step would otherwise force the
        caching allocator into a hipFree/hipMalloc of tens of GB EVERY
        step (measured: ~630 ms of a 680 ms step at 80M vocab).  Buffers
        are keyed by name, grown with 12% headroom, and sliced to size.
        The pipelined loop keeps two steps in flight — callers alternate
        the name's slot suffix."""
        if not hasattr(self, "_workspace"):
            self._workspace = {}
        need = rows * cols
        buf = self._workspace.get(name)
        if buf is None or buf.numel() < need or buf.dtype != dtype:
            cap = max(need + need // 8, 1)
            self._workspace[name] = buf = torch.empty(
                cap, dtype=dtype, device=self.device)
        out = buf[:need]
        return out.view(rows, cols) if cols > 1 else out

    def pull_begin(self, plan, slot: "int | None" = None) -> dict:
        """Stage 1 of a step: route the plan's unique rows to their owners,
        exchange indices, gather + exchange the rows (native dtype on the
        wire — bf16 halves xGMI bytes).  Returns the state consumed by
        train_push().  May be called for step k+1 before train_push(k);
        the two in-flight steps use alternating workspace slots."""
        dev = self.device
        tdtype = self.syn0.dtype
        if slot is None:
            slot = getattr(self, "_ws_slot", 0)
            self._ws_slot = slot ^ 1
        sl = f"{slot}"
        uc, inv_c = torch.unique(plan.group_center, return_inverse=True)
        ut, inv_t = torch.unique(plan.pair_target, return_inverse=True)
        st = {"plan": plan}
        if self.world == 1:
            # no routing: gather straight from the shard (serving-path
            # probes; production world-1 training uses the direct mode)
            cache0 = self._ws("c0" + sl, uc.numel(), self.store_stride,
                              tdtype)
            cache1 = self._ws("c1" + sl, ut.numel(), self.store_stride,
                              tdtype)
            self._gather_native(self.syn0, uc.int().contiguous(), cache0)
            self._gather_native(self.syn1, ut.int().contiguous(), cache1)
            st["gc"] = inv_c.int().contiguous()
            st["gt"] = inv_t.int().contiguous()
            st["cache0"], st["cache1"] = cache0, cache1
            st["orig0"] = st["orig1"] = None
            st["uc_sorted"] = uc
            st["ut_sorted"] = ut
            return st
        pos0, cnt0, loc0 = self._route32(uc)
        pos1, cnt1, loc1 = self._route32(ut)
        rc0, rc1 = self._exchange_counts2(cnt0, cnt1)
        cnt0_h = cnt0.cpu().numpy()
        cnt1_h = cnt1.cpu().numpy()
        if self.is_cuda:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream(dev))
        with self._comm_ctx():
            if self.is_cuda:
                torch.cuda.current_stream(dev).wait_event(ev)
            # index request exchange (int32 local rows)
            idx_recv0 = self._ws("i0" + sl, int(rc0.sum()), 1, torch.int32)
            idx_recv1 = self._ws("i1" + sl, int(rc1.sum()), 1, torch.int32)
            comm.all_to_all_single_v(idx_recv0, loc0, rc0, cnt0_h)
            comm.all_to_all_single_v(idx_recv1, loc1, rc1, cnt1_h)
            if self.is_cuda:
                # loc* were allocated on the compute stream and die with
                # this function — tell the allocator the comm stream still
                # reads them (else their blocks could be re-issued to a
                # compute-stream tensor mid-send)
                loc0.record_stream(torch.cuda.current_stream(dev))
                loc1.record_stream(torch.cuda.current_stream(dev))
            # owners gather requested rows; rows return in owner-major order
            rows_send0 = self._ws("s0" + sl, idx_recv0.numel(),
                                  self.store_stride, tdtype)
            rows_send1 = self._ws("s1" + sl, idx_recv1.numel(),
                                  self.store_stride, tdtype)
            self._gather_native(self.syn0, idx_recv0, rows_send0)
            self._gather_native(self.syn1, idx_recv1, rows_send1)
            cache0 = self._ws("c0" + sl, uc.numel(), self.store_stride,
                              tdtype)
            cache1 = self._ws("c1" + sl, ut.numel(), self.store_stride,
                              tdtype)
            comm.all_to_all_single_v(cache0, rows_send0, cnt0_h, rc0)
            comm.all_to_all_single_v(cache1, rows_send1, cnt1_h, rc1)
            # orig snapshots MUST be ordered after the alltoall on the
            # same (comm) stream — a compute-stream copy could read the
            # cache before the exchange lands
            st["orig0"] = self._ws("o0" + sl, *cache0.shape, tdtype)
            st["orig1"] = self._ws("o1" + sl, *cache1.shape, tdtype)
            st["orig0"].copy_(cache0)
            st["orig1"].copy_(cache1)
            if self.is_cuda:
                done = torch.cuda.Event()
                done.record(torch.cuda.current_stream(dev))
                st["pull_done"] = done
        # cache rows are owner-sorted; remap the plan's inverse indices
        st["gc"] = pos0.index_select(0, inv_c).contiguous()
        st["gt"] = pos1.index_select(0, inv_t).contiguous()
        st["cache0"], st["cache1"] = cache0, cache1
        st["slot"] = sl
        st["idx_recv"] = (idx_recv0, idx_recv1)
        st["splits"] = (cnt0_h, rc0, cnt1_h, rc1)
        return st

    def train_push(self, st: dict, alpha: float) -> None:
        """Stage 2: train the pairs kernel on the pulled cache (compute
        stream), compute native-dtype deltas, exchange them back and
        scatter-add into the owners' shards (adjust).  Push collectives are
        enqueued after any pull_begin already issued — same order on every
        rank."""
        plan = st["plan"]
        if self.is_cuda and "pull_done" in st:
            torch.cuda.current_stream(self.device).wait_event(st["pull_done"])
        if plan.num_pairs > 0:
            self._train_pairs_cache(st["cache0"], st["cache1"], st["gc"],
                                    plan.group_offsets, st["gt"],
                                    plan.pair_label, alpha)
        if self.world == 1:
            if st["uc_sorted"].numel():
                self._write_back(st["uc_sorted"], st["cache0"], 0)
            if st["ut_sorted"].numel():
                self._write_back(st["ut_sorted"], st["cache1"], 1)
            return
        cnt0_h, rc0, cnt1_h, rc1 = st["splits"]
        idx_recv0, idx_recv1 = st["idx_recv"]
        # delta overwrites orig in place (orig is dead after the subtract)
        delta0 = st["orig0"]
        delta1 = st["orig1"]
        self._sub_native_into(st["cache0"], st["orig0"], delta0)
        self._sub_native_into(st["cache1"], st["orig1"], delta1)
        if self.is_cuda:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream(self.device))
        with self._comm_ctx():
            if self.is_cuda:
                torch.cuda.current_stream(self.device).wait_event(ev)
            sl = st["slot"]
            del_recv0 = self._ws("s0" + sl, idx_recv0.numel(),
                                 self.store_stride, delta0.dtype)
            del_recv1 = self._ws("s1" + sl, idx_recv1.numel(),
                                 self.store_stride, delta1.dtype)
            comm.all_to_all_single_v(del_recv0, delta0, rc0, cnt0_h)
            comm.all_to_all_single_v(del_recv1, delta1, rc1, cnt1_h)
            self._scatter_add_native(self.syn0, idx_recv0, del_recv0)
            self._scatter_add_native(self.syn1, idx_recv1, del_recv1)
            if self.is_cuda:
                # later compute-stream work (next train) must not overtake
                push_done = torch.cuda.Event()
                push_done.record(torch.cuda.current_stream(self.device))
                self._last_push = push_done

    def _train_pairs_cache(self, cache0, cache1, gc, go, gt, pl, alpha):
        """Pairs kernel on the native-dtype cache (GPU) / f32 cache (CPU).
        Cache ids are local, so the hybrid threshold does not apply here —
        atomic is all-or-nothing (self.atomic)."""
        if self.is_cuda:
            G = int(gc.numel())
            nb = 1 if self.serial else max(1, min((G + 3) // 4, 8192))
            nt = 64 if self.serial else 256
            if self.serial and self.is_bf16:
                raise ValueError("serial parity mode requires fp32 caches")
            stream = torch.cuda.current_stream(self.device)
            go = go.contiguous()
            pl = pl.contiguous()
            self.native.train_pairs(
                cache0.data_ptr(), cache1.data_ptr(), self.store_stride,
                gc.data_ptr(), go.data_ptr(), G, gt.data_ptr(), pl.data_ptr(),
                float(alpha), self._stats.data_ptr(), nb, nt,
                stream.cuda_stream, 0 if self.serial else 1,
                int(self.atomic), int(self.is_bf16), 2 ** 31 - 1, 0)
            self._inflight = (gc, go, gt, pl, cache0, cache1)
        else:
            st = self.native.train_pairs(
                cache0.numpy(), cache1.numpy(), gc.numpy(),
                go.numpy() if hasattr(go, "numpy") else go,
                gt.numpy(), pl.numpy() if hasattr(pl, "numpy") else pl,
                float(alpha))
            for k in ("pairs", "positives", "words_trained"):
                self._cpu_stats[k] += st[k]
            self._cpu_stats["sum_fplus"] += st["sum_fplus"]

    def _to_plan_t(self, plan) -> "sgns.GroupedPlanT":
        if isinstance(plan, sgns.GroupedPlanT):
            return plan
        dev = self.device
        return sgns.GroupedPlanT(
            torch.from_numpy(np.ascontiguousarray(
                plan.group_center, dtype=np.int32)).to(dev),
            torch.from_numpy(np.ascontiguousarray(
                plan.group_offsets, dtype=np.int64)).to(dev),
            torch.from_numpy(np.ascontiguousarray(
                plan.pair_target, dtype=np.int32)).to(dev),
            torch.from_numpy(np.ascontiguousarray(
                plan.pair_label, dtype=np.float32)).to(dev))

    def train_batch_fused(self, tokens: torch.Tensor, offsets: torch.Tensor,
                          alpha: float, window: int, n_neg: int, seed: int,
                          sent_id_base: int = 0) -> None:
        """World-1 fast path: the shard IS the full table stored at the
        kernel stride, so the fused train kernel runs directly — no plan
        materialisation, no launch-bound cumsum/host sync per step
        (151 -> ~fused-rate words/s; engine.py/bench.py use this at
        world 1, the plan+direct path remains for tests and world > 1)."""
        assert self.world == 1 and self.is_cuda
        if not hasattr(self, "_keep_thr_t"):
            if self.keep_prob is None:
                self._keep_thr_t = None
            else:
                thr = np.minimum(self.keep_prob.astype(np.float64)
                                 * 4294967296.0,
                                 4294967295.0).astype(np.uint32)
                self._keep_thr_t = torch.from_numpy(
                    thr.view(np.int32)).to(self.device)
        if not hasattr(self, "_table_t"):
            self._table_t = torch.from_numpy(self.table).to(self.device)
        num_sent = int(offsets.numel() - 1)
        if num_sent <= 0:
            return
        nblocks = max(1, min((num_sent + 3) // 4, 2048))
        avg_len = max(1, int(tokens.numel()) // num_sent)
        pos_blocks = max(1, min(min(max(1, 8192 // num_sent),
                                    (avg_len + 95) // 96), 11))
        stream = torch.cuda.current_stream(self.device)
        self.native.sgns_train(
            self.syn0.data_ptr(), self.syn1.data_ptr(), int(self.is_bf16),
            self.store_stride, tokens.data_ptr(), offsets.data_ptr(),
            num_sent,
            0 if self._keep_thr_t is None else self._keep_thr_t.data_ptr(),
            self._table_t.data_ptr(), int(self._table_t.numel()),
            float(alpha), int(window), int(n_neg),
            seed & 0xFFFFFFFFFFFFFFFF, int(sent_id_base),
            int(self.window_mode == "reference"), self.atomic_below,
            self._stats.data_ptr(), nblocks, pos_blocks, 256,
            stream.cuda_stream, 0, 0,
            3 if self.store_stride <= 512 else 0, self.atomic_floor,
            self.shared_neg)

    def train_step(self, tokens: np.ndarray, offsets: np.ndarray,
                   alpha: float, window: int, n_neg: int,
                   rng: np.random.Generator, plan=None) -> None:
        """One (unpipelined) data-parallel step over this rank's batch.
        ``plan`` may be a host GroupedPlan or a device GroupedPlanT.
        Ranks with no data still participate in the collectives.  The
        pipelined form is pull_begin()/train_push() (engine.py uses it)."""
        if plan is None:
            plan = self.make_plan(tokens, offsets, window, n_neg, rng)
        plan = self._to_plan_t(plan)
        if (self.world == 1 and self.is_cuda and not self.serial
                and getattr(self, "use_direct", True)
                and isinstance(plan, sgns.GroupedPlanT)):
            # world 1: row r = word r and the shard is stored at the
            # kernel stride — train DIRECTLY on the tables (no
            # unique/pull/write-back at all)
            self._train_pairs_direct(plan, alpha)
            return
        st = self.pull_begin(plan)
        self.train_push(st, alpha)

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
        # direct mode: plan ids are global words, so the hybrid
        # atomic_below threshold applies (atomics on the Zipf head only)
        atomic_flag = self.atomic_below > 0
        self.native.train_pairs(
            self.syn0.data_ptr(), self.syn1.data_ptr(), self.store_stride,
            gc.data_ptr(), go.data_ptr(), G, pt.data_ptr(), pl.data_ptr(),
            float(alpha), self._stats.data_ptr(), nb, 256,
            stream.cuda_stream, 1, int(atomic_flag),
            int(self.is_bf16), self.atomic_below, self.atomic_floor)
        self._inflight = (gc, go, pt, pl)

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
        if self.is_cuda:
            # quiesce in-flight comm-stream pushes/pulls so the D2H
            # snapshot reads a settled table (mid-training checkpoints
            # run inside the pipelined loop)
            torch.cuda.synchronize(self.device)
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

    def load_checkpoint(self, path: str, block_rows: int = 1 << 20) -> None:
        """Streaming resume from ANY checkpoint layout/shard count: each
        rank scans the shard files block-wise and keeps the rows it owns
        (r % world == rank).  Host memory stays O(block_rows*dim); a
        same-world row_mod checkpoint reduces to a straight shard read."""
        import json
        import os
        from ..serving import _shard_row_iter
        with open(os.path.join(path, "shards", "index.json")) as f:
            index = json.load(f)
        if index["vocab"] != self.vocab_size or index["dim"] != self.dim:
            raise ValueError(
                f"checkpoint is vocab={index['vocab']} dim={index['dim']}; "
                f"engine is vocab={self.vocab_size} dim={self.dim}")
        if not index.get("has_syn1", False):
            raise ValueError("checkpoint has no syn1 — cannot resume")
        is_bf16_file = index["dtype"] == "bfloat16"
        dt = np.uint16 if is_bf16_file else np.dtype(index["dtype"])
        D = self.dim
        for t, name in ((self.syn0, "syn0"), (self.syn1, "syn1")):
            for s in range(index["num_shards"]):
                first, count, stride = _shard_row_iter(index, s)
                fp = os.path.join(path, "shards", f"{name}-{s:05d}.bin")
                with open(fp, "rb") as f:
                    for j0 in range(0, count, block_rows):
                        j1 = min(count, j0 + block_rows)
                        block = np.fromfile(f, dtype=dt, count=(j1 - j0) * D) \
                            .reshape(j1 - j0, D)
                        gids = first + np.arange(j0, j1,
                                                 dtype=np.int64) * stride
                        mine = (gids % self.world) == self.rank
                        if not mine.any():
                            continue
                        picked = np.ascontiguousarray(block[mine])
                        rows = (torch.from_numpy(picked)
                                .view(torch.bfloat16).float()
                                if is_bf16_file else
                                torch.from_numpy(picked.astype(
                                    np.float32, copy=False)))
                        t[torch.from_numpy(gids[mine] // self.world),
                          :D] = rows.to(t.dtype).to(self.device)

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
