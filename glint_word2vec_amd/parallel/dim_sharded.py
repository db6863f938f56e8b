"""Dimension-sharded multi-GPU SGNS engine (CIKM'16 scheme on RCCL/xGMI —
DESIGN.md).

Each rank owns a contiguous dim-slice of BOTH tables.  Every rank walks the
same token batch with the same counter-based RNG, so pair enumeration is
identical everywhere.  Per chunk:

    count pairs -> partial dots over my slice -> allreduce(f) ->
    sigmoid/gradient + slice updates (fully local)

Network traffic is dimension-independent: ~4 bytes per pair (the f scalar),
never an index, never a row — the reference's network-efficiency claim
(README.md:7-9) realized with one RCCL allreduce.

Works on CUDA (HIP kernels) and CPU (C++ twins, used by the gloo
multi-process tests).
"""
from __future__ import annotations

import logging
from typing import Optional

import numpy as np
import torch

from ..vocab import build_unigram_table, keep_probabilities
from . import comm
from ..ops.gpu import GpuStats

log = logging.getLogger("glint_word2vec_amd")


def slice_bounds(dim: int, rank: int, world: int) -> tuple[int, int]:
    lo = rank * dim // world
    hi = (rank + 1) * dim // world
    return lo, hi


class DimShardedSgns:
    def __init__(self, vocab_size: int, dim: int, dtype: str = "float32",
                 device: str = "cuda", seed: int = 1,
                 counts: Optional[np.ndarray] = None,
                 table_size: int = 1_000_000, subsample: float = 0.0,
                 window_mode: str = "canonical", chunk_words: int = 1 << 19,
                 f_correction: bool = True, atomic: bool = True,
                 atomic_below: "int | None" = None, atomic_floor: int = 0,
                 shared_negatives: bool = False,
                 narrow: "bool | None" = None,
                 init_full_limit: int = 1 << 28):
        self.rank, self.world = (comm.init_from_env() if torch.distributed.is_available()
                                 else (0, 1))
        self.vocab_size = vocab_size
        self.dim = dim
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        self.is_bf16 = dtype == "bfloat16"
        self.window_mode = window_mode
        self.chunk_words = chunk_words
        self.f_correction = f_correction
        self.atomic = atomic
        # row-id threshold for atomics (update_mode="hybrid"); None derives
        # all-or-nothing from the atomic flag
        self.atomic_below = ((2 ** 31 - 1 if atomic else 0)
                             if atomic_below is None else int(atomic_below))
        self.atomic_floor = int(atomic_floor)
        self.shared_neg = int(shared_negatives)
        self.lo, self.hi = slice_bounds(dim, self.rank, self.world)
        self.width = self.hi - self.lo

        if self.is_cuda:
            from .. import _hip_native
            self.native = _hip_native
            # narrow storage: stride = round_up(width, 8) removes the
            # 64-element padding waste for thin dim slices (NOTES backlog
            # #2, measured in benchmarks/results.md); requires the pair2
            # masked kernels, so serial-parity tests pass narrow=False.
            # Auto only when padding waste is >= 25%: a non-64-multiple
            # stride misaligns row starts, which costs far more than a
            # small byte saving on wide bandwidth-bound slices (measured:
            # dim 300 at stride 304 runs 2.4x SLOWER than padded 320;
            # width 38 at stride 40 runs 2.6% faster than padded 64)
            # round-2 re-measure at the 8-GPU slice shape (width 38):
            # padded stride 64 = 230.4M vs narrow stride 40 = 216.0M
            # words/s per rank (the kernels' predicates changed since the
            # round-1 +2.6% narrow result), and the capacity saving is
            # irrelevant against 288 GB HBM (config 4 at world 8: 12.8 vs
            # 20.5 GB).  Auto-narrow is therefore OFF; explicit
            # narrow=True keeps the masked storage available.
            if narrow is None:
                narrow = False
            self.narrow = bool(narrow)
            self.stride = ((self.width + 7) // 8 * 8 if self.narrow
                           else self.native.round_stride(max(self.width, 1)))
        else:
            from .. import _cpu_native
            self.native = _cpu_native
            self.narrow = False
            self.stride = self.width
        tdtype = torch.bfloat16 if self.is_bf16 else torch.float32
        if self.is_bf16 and not self.is_cuda:
            raise ValueError("bf16 dim-sharded path is GPU-only")
        self.syn0 = torch.zeros((vocab_size, self.stride), dtype=tdtype,
                                device=self.device)
        self.syn1 = torch.zeros((vocab_size, self.stride), dtype=tdtype,
                                device=self.device)
        self._init_slices(seed, init_full_limit)

        counts = (np.ones(vocab_size, dtype=np.int64) if counts is None
                  else counts)
        table = build_unigram_table(counts, table_size)
        self.table = torch.from_numpy(table).to(self.device)
        self.keep_prob: Optional[np.ndarray] = None
        self.keep_thr: Optional[torch.Tensor] = None
        if subsample > 0:
            kp = keep_probabilities(counts, int(counts.sum()), subsample)
            self.keep_prob = kp
            thr = np.minimum(kp.astype(np.float64) * 4294967296.0,
                             4294967295.0).astype(np.uint32)
            self.keep_thr = torch.from_numpy(thr.view(np.int32)).to(self.device)
        self._stats = torch.zeros(4, dtype=torch.int64, device=self.device)
        self._cpu_stats = dict(pairs=0, positives=0, words_trained=0,
                               sum_fplus=0.0)
        self.serial = False   # tests: single-wave launches (oracle order)
        # update-phase kernel variant: 1 = two pairs/wave, 3 = + 2-deep
        # block pipelining (A/B via benchmarks/narrow_probe.py)
        self.pair_mode = 1

    def _init_slices(self, seed: int, full_limit: int) -> None:
        """Same init as the single-GPU path: full-matrix U(-.5/dim,.5/dim)
        sliced by column, so results are world-size invariant (tested).  For
        huge vocab*dim the init is generated per row-block."""
        tdtype = self.syn0.dtype
        block = max(1, full_limit // max(self.dim, 1))
        rng = np.random.default_rng(seed)
        for r0 in range(0, self.vocab_size, block):
            r1 = min(self.vocab_size, r0 + block)
            full = (rng.random((r1 - r0, self.dim), dtype=np.float32) - 0.5) / self.dim
            sl = torch.from_numpy(np.ascontiguousarray(full[:, self.lo:self.hi]))
            self.syn0[r0:r1, :self.width] = sl.to(tdtype).to(self.device)

    def load_host(self, syn0, syn1) -> None:
        """Initialise this rank's dim slice from full host f32 matrices
        (training resume)."""
        tdtype = self.syn0.dtype
        for host, dev in ((syn0, self.syn0), (syn1, self.syn1)):
            sl = torch.from_numpy(np.ascontiguousarray(
                host[:, self.lo:self.hi], dtype=np.float32))
            dev[:, :self.width] = sl.to(tdtype).to(self.device)

    # ------------------------------------------------------------------
    def train_step(self, tokens: torch.Tensor, offsets: torch.Tensor,
                   alpha: float, window: int, n_neg: int, seed: int,
                   sent_id_base: int = 0,
                   offsets_host: Optional[np.ndarray] = None) -> None:
        """One step over a batch.  ``tokens``/``offsets`` must be identical
        on every rank (dim-sharding splits compute by dimension, not data)."""
        if offsets_host is None:
            offsets_host = offsets.cpu().numpy()
        num_sent = len(offsets_host) - 1
        if num_sent <= 0:
            return
        # chunk boundaries: greedily pack sentences up to chunk_words tokens
        sent_lens = np.diff(offsets_host)
        chunks = []
        s0 = 0
        acc = 0
        for s in range(num_sent):
            acc += sent_lens[s]
            if acc >= self.chunk_words or s == num_sent - 1:
                chunks.append((s0, s + 1))
                s0, acc = s + 1, 0
        if self.is_cuda:
            if self.narrow and self.serial:
                raise ValueError("serial parity mode requires narrow=False")
            if (self.world == 1 and not self.serial and not self.narrow
                    and getattr(self, "single_pass_world1", True)):
                # world 1 collapses the CIKM scheme: the local partial IS
                # the full dot, so the dots/allreduce/update pipeline
                # reduces EXACTLY to the fused single-kernel form —
                # dot + sigmoid + update in one pass, zero redundant row
                # reads (the "single-pass variant", VERDICT round-1 #6;
                # 105.8M -> fused-rate, ~169M at the final hybrid default).
                # The phase pipeline
                # stays in use for serial parity tests and world > 1.
                self._train_step_gpu_fused(tokens, offsets, alpha, window,
                                           n_neg, seed, sent_id_base)
                return
            self._train_step_gpu(tokens, offsets, chunks, alpha, window,
                                 n_neg, seed, sent_id_base)
        else:
            for (a, b) in chunks:
                self._train_chunk_cpu(tokens, offsets, a, b, alpha, window,
                                      n_neg, seed, sent_id_base)

    def _train_step_gpu_fused(self, tokens, offsets, alpha, window, n_neg,
                              seed, sent_id_base):
        """World-1 fast path: one fused sgns_train launch over the full-
        width slice (lo=0, hi=dim, stride == round_stride(dim) — identical
        table layout to ops.gpu.GpuSgns)."""
        assert self.lo == 0 and self.hi == self.dim and not self.narrow
        num_sent = len(offsets) - 1
        nthreads = 256
        nblocks = max(1, min((num_sent + 3) // 4, 2048))
        avg_len = max(1, int(tokens.numel()) // max(num_sent, 1))
        pos_blocks = min(max(1, 8192 // max(num_sent, 1)),
                         (avg_len + 95) // 96)
        pos_blocks = max(1, min(pos_blocks, 11))
        stream = torch.cuda.current_stream(self.device)
        self.native.sgns_train(
            self.syn0.data_ptr(), self.syn1.data_ptr(), int(self.is_bf16),
            self.stride, tokens.data_ptr(), offsets.data_ptr(), num_sent,
            0 if self.keep_thr is None else self.keep_thr.data_ptr(),
            self.table.data_ptr(), int(self.table.numel()), float(alpha),
            int(window), int(n_neg), seed & 0xFFFFFFFFFFFFFFFF,
            int(sent_id_base), int(self.window_mode == "reference"),
            self.atomic_below, self._stats.data_ptr(), nblocks, pos_blocks,
            nthreads, stream.cuda_stream, 0, 0,
            3 if self.stride <= 512 else 0, self.atomic_floor,
            self.shared_neg)

    def _train_step_gpu(self, tokens, offsets, chunks, alpha, window, n_neg,
                        seed, sent_id_base):
        """Pipelined phases: one pair-count pass + host sync per STEP, then
        per chunk {dots -> allreduce(f slice) on a comm stream -> update},
        with chunk k's allreduce overlapping chunk k+1's dots.  The one-
        chunk lookahead (dots k+1 run before update k lands) is absorbed by
        the f-correction."""
        device = self.device
        comp = torch.cuda.current_stream(device)
        if not hasattr(self, "_comm_stream"):
            self._comm_stream = torch.cuda.Stream(device)
        cs = self._comm_stream
        num_sent = len(offsets) - 1
        seed &= 0xFFFFFFFFFFFFFFFF
        ref = int(self.window_mode == "reference")
        kthr = 0 if self.keep_thr is None else self.keep_thr.data_ptr()
        nb_all = 1 if self.serial else max(1, min((num_sent + 3) // 4, 8192))
        nt = 64 if self.serial else 256
        counts = torch.empty(num_sent, dtype=torch.int64, device=device)
        self.native.count_pairs(
            tokens.data_ptr(), offsets.data_ptr(), num_sent, kthr,
            self.table.data_ptr(), int(self.table.numel()), window, n_neg,
            seed, sent_id_base, ref, counts.data_ptr(), nb_all, nt,
            comp.cuda_stream, self.shared_neg)
        poff = torch.zeros(num_sent + 1, dtype=torch.int64, device=device)
        torch.cumsum(counts, 0, out=poff[1:])
        poff_host = poff.cpu().numpy()          # one sync per step
        total = int(poff_host[-1])
        if total == 0:
            return
        f = torch.zeros(total, dtype=torch.float32, device=device)
        f_loc = torch.empty_like(f) if self.f_correction else None
        prev = None   # (chunk, ar_event)

        def launch_update(c):
            n = c[1] - c[0]
            nb = 1 if self.serial else max(1, min((n + 3) // 4, 8192))
            self.native.update_slice(
                self.syn0.data_ptr(), self.syn1.data_ptr(), int(self.is_bf16),
                self.stride, tokens.data_ptr(),
                offsets[c[0]:c[1] + 1].data_ptr(), n, kthr,
                self.table.data_ptr(), int(self.table.numel()), float(alpha),
                window, n_neg, seed, sent_id_base + c[0], ref,
                poff[c[0]:c[1] + 1].data_ptr(), f.data_ptr(),
                0 if f_loc is None else f_loc.data_ptr(),
                float(self.dim) / max(self.width, 1),
                self.atomic_below, self._stats.data_ptr(),
                nb, nt, comp.cuda_stream, 0, 0,
                0 if self.serial else self.pair_mode,
                self.width if self.narrow else 0, self.atomic_floor,
                self.shared_neg)

        for (a, b) in chunks:
            n = b - a
            lo, hi = int(poff_host[a]), int(poff_host[b])
            if hi == lo:
                continue
            nb = 1 if self.serial else max(1, min((n + 3) // 4, 8192))
            self.native.dots_slice(
                self.syn0.data_ptr(), self.syn1.data_ptr(), int(self.is_bf16),
                self.stride, tokens.data_ptr(), offsets[a:b + 1].data_ptr(),
                n, kthr, self.table.data_ptr(), int(self.table.numel()),
                window, n_neg, seed, sent_id_base + a, ref,
                poff[a:b + 1].data_ptr(), f.data_ptr(), nb, nt,
                comp.cuda_stream, 0 if self.serial else 1,
                self.width if self.narrow else 0, self.shared_neg)
            if f_loc is not None:
                f_loc[lo:hi] = f[lo:hi]
            ev = torch.cuda.Event()
            ev.record(comp)
            with torch.cuda.stream(cs):
                cs.wait_event(ev)
                comm.all_reduce_sum_compressed(f[lo:hi])
                ar_ev = torch.cuda.Event()
                ar_ev.record(cs)
            if prev is not None:
                comp.wait_event(prev[1])
                launch_update(prev[0])
            prev = ((a, b), ar_ev)
        if prev is not None:
            comp.wait_event(prev[1])
            launch_update(prev[0])

    def _train_chunk_cpu(self, tokens, offsets, s0, s1, alpha, window, n_neg,
                         seed, sent_id_base):
        n = s1 - s0
        off_view = offsets[s0:s1 + 1]
        base = sent_id_base + s0
        if True:
            tok_np = tokens.numpy()
            off_np = off_view.numpy()
            tab_np = self.table.numpy()
            wm = self.window_mode
            cnts = self.native.count_pairs(
                tok_np, off_np, self.keep_prob, tab_np, window, n_neg,
                seed & 0xFFFFFFFFFFFFFFFF, base, wm,
                shared_negatives=self.shared_neg)
            poff = np.zeros(n + 1, dtype=np.int64)
            np.cumsum(cnts, out=poff[1:])
            total = int(poff[-1])
            if total == 0:
                return
            f = torch.zeros(total, dtype=torch.float32)
            s0_np = self.syn0.numpy()
            s1_np = self.syn1.numpy()
            self.native.dots_slice(s0_np, s1_np, tok_np, off_np,
                                   self.keep_prob, tab_np, window, n_neg,
                                   seed & 0xFFFFFFFFFFFFFFFF, base, wm, poff,
                                   f.numpy(),
                                   shared_negatives=self.shared_neg)
            f_loc = f.numpy().copy() if self.f_correction else None
            comm.all_reduce_sum(f)
            st = self.native.update_slice(s0_np, s1_np, tok_np, off_np,
                                          self.keep_prob, tab_np, float(alpha),
                                          window, n_neg,
                                          seed & 0xFFFFFFFFFFFFFFFF, base, wm,
                                          poff, f.numpy(), f_loc,
                                          float(self.dim) /
                                          max(self.width, 1),
                                          shared_negatives=self.shared_neg)
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

    def save_checkpoint(self, path: str, config, vocab,
                        num_shards: int = 8,
                        block_rows: int = 1 << 20) -> None:
        """Streamed checkpoint: row blocks are assembled from the per-rank
        dim slices (allgather over the world) and written by rank 0 —
        host memory stays O(block_rows * dim) instead of O(vocab * dim)
        (NOTES backlog #5).  Collective: every rank must call this."""
        from ..checkpoint import save_model_streaming
        maxw = max(slice_bounds(self.dim, r, self.world)[1] -
                   slice_bounds(self.dim, r, self.world)[0]
                   for r in range(self.world))

        def row_block(which, r0, r1):
            t = self.syn0 if which == 0 else self.syn1
            my = t[r0:r1, :self.width].float()
            if self.world == 1:
                return my.cpu().numpy()
            padded = torch.zeros((r1 - r0, maxw), dtype=torch.float32,
                                 device=self.device)
            padded[:, :self.width] = my
            gathered = [torch.empty_like(padded) for _ in range(self.world)]
            torch.distributed.all_gather(gathered, padded)
            full = np.empty((r1 - r0, self.dim), dtype=np.float32)
            for r in range(self.world):
                lo, hi = slice_bounds(self.dim, r, self.world)
                full[:, lo:hi] = gathered[r][:, :hi - lo].cpu().numpy()
            return full

        save_model_streaming(path, config, vocab, row_block,
                             num_shards=num_shards, block_rows=block_rows,
                             write=self.rank == 0)
        comm.barrier()

    def to_host(self) -> tuple[np.ndarray, np.ndarray]:
        """Assemble the full [vocab, dim] matrices on every rank (allgather
        of slices; small-model path for save/model ops)."""
        out = []
        for t in (self.syn0, self.syn1):
            my = t[:, :self.width].float()
            if self.world == 1:
                out.append(my.cpu().numpy().copy())
                continue
            maxw = max(slice_bounds(self.dim, r, self.world)[1] -
                       slice_bounds(self.dim, r, self.world)[0]
                       for r in range(self.world))
            padded = torch.zeros((self.vocab_size, maxw), dtype=torch.float32,
                                 device=self.device)
            padded[:, :self.width] = my
            gathered = [torch.empty_like(padded) for _ in range(self.world)]
            torch.distributed.all_gather(gathered, padded)
            full = np.empty((self.vocab_size, self.dim), dtype=np.float32)
            for r in range(self.world):
                lo, hi = slice_bounds(self.dim, r, self.world)
                full[:, lo:hi] = gathered[r][:, :hi - lo].cpu().numpy()
            out.append(full)
        return out[0], out[1]
