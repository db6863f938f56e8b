"""GPU-resident SGNS state + launch wrappers around the fused HIP kernels.

The embedding tables live in HBM as [vocab, stride] tensors (stride = dim
rounded up to the kernel's supported chunk counts); padding columns are
zero-initialised and provably stay zero (every update scales existing row
values), so dots over the padded width are exact.

On a machine with a GPU the HIP extension is REQUIRED: there is no silent
eager/torch fallback — a missing or broken _hip_native raises immediately.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from ..vocab import keep_probabilities


def _load_native():
    try:
        from .. import _hip_native
        return _hip_native
    except ImportError as e:
        raise RuntimeError(
            "glint_word2vec_amd._hip_native is not built but a GPU path was "
            "requested. Build it in-tree with `python setup.py build_ext "
            "--inplace` (hipcc, gfx950). Refusing to fall back to eager."
        ) from e


def round_stride(dim: int) -> int:
    return _load_native().round_stride(dim)


@dataclass
class GpuStats:
    pairs: int
    positives: int
    words_trained: int
    sum_fplus: float


class GpuSgns:
    """Owns the device tables and launches the fused kernels.

    The full-table (non-sharded) case: both matrices resident on one GPU —
    BASELINE.json config 2.  The sharded engine (parallel/engine.py) holds
    one GpuSgns per rank over its row shard plus staging buffers.
    """

    def __init__(self, vocab_size: int, dim: int, dtype: str = "float32",
                 device: str = "cuda", seed: int = 1,
                 init: str = "word2vec",
                 syn0_host: Optional[np.ndarray] = None,
                 syn1_host: Optional[np.ndarray] = None):
        if dtype not in ("float32", "bfloat16"):
            raise ValueError(f"GpuSgns needs a concrete dtype, got {dtype!r}")
        self.native = _load_native()
        self.device = torch.device(device)
        self.vocab_size = vocab_size
        self.dim = dim
        self.stride = self.native.round_stride(dim)
        self.is_bf16 = dtype == "bfloat16"
        tdtype = torch.bfloat16 if self.is_bf16 else torch.float32
        self.syn0 = torch.zeros((vocab_size, self.stride), dtype=tdtype,
                                device=self.device)
        self.syn1 = torch.zeros((vocab_size, self.stride), dtype=tdtype,
                                device=self.device)
        if syn0_host is not None:
            self.load_host(syn0_host, syn1_host)
        elif init == "word2vec":
            # blockwise so 80M-vocab tables never materialise on host
            g = torch.Generator(device="cpu").manual_seed(seed)
            block = max(1, (1 << 28) // max(dim, 1))
            for r0 in range(0, vocab_size, block):
                r1 = min(vocab_size, r0 + block)
                w = (torch.rand((r1 - r0, dim), generator=g) - 0.5) / dim
                self.syn0[r0:r1, :dim] = w.to(tdtype).to(self.device)
        # stats buffer: [pairs u64, positives u64, words u64, sum_fplus f64]
        self._stats = torch.zeros(4, dtype=torch.int64, device=self.device)
        self.keep_thr: Optional[torch.Tensor] = None
        self.table: Optional[torch.Tensor] = None
        self.exp_table: Optional[torch.Tensor] = None

    # --- state management --------------------------------------------------
    def load_host(self, syn0: np.ndarray, syn1: Optional[np.ndarray]) -> None:
        tdtype = self.syn0.dtype
        s0 = torch.from_numpy(np.ascontiguousarray(syn0, dtype=np.float32))
        self.syn0[:, :self.dim] = s0.to(tdtype).to(self.device)
        if syn1 is not None:
            s1 = torch.from_numpy(np.ascontiguousarray(syn1, dtype=np.float32))
            self.syn1[:, :self.dim] = s1.to(tdtype).to(self.device)

    def to_host(self) -> tuple[np.ndarray, np.ndarray]:
        s0 = self.syn0[:, :self.dim].float().cpu().numpy()
        s1 = self.syn1[:, :self.dim].float().cpu().numpy()
        return s0, s1

    def set_subsample(self, counts: np.ndarray, train_words: int,
                      ratio: float) -> None:
        if ratio <= 0:
            self.keep_thr = None
            return
        kp = keep_probabilities(counts, train_words, ratio).astype(np.float64)
        thr = np.minimum(kp * 4294967296.0, 4294967295.0).astype(np.uint32)
        self.keep_thr = torch.from_numpy(thr.view(np.int32)).to(self.device)

    def set_sigmoid_lut(self, table: Optional[np.ndarray]) -> None:
        """Reference getSigmoid parity mode (EXP_TABLE_SIZE-entry LUT)."""
        self.exp_table = (None if table is None else
                          torch.from_numpy(np.ascontiguousarray(
                              table, dtype=np.float32)).to(self.device))

    def set_table(self, table: np.ndarray) -> None:
        self.table = torch.from_numpy(np.ascontiguousarray(table, dtype=np.int32)) \
            .to(self.device)

    # --- ops ---------------------------------------------------------------
    def train_batch(self, tokens: torch.Tensor, offsets: torch.Tensor,
                    alpha: float, window: int, n_neg: int, seed: int,
                    sent_id_base: int = 0, window_mode: str = "canonical",
                    atomic: bool = True, atomic_below: Optional[int] = None,
                    atomic_floor: int = 0, shared_negatives: bool = False,
                    blocks: Optional[int] = None,
                    serial: bool = False,
                    pair_mode: Optional[int] = None,
                    stream: Optional[torch.cuda.Stream] = None) -> None:
        """Launch the fused train kernel (async on the given/current stream).
        Stats accumulate on-device; read with read_stats().
        atomic_below: None = all rows atomic; K > 0 = rows < K only;
        -1 = positive pairs (+ centers) only — the quality-critical 1/6 of
        update traffic (see benchmarks/quality_probe.py results)."""
        assert self.table is not None, "call set_table first"
        if pair_mode is None:
            # two pairs per wave (32-lane halves) wins up to dim ~512; at
            # NC > 8 the doubled per-wave registers (c_row/grad/t_row of
            # 2*NC floats each) collapse occupancy — one 64-lane pair per
            # wave runs 2.5x faster at dim 1024 (benchmarks/results.md
            # round-2 A/B: 15.0M vs 5.9M words/s at dim=1024 neg=25)
            # 3 = two pairs/wave with 2-deep block pipelining (round-2
            # A/B: 169.7->173.4M hybrid, 138.8->153.5M at vocab 10M — the
            # atomic/store traffic no longer clogs the next load's vmcnt)
            pair_mode = 3 if self.stride <= 512 else 0
        num_sent = int(offsets.numel() - 1)
        if num_sent <= 0:
            return
        if serial:
            nblocks, nthreads, pos_blocks = 1, 64, 1   # one wave: oracle order
        else:
            nthreads = 256
            if blocks is None:
                # 4 waves per block; oversubscribe the 256 CUs
                nblocks = max(1, min((num_sent + 3) // 4, 2048))
            else:
                nblocks = blocks
            # long-sentence batches: split positions over grid.y so ~8k
            # waves stay in flight regardless of sentence count
            avg_len = max(1, int(tokens.numel()) // num_sent)
            pos_blocks = min(max(1, 8192 // max(num_sent, 1)),
                             (avg_len + 95) // 96)
            pos_blocks = max(1, min(pos_blocks, 11))
        s = stream if stream is not None else torch.cuda.current_stream(self.device)
        self.native.sgns_train(
            self.syn0.data_ptr(), self.syn1.data_ptr(), int(self.is_bf16),
            self.stride, tokens.data_ptr(), offsets.data_ptr(), num_sent,
            0 if self.keep_thr is None else self.keep_thr.data_ptr(),
            self.table.data_ptr(), int(self.table.numel()), float(alpha),
            int(window), int(n_neg), seed & 0xFFFFFFFFFFFFFFFF,
            int(sent_id_base), int(window_mode == "reference"),
            0 if not atomic else (2 ** 31 - 1 if atomic_below is None
                                  else int(atomic_below)),
            self._stats.data_ptr(), nblocks, pos_blocks, nthreads,
            s.cuda_stream,
            0 if self.exp_table is None else self.exp_table.data_ptr(),
            0 if self.exp_table is None else int(self.exp_table.numel()),
            0 if serial else int(pair_mode), int(atomic_floor),
            int(shared_negatives))

    def read_stats(self, reset: bool = True) -> GpuStats:
        h = self._stats.cpu()
        pairs, pos, words = int(h[0]), int(h[1]), int(h[2])
        sum_fplus = float(h[3:4].view(torch.float64)[0])
        if reset:
            self._stats.zero_()
        return GpuStats(pairs, pos, words, sum_fplus)

    def pull_average(self, tokens: torch.Tensor, offsets: torch.Tensor) -> torch.Tensor:
        num_sent = int(offsets.numel() - 1)
        out = torch.empty((num_sent, self.stride), dtype=torch.float32,
                          device=self.device)
        s = torch.cuda.current_stream(self.device)
        self.native.pull_average(
            self.syn0.data_ptr(), int(self.is_bf16), self.stride,
            tokens.data_ptr(), offsets.data_ptr(), num_sent, out.data_ptr(),
            max(1, min((num_sent + 3) // 4, 8192)), s.cuda_stream)
        return out[:, :self.dim]

    def norms(self) -> torch.Tensor:
        out = torch.empty(self.vocab_size, dtype=torch.float32, device=self.device)
        s = torch.cuda.current_stream(self.device)
        self.native.norms(self.syn0.data_ptr(), int(self.is_bf16),
                          self.vocab_size, self.stride, out.data_ptr(),
                          2048, s.cuda_stream)
        return out

    def save_checkpoint(self, path: str, config, vocab,
                        num_shards: int = 8) -> None:
        """Stream the tables from HBM to a checkpoint without materialising
        them on host (80M-vocab models are ~100 GB on-device)."""
        from ..checkpoint import save_model_streaming

        def row_block(which, r0, r1):
            t = self.syn0 if which == 0 else self.syn1
            return t[r0:r1, :self.dim].float().cpu().numpy()

        save_model_streaming(path, config, vocab, row_block,
                             num_shards=num_shards)

    def multiply(self, vec: torch.Tensor, norms: Optional[torch.Tensor]
                 = None) -> torch.Tensor:
        """Whole-table GEMV scores (Glint multiply, mllib:598) — a
        hand-written wave-per-row kernel: rocBLAS bf16 GEMV sustains only
        ~1.6 TB/s on this tall-skinny shape, the kernel streams the table
        at HBM rate (~5x, benchmarks/serving_probe.py).  ``norms`` (>0,
        pre-clamped) fuses the cosine divide."""
        q = torch.zeros(self.stride, dtype=torch.float32, device=self.device)
        q[:self.dim] = vec.float().to(self.device)
        out = torch.empty(self.vocab_size, dtype=torch.float32,
                          device=self.device)
        st = torch.cuda.current_stream(self.device)
        self.native.scores(self.syn0.data_ptr(), int(self.is_bf16),
                           self.vocab_size, self.stride, q.data_ptr(),
                           0 if norms is None else norms.data_ptr(),
                           out.data_ptr(), 2048, st.cuda_stream)
        self._scores_args = (q, out) if norms is None else (q, out, norms)
        return out

    def synonyms_query(self, vec: torch.Tensor, k: int):
        """Single-query findSynonyms core as a replayed hipGraph: the
        GEMV + normalise + top-k chain is launch-latency-bound at ~2.1k
        q/s when enqueued kernel by kernel; capturing it once and
        replaying turns per-query host work into one graph launch
        (guide: hipGraphs for launch-bound inner loops).  Returns
        (cos_values[k], row_indices[k]) device tensors — valid until the
        next call."""
        if getattr(self, "_syn_graph_k", None) != k:
            self._syn_norms = self.norms().clamp_min(1e-12)
            self._g_in = torch.zeros(1, self.dim, dtype=torch.float32,
                                     device=self.device)
            side = torch.cuda.Stream(self.device)
            side.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(side):
                for _ in range(2):   # warmup allocations before capture
                    cos = self.multiply(self._g_in[0], norms=self._syn_norms)
                    torch.topk(cos, k)
            torch.cuda.current_stream(self.device).wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                cos = self.multiply(self._g_in[0], norms=self._syn_norms)
                self._g_val, self._g_idx = torch.topk(cos, k)
            self._syn_graph = g
            self._syn_graph_k = k
        self._g_in.copy_(vec.reshape(1, -1).to(torch.float32))
        self._syn_graph.replay()
        return self._g_val, self._g_idx

    def multiply_batch(self, vecs: torch.Tensor) -> torch.Tensor:
        """Multi-query scores: one rocBLAS GEMM [Q, stride] x
        [stride, vocab] -> [Q, vocab] — the batched-serving findSynonyms
        path.  The [Q, vocab] layout matters: the follow-up top-k then
        reduces along the CONTIGUOUS last dim (topk over dim 0 of a
        [vocab, Q] tensor ran 30x slower at Q=4096,
        benchmarks/serving_probe.py)."""
        q = torch.zeros((vecs.shape[0], self.stride), dtype=self.syn0.dtype,
                        device=self.device)
        q[:, :self.dim] = vecs.to(self.syn0.dtype).to(self.device)
        return (q @ self.syn0.T).float()
