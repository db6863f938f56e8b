"""Sharded model serving — the reference's distributed-model load+serve
path (mllib:683-726: ``load`` restarts the PS cluster and loads shards
SERVER-side, keeping the model sharded; findSynonyms/transform then run
against the sharded state, mllib:598/486, ml:453).

Here the GPUs are the servers: ``ShardedWord2VecModel.load`` streams each
rank's rows (row r lives on rank r % world) straight from the checkpoint
shard files into HBM — host memory stays O(block), never O(vocab*dim) —
and the model ops run as per-shard kernels + small collectives:

  findSynonyms  = per-shard GEMM scores (rocBLAS) + local top-k
                  -> allgather of k candidates -> merge     (mllib:598)
  norms         = per-shard row-norm kernel, cached         (mllib:486)
  transform     = per-shard partial sentence sums -> allreduce (ml:453)

Word <-> index lookups come from the mmap'd C++ ``WordFileIndex`` (one
word per line, line number == row index — the reference's words-file
layout, mllib:714-715), ~24 bytes/word instead of the reference's 8 GB
driver-side broadcast map (README.md:71-73).

Runs at any world size (world 1 = single-GPU serving) and on CPU under
gloo for tests.  Every rank calls every op collectively and every rank
returns the full result (the natural shape for a replicated-frontend
serving tier).
"""
from __future__ import annotations

import json
import logging
import os
from typing import Iterable, List, Optional, Sequence, Union

import numpy as np
import torch

from . import _cpu_native
from .config import Word2VecConfig
from .parallel import comm

log = logging.getLogger("glint_word2vec_amd")


class LazyVocab:
    """Vocabulary view over the checkpoint's words file (mmap'd C++ index):
    word->row and row->word without materialising Python strings."""

    def __init__(self, words_path: str):
        self._ix = _cpu_native.WordFileIndex(words_path)

    @property
    def num_words(self) -> int:
        return len(self._ix)

    def __contains__(self, word: str) -> bool:
        return self._ix.lookup(word) >= 0

    def __getitem__(self, word: str) -> int:
        i = self._ix.lookup(word)
        if i < 0:
            raise KeyError(word)
        return i

    def get(self, word: str, default: int = -1) -> int:
        i = self._ix.lookup(word)
        return i if i >= 0 else default

    def word(self, i: int) -> str:
        return self._ix.word(i)

    def lookup_many(self, words: Sequence[str]) -> np.ndarray:
        return self._ix.lookup_many(list(words))


def _shard_row_iter(index: dict, shard: int):
    """Yield (global_row_of_first, rows_in_shard) addressing for a shard
    file: row j of shard s is global ``s + j*k`` (row_mod) or
    ``bounds[s] + j`` (row_range)."""
    k = index["num_shards"]
    if index.get("layout", "row_mod") == "row_mod":
        first = shard
        count = (index["vocab"] - shard + k - 1) // k
        stride = k
    else:
        b = index["bounds"]
        first = b[shard]
        count = b[shard + 1] - b[shard]
        stride = 1
    return first, count, stride


class ShardedWord2VecModel:
    """Distributed (or single-GPU) serving model over a checkpoint.

    ``load`` is collective: every rank streams its own rows of syn0 from
    whatever shard layout/count the checkpoint was written with.  syn1 is
    not loaded (serving needs only the input embeddings, like the
    reference's model ops)."""

    def __init__(self, config: Word2VecConfig, vocab, shard: torch.Tensor,
                 rank: int, world: int, dim: int):
        self.config = config
        self.vocab = vocab
        self.shard = shard            # [shard_size, dim] on the device
        self.rank = rank
        self.world = world
        self.dim = dim
        self.device = shard.device
        self.is_cuda = shard.device.type == "cuda"
        self._norms: Optional[torch.Tensor] = None   # local shard norms

    # ------------------------------------------------------------------
    @classmethod
    def load(cls, path: str, device: str = "auto", dtype: str = "auto",
             block_rows: int = 1 << 20) -> "ShardedWord2VecModel":
        """Streaming sharded load (the PS-side load, mllib:717-722).
        Host RSS stays O(block_rows*dim) + the word index (~24 B/word)."""
        rank, world = comm.init_from_env()
        if device == "auto":
            device = "cuda" if torch.cuda.is_available() else "cpu"
        if device.startswith("cuda") and torch.cuda.is_available():
            device = f"cuda:{torch.cuda.current_device()}"
        with open(os.path.join(path, "metadata")) as f:
            meta = json.load(f)
        config = Word2VecConfig.from_dict(meta["paramMap"])
        with open(os.path.join(path, "shards", "index.json")) as f:
            index = json.load(f)
        V, D = index["vocab"], index["dim"]
        tdtype = (torch.bfloat16 if (dtype == "bfloat16" or (
            dtype == "auto" and device.startswith("cuda")))
            else torch.float32)
        # "bfloat16" checkpoints are stored as raw u16 (checkpoint.py
        # _DTYPES): half the disk/IO of f32 — the natural format for
        # 100 GB-class models
        is_bf16_file = index["dtype"] == "bfloat16"
        file_dt = np.uint16 if is_bf16_file else np.dtype(index["dtype"])
        shard_size = (V - rank + world - 1) // world
        shard = torch.empty((shard_size, D), dtype=tdtype,
                            device=torch.device(device))
        for s in range(index["num_shards"]):
            first, count, stride = _shard_row_iter(index, s)
            fp = os.path.join(path, "shards", f"syn0-{s:05d}.bin")
            with open(fp, "rb") as f:
                for j0 in range(0, count, block_rows):
                    j1 = min(count, j0 + block_rows)
                    buf = np.fromfile(f, dtype=file_dt,
                                      count=(j1 - j0) * D)
                    block = buf.reshape(j1 - j0, D)
                    # global ids of this block, then the ones this rank owns
                    gids = first + np.arange(j0, j1, dtype=np.int64) * stride
                    mine = (gids % world) == rank
                    if not mine.any():
                        continue
                    picked = np.ascontiguousarray(block[mine])
                    rows = (torch.from_numpy(picked).view(torch.bfloat16)
                            .float() if is_bf16_file else
                            torch.from_numpy(picked.astype(np.float32,
                                                           copy=False)))
                    shard[torch.from_numpy(gids[mine] // world)] = \
                        rows.to(tdtype).to(shard.device)
        vocab = LazyVocab(os.path.join(path, "words"))
        if vocab.num_words != V:
            raise ValueError(f"words file has {vocab.num_words} entries, "
                             f"index says {V}")
        m = cls(config, vocab, shard, rank, world, D)
        log.info("sharded load: rank %d/%d holds %d of %d rows (%s, %s)",
                 rank, world, shard_size, V, device, tdtype)
        return m

    # ------------------------------------------------------------------
    @property
    def num_words(self) -> int:
        return self.vocab.num_words

    @property
    def vector_size(self) -> int:
        return self.dim

    def _local_norms(self) -> torch.Tensor:
        """Lazily cached per-shard row norms (mllib:486)."""
        if self._norms is None:
            self._norms = self.shard.float().norm(dim=1)
        return self._norms

    def _global_ids(self, local: torch.Tensor) -> torch.Tensor:
        return local * self.world + self.rank

    def get_vector(self, word_or_id: Union[str, int]) -> np.ndarray:
        """Row pull from its owner + broadcast (the Glint single-row pull,
        mllib:514)."""
        idx = (self.vocab[word_or_id] if isinstance(word_or_id, str)
               else int(word_or_id))
        owner = idx % self.world
        v = torch.zeros(self.dim, dtype=torch.float32, device=self.device)
        if self.rank == owner:
            v.copy_(self.shard[idx // self.world].float())
        comm.broadcast_(v, src=owner)
        return v.cpu().numpy()

    transform_word = get_vector

    def transform_words(self, words: Iterable[str]) -> np.ndarray:
        """Batched word lookup (mllib:529-543): owner-partial scatter +
        allreduce."""
        ids = [self.vocab[w] for w in words]
        out = torch.zeros((len(ids), self.dim), dtype=torch.float32,
                          device=self.device)
        idx = torch.tensor(ids, dtype=torch.long, device=self.device)
        mine = (idx % self.world) == self.rank
        if bool(mine.any()):
            out[mine] = self.shard[idx[mine] // self.world].float()
        comm.all_reduce_sum(out)
        return out.cpu().numpy()

    def transform_sentences(self, sentences: Sequence[Sequence[str]]
                            ) -> np.ndarray:
        """Sentence-average transform (pullAverage, ml:453): each rank sums
        the rows it owns per sentence, one allreduce merges."""
        sums = torch.zeros((len(sentences), self.dim), dtype=torch.float32,
                           device=self.device)
        # one batched word->id lookup over the whole request (C++ mmap
        # index), then owner-partial sums
        flat: List[str] = []
        sent_of = np.empty(sum(len(x) for x in sentences), dtype=np.int64)
        pos = 0
        for si, sent in enumerate(sentences):
            flat.extend(sent)
            sent_of[pos:pos + len(sent)] = si
            pos += len(sent)
        ids = (self.vocab.lookup_many(flat)
               if hasattr(self.vocab, "lookup_many") else
               np.array([self.vocab.get(w) for w in flat], dtype=np.int64))
        known = ids >= 0
        counts = np.bincount(sent_of[known],
                             minlength=len(sentences)).astype(np.float32)
        mine = known & (ids % self.world == self.rank)
        if mine.any():
            rows = self.shard[torch.from_numpy(ids[mine] // self.world)
                              .to(self.device)].float()
            sums.index_add_(0, torch.from_numpy(sent_of[mine])
                            .to(self.device), rows)
        comm.all_reduce_sum(sums)
        out = sums.cpu().numpy()
        nz = counts > 0
        out[nz] /= counts[nz, None]
        return out

    # alias matching GlintWord2VecModel's GPU serving name
    transform_sentences_gpu = transform_sentences

    def transform(self, x, input_col: str = "sentence",
                  output_col: str = "vector"):
        """Drop-in transform (the dense model's polymorphic surface,
        ml:432-460): word -> vector; token sequence -> average vector;
        DataFrame -> copy with an output column of averaged vectors."""
        if isinstance(x, str):
            return self.get_vector(x)
        if hasattr(x, "columns"):   # DataFrame
            col = input_col if input_col in x.columns else x.columns[0]
            out = x.copy()
            out[output_col] = list(self.transform_sentences(list(x[col])))
            return out
        return self.transform_sentences([list(x)])[0]

    def _query_vec(self, word_or_vec) -> tuple:
        if isinstance(word_or_vec, str):
            vec = self.get_vector(word_or_vec)
            return word_or_vec, vec
        return None, np.asarray(word_or_vec, dtype=np.float32)

    def find_synonyms(self, word_or_vec, num: int) -> List[tuple]:
        """Top-``num`` cosine neighbours: sharded GEMV + local top-k +
        allgather merge (mllib:554-630; the multiply op at :598)."""
        return self.find_synonyms_batch([word_or_vec], num)[0]

    def find_synonyms_batch(self, queries: Sequence, num: int,
                            max_score_bytes: int = 4 << 30
                            ) -> List[List[tuple]]:
        """Batched multi-query findSynonyms: sharded GEMM + topk per
        shard + one allgather (serving throughput path;
        benchmarks/serving_probe.py).  Internally chunks the query batch
        so the [q, shard] f32 score tensor stays under
        ``max_score_bytes`` (an 80M-row shard at Q=4096 would otherwise
        materialise 1.3 TB)."""
        shard_rows = max(int(self.shard.shape[0]), 1)
        q_chunk = max(1, int(max_score_bytes // (shard_rows * 4)))
        if len(queries) > q_chunk:
            out: List[List[tuple]] = []
            for i in range(0, len(queries), q_chunk):
                out.extend(self.find_synonyms_batch(
                    queries[i:i + q_chunk], num, max_score_bytes))
            return out
        Q = len(queries)
        skip_words = []
        vecs = np.empty((Q, self.dim), dtype=np.float32)
        for qi, q in enumerate(queries):
            w, v = self._query_vec(q)
            skip_words.append(w)
            n = np.linalg.norm(v)
            vecs[qi] = v / n if n > 0 else v
        qs = torch.from_numpy(vecs).to(self.shard.dtype).to(self.device)
        scores = (qs @ self.shard.T).float()            # [Q, shard]
        norms = self._local_norms()
        cos = torch.where(norms[None, :] > 0, scores / norms[None, :],
                          torch.zeros((), device=self.device))
        k = min(num + 1, cos.shape[1])
        val, loc = torch.topk(cos, k, dim=1)            # [Q, k]
        val = val.T.contiguous()                        # [k, Q] for gather
        gid = self._global_ids(loc.long()).T.contiguous()
        # pad to a fixed k across ranks (tiny shards), then allgather
        kmax = min(num + 1, self.num_words)
        if k < kmax:
            pad = torch.full((kmax - k, Q), -2.0, device=self.device)
            val = torch.cat([val, pad])
            gid = torch.cat([gid, torch.zeros_like(pad, dtype=torch.long)])
        if self.world > 1:
            gval = [torch.empty_like(val) for _ in range(self.world)]
            ggid = [torch.empty_like(gid) for _ in range(self.world)]
            torch.distributed.all_gather(gval, val.contiguous())
            torch.distributed.all_gather(ggid, gid.contiguous())
            val = torch.cat(gval)                        # [world*kmax, Q]
            gid = torch.cat(ggid)
        val = val.cpu().numpy()
        gid = gid.cpu().numpy()
        results: List[List[tuple]] = []
        for qi in range(Q):
            order = np.argsort(-val[:, qi])
            out = []
            for j in order:
                if val[j, qi] <= -2.0:
                    continue
                w = self.vocab.word(int(gid[j, qi]))
                if w == skip_words[qi]:
                    continue
                out.append((w, float(val[j, qi])))
                if len(out) == num:
                    break
            results.append(out)
        return results

    def find_synonyms_df(self, word_or_vec, num: int):
        """DataFrame variant (ml:390-420 parity, like the dense model)."""
        import pandas as pd
        return pd.DataFrame(self.find_synonyms(word_or_vec, num),
                            columns=["word", "similarity"])

    def analogy(self, pos: List[str], neg: List[str],
                num: int = 10) -> List[tuple]:
        v = np.zeros(self.dim, dtype=np.float32)
        for w in pos:
            v += self.get_vector(w)
        for w in neg:
            v -= self.get_vector(w)
        skip = set(pos) | set(neg)
        res = self.find_synonyms(v, num + len(skip))
        return [(w, c) for w, c in res if w not in skip][:num]

    def to_local(self, max_bytes: int = 16 << 30):
        """Assemble a host LocalWord2VecModel (the reference's toLocal,
        mllib:651-659).  Deliberately guarded: materialising an 80M-vocab
        matrix needs ~96 GB of host RAM — raise ``max_bytes`` explicitly
        for giant models."""
        need = self.num_words * self.dim * 4
        if need > max_bytes:
            raise ValueError(
                f"toLocal would materialise {need / 2**30:.1f} GB on host; "
                f"pass max_bytes >= {need} to allow it, or use "
                "export_text() for a streaming export")
        from .estimator import LocalWord2VecModel
        full = np.empty((self.num_words, self.dim), dtype=np.float32)
        block = 1 << 20
        for r0 in range(0, self.num_words, block):
            r1 = min(self.num_words, r0 + block)
            full[r0:r1] = self._pull_range(r0, r1)
        words = [self.vocab.word(i) for i in range(self.num_words)]
        return LocalWord2VecModel(words, full)

    def _pull_range(self, r0: int, r1: int) -> np.ndarray:
        """Rows [r0, r1) assembled from their owners (partial + allreduce
        merge, O(block) at a time)."""
        ids = torch.arange(r0, r1, dtype=torch.long, device=self.device)
        out = torch.zeros((r1 - r0, self.dim), dtype=torch.float32,
                          device=self.device)
        mine = (ids % self.world) == self.rank
        if bool(mine.any()):
            out[mine] = self.shard[ids[mine] // self.world].float()
        comm.all_reduce_sum(out)
        return out.cpu().numpy()

    def export_text(self, path: str, block: int = 1 << 18) -> None:
        """Streaming word2vec-text export (toLocal().save without the host
        matrix): rank 0 writes; every rank participates in the pulls."""
        f = open(path, "w", encoding="utf-8") if self.rank == 0 else None
        try:
            if f:
                f.write(f"{self.num_words} {self.dim}\n")
            for r0 in range(0, self.num_words, block):
                r1 = min(self.num_words, r0 + block)
                rows = self._pull_range(r0, r1)
                if f:
                    for i in range(r0, r1):
                        v = rows[i - r0]
                        f.write(self.vocab.word(i) + " "
                                + " ".join(f"{x:.6g}" for x in v) + "\n")
        finally:
            if f:
                f.close()
        comm.barrier()

    def save(self, path: str, num_shards: int = 8,
             block: int = 1 << 20) -> None:
        """Re-save the (serving, syn0-only) model as a checkpoint —
        streaming, rank 0 writes (the reference's matrix.save,
        mllib:493-498).  Collective."""
        from .checkpoint import save_model_streaming

        class _V:
            num_words = self.num_words
            counts = None

            @staticmethod
            def save_words(p):
                with open(p, "w", encoding="utf-8") as f:
                    for i in range(self.num_words):
                        f.write(self.vocab.word(i) + "\n")

        def row_block(which, r0, r1):
            return self._pull_range(r0, r1)

        save_model_streaming(path, self.config, _V, row_block,
                             num_shards=num_shards, block_rows=block,
                             has_syn1=False, write=self.rank == 0)
        comm.barrier()

    def stop(self, terminate_other_clients: bool = False) -> None:
        try:
            import torch.distributed as dist
            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass
