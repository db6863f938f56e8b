"""User-facing API: ``GlintWord2Vec`` (estimator) and ``GlintWord2VecModel``.

Mirrors the reference's Spark-ML surface (ml.feature.ServerSideGlintWord2Vec
fit/transform/findSynonyms/getVectors/save/load/toLocal/stop — ml:284-497,
mllib:460-726) without Spark: corpora are paths, iterables of token lists,
or pandas DataFrames with an ``input_col`` of token lists.  The reference's
fluent ``setX`` setters are provided as aliases.
"""
from __future__ import annotations

import logging
import time
from typing import Iterable, List, Optional, Sequence, Union

import numpy as np

from .config import Word2VecConfig
from .checkpoint import load_model, save_model, save_word2vec_text
from .data import batch_sentences, read_text_corpus
from .models import sgns
from .vocab import (Vocabulary, build_unigram_table, build_vocab,
                    encode_sentences, keep_probabilities)

log = logging.getLogger("glint_word2vec_amd")

CorpusLike = Union[str, Iterable[Sequence[str]]]


def _corpus_reader(corpus: CorpusLike, input_col: Optional[str]):
    """Return a zero-arg callable yielding token sequences (re-iterable)."""
    if isinstance(corpus, str):
        return lambda: read_text_corpus(corpus)
    try:
        import pandas as pd  # noqa
        if hasattr(corpus, "columns") and input_col is not None:
            col = corpus[input_col]
            return lambda: iter(col.tolist())
    except ImportError:
        pass
    if hasattr(corpus, "__iter__") and not hasattr(corpus, "__next__"):
        return lambda: iter(corpus)
    # one-shot iterator: materialise
    sents = [list(s) for s in corpus]
    return lambda: iter(sents)


class GlintWord2Vec:
    """Estimator.  Keyword args are Word2VecConfig fields; reference-style
    fluent setters are also available (setVectorSize, setStepSize, ...)."""

    def __init__(self, input_col: str = "sentence", output_col: str = "vector",
                 config: Optional[Word2VecConfig] = None, **kwargs):
        self.input_col = input_col
        self.output_col = output_col
        self.config = config if config is not None else Word2VecConfig(**kwargs)

    # --- reference-parity fluent setters (ml:40-222) ----------------------
    def _set(self, **kw) -> "GlintWord2Vec":
        for k, v in kw.items():
            setattr(self.config, k, v)
        self.config.validate()
        return self

    def setVectorSize(self, v):        return self._set(vector_size=v)
    def setStepSize(self, v):          return self._set(learning_rate=v)
    def setLearningRate(self, v):      return self._set(learning_rate=v)
    def setNumPartitions(self, v):     return self._set(num_partitions=v)
    def setNumIterations(self, v):     return self._set(num_iterations=v)
    def setMaxIter(self, v):           return self._set(num_iterations=v)
    def setSeed(self, v):              return self._set(seed=v)
    def setMinCount(self, v):          return self._set(min_count=v)
    def setMaxSentenceLength(self, v): return self._set(max_sentence_length=v)
    def setWindowSize(self, v):        return self._set(window=v)
    def setBatchSize(self, v):         return self._set(batch_size=v)
    def setN(self, v):                 return self._set(n=v)
    def setSubsampleRatio(self, v):    return self._set(subsample_ratio=v)
    def setUnigramTableSize(self, v):  return self._set(unigram_table_size=v)
    def setCheckpointEvery(self, v):   return self._set(checkpoint_every=v)
    def setNumParameterServers(self, v):
        # PS count maps to GPU shard count in this framework.
        return self._set(num_shards=v)

    def setParameterServerHost(self, v):
        # No separate PS cluster exists: the GPUs of this node are the
        # servers.  Accepted for drop-in compatibility.
        if v:
            log.warning("setParameterServerHost(%r) ignored: parameter "
                        "servers are the local GPUs; launch under torchrun "
                        "for multi-GPU", v)
        return self

    def setParameterServerConfig(self, v):
        if v:
            log.warning("setParameterServerConfig ignored (no Akka/HOCON "
                        "tier); use Word2VecConfig device/engine fields")
        return self

    def setInputCol(self, v):
        self.input_col = v
        return self

    def setOutputCol(self, v):
        self.output_col = v
        return self

    # ----------------------------------------------------------------------
    def fit(self, corpus: CorpusLike, save_path: Optional[str] = None,
            materialize: bool = True,
            init_from: Optional[str] = None) -> "GlintWord2VecModel":
        """Train.  ``save_path``: checkpoint straight from the engine
        (streamed from HBM / per-rank shards — required path for 80M-vocab
        models whose matrices must not materialise on host).
        ``materialize=False``: skip host assembly and return None (load
        later with GlintWord2VecModel.load).
        ``init_from``: resume training from a saved checkpoint — its vocab
        and matrices seed the run; the corpus is encoded against the
        checkpoint's vocabulary (continue training on new data)."""
        cfg = self.config
        seed = cfg.seed if cfg.seed is not None else np.random.SeedSequence().entropy % (2 ** 63)
        seed = int(seed)
        init_tables = None
        if init_from is not None:
            from .checkpoint import load_model
            ck_cfg, vocab, ck0, ck1 = load_model(init_from)
            if ck1 is None:
                raise ValueError("checkpoint has no syn1 — cannot resume "
                                 "training (saved with has_syn1=False)")
            if ck0.shape[1] != cfg.vector_size:
                raise ValueError(
                    f"checkpoint dim {ck0.shape[1]} != vector_size "
                    f"{cfg.vector_size}")
            init_tables = (ck0, ck1)
            reader = _corpus_reader(corpus, self.input_col)
            native = None
            if isinstance(corpus, str):
                try:
                    from . import _cpu_native as native
                except ImportError:
                    native = None
        else:
            native = None
            if isinstance(corpus, str):
                try:
                    from . import _cpu_native as native
                except ImportError:
                    native = None
            if native is not None:
                # native corpus pipeline (csrc/cpu_sgns.cpp): count/sort/
                # encode in C++ — same semantics as the Python path (tested)
                words, counts, total = native.build_vocab_file(corpus,
                                                               cfg.min_count)
                vocab = Vocabulary(words=list(words), counts=counts,
                                   index={w: i for i, w in enumerate(words)},
                                   train_words_count=int(total))
                reader = _corpus_reader(corpus, self.input_col)
            else:
                reader = _corpus_reader(corpus, self.input_col)
                vocab = build_vocab(reader(), min_count=cfg.min_count)
        if vocab.num_words == 0:
            raise ValueError("empty vocabulary — corpus has no word above min_count")
        log.info("vocab: %d words, %d train words", vocab.num_words, vocab.train_words_count)

        max_sent = min(cfg.max_sentence_length, 1024)
        if cfg.max_sentence_length > 1024:
            # the GPU walker compacts each sentence into a 1024-entry LDS
            # buffer (csrc/hip kMaxSent; 4 waves x 4 KB LDS per block), so
            # longer sentences are chunked at 1024 — the same semantics the
            # reference applies at its default maxSentenceLength=1000
            # (mllib:88-97), just at a different boundary.
            log.warning(
                "max_sentence_length=%d exceeds the kernel sentence buffer; "
                "sentences are chunked at 1024 tokens instead",
                cfg.max_sentence_length)
        if native is not None:
            enc_tokens, enc_offsets = native.encode_corpus(
                corpus, list(vocab.words), max_sent)

            def batches():
                from .data import batches_from_arrays
                return batches_from_arrays(enc_tokens, enc_offsets,
                                           cfg.words_per_step)
        else:
            def batches():
                return batch_sentences(
                    encode_sentences(reader(), vocab, max_sent),
                    cfg.words_per_step)
        self._batches = batches   # engines pick this up via fit internals

        device = cfg.device
        if device == "auto":
            import torch
            device = "cuda" if torch.cuda.is_available() else "cpu"
        if device == "cuda" or cfg.engine in ("dim", "row", "dp"):
            # single- or multi-GPU engines; dim/row/dp also run on CPU under
            # gloo (multi-process tests, torchrun without GPUs)
            from .parallel.engine import train_gpu
            syn0, syn1 = train_gpu(cfg, vocab, batches, seed,
                                   save_path=save_path,
                                   materialize=materialize,
                                   init_tables=init_tables)
        else:
            syn0, syn1 = self._fit_cpu(cfg, vocab, batches, seed,
                                       init_tables=init_tables)
            if save_path is not None:
                from .checkpoint import save_model
                save_model(save_path, cfg, vocab, syn0, syn1,
                           num_shards=cfg.num_shards or 1)
        if not materialize:
            return None
        return GlintWord2VecModel(config=cfg, vocab=vocab, syn0=syn0, syn1=syn1,
                                  output_col=self.output_col, input_col=self.input_col)

    # --- single-process CPU trainer (BASELINE config 1) --------------------
    def _fit_cpu(self, cfg: Word2VecConfig, vocab: Vocabulary, batches_fn,
                 seed: int, init_tables=None):
        if init_tables is not None:
            syn0 = np.ascontiguousarray(init_tables[0], dtype=np.float32)
            syn1 = np.ascontiguousarray(init_tables[1], dtype=np.float32)
        else:
            syn0, syn1 = sgns.init_tables(vocab.num_words, cfg.vector_size,
                                          seed)
        kp = None
        if not cfg.legacy_subsample and cfg.subsample_ratio > 0:
            kp = keep_probabilities(vocab.counts, vocab.train_words_count,
                                    cfg.subsample_ratio)
            if np.all(kp >= 1.0):
                kp = None
        table = build_unigram_table(vocab.counts, cfg.unigram_table_size,
                                    cfg.unigram_power)
        exp_table = (sgns.create_exp_table() if cfg.sigmoid_mode == "lut"
                     else None)
        try:
            from . import _cpu_native
        except ImportError:
            _cpu_native = None
            log.warning("_cpu_native extension not built; falling back to the "
                        "slow vectorized path (run `python setup.py build_ext "
                        "--inplace`)")
        rng = np.random.default_rng(seed)
        total_words = vocab.train_words_count * cfg.num_iterations
        processed = 0
        sent_base = 0
        t0 = time.time()
        for it in range(cfg.num_iterations):
            for batch in batches_fn():
                alpha = cfg.learning_rate * max(
                    1e-4, 1.0 - processed / (total_words + 1))
                if _cpu_native is not None:
                    st = _cpu_native.train_batch(
                        syn0, syn1, batch.tokens, batch.offsets, kp, table,
                        alpha, cfg.window, cfg.n, seed, sent_base,
                        cfg.window_mode, cfg.num_partitions, exp_table,
                        int(cfg.shared_negatives))
                    npos, sum_fp = st["positives"], st["sum_fplus"]
                else:
                    plan = sgns.make_plan(batch.tokens, batch.offsets, kp,
                                          table, cfg.window, cfg.n, rng,
                                          cfg.window_mode)
                    npos, sum_fp = sgns.train_plan_minibatched(
                        syn0, syn1, plan, alpha, minibatch=256)
                sent_base += batch.num_sentences
                processed += batch.num_tokens
                wps = processed / max(time.time() - t0, 1e-9)
                log.info("iter %d: %d/%d words, alpha=%.5f, %.0f words/s, "
                         "mean_fplus=%.4f", it, processed, total_words, alpha,
                         wps, sum_fp / max(npos, 1))
        return syn0, syn1


class LocalWord2VecModel:
    """`toLocal` result (mllib:651-659): plain in-memory word->vector map."""

    def __init__(self, words: List[str], vectors: np.ndarray):
        self.words = words
        self.vectors = vectors
        self.index = {w: i for i, w in enumerate(words)}

    def __getitem__(self, word: str) -> np.ndarray:
        return self.vectors[self.index[word]]

    def save(self, path: str) -> None:
        save_word2vec_text(path, self.words, self.vectors)


class GlintWord2VecModel:
    """Fitted model.  Holds the input matrix syn0 (and syn1 when kept for
    resume); model ops follow the reference op semantics.
    """

    def __init__(self, config: Word2VecConfig, vocab: Vocabulary,
                 syn0: np.ndarray, syn1: Optional[np.ndarray] = None,
                 input_col: str = "sentence", output_col: str = "vector"):
        self.config = config
        self.vocab = vocab
        self.syn0 = syn0
        self.syn1 = syn1
        self.input_col = input_col
        self.output_col = output_col
        self._norms: Optional[np.ndarray] = None
        self._gpu = None          # GpuSgns when to_gpu() was called

    # --- helpers ----------------------------------------------------------
    @property
    def num_words(self) -> int:
        return self.vocab.num_words

    @property
    def vector_size(self) -> int:
        return int(self.syn0.shape[1])

    def _f32(self) -> np.ndarray:
        return self.syn0.astype(np.float32, copy=False)

    def norms(self) -> np.ndarray:
        """Lazily cached row norms (mllib:486)."""
        if self._norms is None:
            if self._gpu is not None:
                self._norms = self._gpu.norms().cpu().numpy()
            else:
                self._norms = np.linalg.norm(self._f32(), axis=1)
        return self._norms

    # --- GPU-resident serving (the PS-serving analog; inverse of toLocal) --
    def to_gpu(self, device: str = "cuda", dtype: str = "float32"
               ) -> "GlintWord2VecModel":
        """Upload syn0 to a GPU; findSynonyms (rocBLAS GEMV + norms kernel)
        and sentence-average transform (pull_average kernel) then run
        device-side — the reference's server-side model ops."""
        from .ops.gpu import GpuSgns
        self._gpu = GpuSgns(self.num_words, self.vector_size, dtype=dtype,
                            device=device, syn0_host=self._f32())
        self._norms = None
        self._norms_t = None
        return self

    def transform_sentences_gpu(self, sentences) -> np.ndarray:
        """Batched sentence-average transform on the GPU (ml:443-456)."""
        assert self._gpu is not None, "call to_gpu() first"
        import torch
        idx_lists = [[self.vocab[w] for w in s if w in self.vocab]
                     for s in sentences]
        offsets = np.zeros(len(idx_lists) + 1, dtype=np.int32)
        np.cumsum([len(x) for x in idx_lists], out=offsets[1:])
        tokens = np.concatenate([np.asarray(x, dtype=np.int32)
                                 for x in idx_lists if x] or
                                [np.zeros(0, dtype=np.int32)])
        dev = self._gpu.device
        out = self._gpu.pull_average(
            torch.from_numpy(tokens).to(dev),
            torch.from_numpy(offsets).to(dev))
        import torch as _t
        _t.cuda.synchronize(dev)
        return out.cpu().numpy()

    # --- transform --------------------------------------------------------
    def transform(self, x):
        """word -> vector; token sequence -> average vector (ml:432-460);
        pandas DataFrame -> copy with output_col of averaged vectors."""
        if isinstance(x, str):
            return self._f32()[self.vocab[x]].copy()
        if hasattr(x, "columns"):  # DataFrame
            out = x.copy()
            out[self.output_col] = [self.transform_sentence(s)
                                    for s in x[self.input_col]]
            return out
        return self.transform_sentence(x)

    def transform_sentence(self, tokens: Sequence[str]) -> np.ndarray:
        idx = [self.vocab[w] for w in tokens if w in self.vocab]
        if not idx:
            return np.zeros(self.vector_size, dtype=np.float32)
        return self._f32()[idx].mean(axis=0)

    def transform_words(self, words: Iterable[str]) -> np.ndarray:
        """Batched per-word lookup (mllib:529-543)."""
        idx = [self.vocab[w] for w in words]
        return self._f32()[idx]

    # --- similarity -------------------------------------------------------
    def find_synonyms(self, word_or_vec, num: int) -> List[tuple]:
        """Top-``num`` cosine-similar words (mllib:554-630).  When queried by
        word, the word itself is excluded."""
        if isinstance(word_or_vec, str):
            query_word = word_or_vec
            vec = self._f32()[self.vocab[word_or_vec]]
        else:
            query_word = None
            vec = np.asarray(word_or_vec, dtype=np.float32)
        qn = np.linalg.norm(vec)
        if qn > 0:
            vec = vec / qn
        if self._gpu is not None:
            import torch
            # eager scores kernel + topk: measured FASTER than the
            # hipGraph replay once multiply() became a hand-written
            # kernel (2.9k vs 2.4k q/s — the graph re-runs the query
            # staging copies; synonyms_query remains available)
            if getattr(self, "_norms_t", None) is None:
                self._norms_t = self._gpu.norms().clamp_min(1e-12)
            cos_t = self._gpu.multiply(torch.from_numpy(vec),
                                       norms=self._norms_t)
            k = min(num + 1, self.num_words)
            val, idx = torch.topk(cos_t, k)
            val = val.cpu().numpy()
            idx = idx.cpu().numpy()
            out = []
            for j in range(len(idx)):
                w = self.vocab.words[int(idx[j])]
                if w == query_word:
                    continue
                out.append((w, float(val[j])))
                if len(out) == num:
                    break
            return out
        else:
            scores = self._f32() @ vec          # `multiply` (mllib:598)
        norms = self.norms()
        with np.errstate(divide="ignore", invalid="ignore"):
            cos = np.where(norms > 0, scores / norms, 0.0)
        k = min(num + 1, len(cos))
        top = np.argpartition(-cos, k - 1)[:k]
        top = top[np.argsort(-cos[top])]
        out = []
        for i in top:
            w = self.vocab.words[i]
            if w == query_word:
                continue
            out.append((w, float(cos[i])))
            if len(out) == num:
                break
        return out

    def find_synonyms_batch(self, queries, num: int,
                            max_score_bytes: int = 4 << 30):
        """Batched multi-query findSynonyms: one GEMM over all queries +
        one top-k (GPU when to_gpu() was called, else BLAS on host).
        Returns a list of (word, cosine) lists, query word excluded.
        Chunks the query batch so the [q, vocab] score tensor stays
        under ``max_score_bytes``."""
        q_chunk = max(1, int(max_score_bytes // (max(self.num_words, 1) * 4)))
        if len(queries) > q_chunk:
            out = []
            for i in range(0, len(queries), q_chunk):
                out.extend(self.find_synonyms_batch(
                    queries[i:i + q_chunk], num, max_score_bytes))
            return out
        Q = len(queries)
        vecs = np.empty((Q, self.vector_size), dtype=np.float32)
        skip = []
        for i, q in enumerate(queries):
            if isinstance(q, str):
                skip.append(q)
                vecs[i] = self._f32()[self.vocab[q]]
            else:
                skip.append(None)
                vecs[i] = np.asarray(q, dtype=np.float32)
        qn = np.linalg.norm(vecs, axis=1, keepdims=True)
        np.divide(vecs, qn, out=vecs, where=qn > 0)
        if self._gpu is not None:
            import torch
            scores = self._gpu.multiply_batch(torch.from_numpy(vecs))
            norms = (self._gpu.norms() if self._norms is None
                     else torch.from_numpy(self._norms)
                     .to(self._gpu.device))
            cos = torch.where(norms[None, :] > 0, scores / norms[None, :],
                              torch.zeros((), device=scores.device))
            k = min(num + 1, cos.shape[1])
            val, idx = torch.topk(cos, k, dim=1)
            val = val.cpu().numpy()
            idx = idx.cpu().numpy()
        else:
            scores = self._f32() @ vecs.T
            norms = self.norms()
            with np.errstate(divide="ignore", invalid="ignore"):
                cos = np.where(norms[:, None] > 0, scores / norms[:, None],
                               0.0)
            k = min(num + 1, cos.shape[0])
            part = np.argpartition(-cos, k - 1, axis=0)[:k]
            idx = np.take_along_axis(
                part, np.argsort(-np.take_along_axis(cos, part, axis=0),
                                 axis=0), axis=0).T
            val = np.take_along_axis(cos.T, idx, axis=1)
        out = []
        for qi in range(Q):
            res = []
            for j in range(idx.shape[1]):
                w = self.vocab.words[int(idx[qi, j])]
                if w == skip[qi]:
                    continue
                res.append((w, float(val[qi, j])))
                if len(res) == num:
                    break
            out.append(res)
        return out

    def find_synonyms_df(self, word_or_vec, num: int):
        """DataFrame variant of findSynonyms (ml:390-420 returns a
        (word, similarity) DataFrame)."""
        import pandas as pd
        rows = self.find_synonyms(word_or_vec, num)
        return pd.DataFrame(rows, columns=["word", "similarity"])

    def analogy(self, pos: List[str], neg: List[str], num: int = 10) -> List[tuple]:
        """wien - oesterreich + deutschland -> berlin style queries
        (IT spec :327-382)."""
        v = np.zeros(self.vector_size, dtype=np.float32)
        for w in pos:
            v += self._f32()[self.vocab[w]]
        for w in neg:
            v -= self._f32()[self.vocab[w]]
        skip = set(pos) | set(neg)
        res = self.find_synonyms(v, num + len(skip))
        return [(w, c) for w, c in res if w not in skip][:num]

    # --- export / persistence --------------------------------------------
    def get_vectors(self) -> dict:
        f = self._f32()
        return {w: f[i].copy() for i, w in enumerate(self.vocab.words)}

    def get_vectors_df(self):
        """DataFrame of (word, vector) — the ml-layer getVectors shape
        (ml:342-364)."""
        import pandas as pd
        f = self._f32()
        return pd.DataFrame({"word": list(self.vocab.words),
                             "vector": [f[i].copy() for i in
                                        range(self.num_words)]})

    def to_local(self) -> LocalWord2VecModel:
        return LocalWord2VecModel(list(self.vocab.words), self._f32().copy())

    def save(self, path: str, num_shards: int = 1) -> None:
        save_model(path, self.config, self.vocab, self.syn0, self.syn1,
                   num_shards=num_shards)

    @classmethod
    def load(cls, path: str) -> "GlintWord2VecModel":
        config, vocab, syn0, syn1 = load_model(path)
        return cls(config=config, vocab=vocab, syn0=syn0, syn1=syn1)

    @classmethod
    def load_sharded(cls, path: str, device: str = "auto",
                     dtype: str = "auto"):
        """Load WITHOUT host materialisation: rows stream from the shard
        files straight to each rank's device and serving ops (findSynonyms
        / transform / getVector) run sharded — the reference's PS-side
        load+serve (mllib:683-726, :598).  Works at world 1 (one GPU) or
        under torchrun (one rank per GPU); returns a
        serving.ShardedWord2VecModel."""
        from .serving import ShardedWord2VecModel
        return ShardedWord2VecModel.load(path, device=device, dtype=dtype)

    def stop(self, terminate_other_clients: bool = False) -> None:
        """Release training resources (reference: stops PS cluster,
        mllib:664-667).  Tears down the torch.distributed group if this
        process created one."""
        try:
            import torch.distributed as dist
            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass
