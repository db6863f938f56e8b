"""Training engines.

``train_gpu`` dispatches on world size and ``config.engine``:
  * fused   — full-table single-GPU (BASELINE config 2): both matrices in
    HBM, one fused-kernel launch per step, H2D upload on a copy stream;
  * dp      — replicated tables + periodic delta-allreduce (replicated.py);
  * dim     — dimension-sharded CIKM scheme (dim_sharded.py);
  * row     — row-sharded alltoallv pull/push (row_sharded.py).
"""
from __future__ import annotations

import logging
import time
from typing import Callable, Tuple

import numpy as np
import torch

from ..config import Word2VecConfig
from ..ops.gpu import GpuSgns
from ..vocab import Vocabulary, build_unigram_table

log = logging.getLogger("glint_word2vec_amd")


def _maybe_mid_checkpoint(cfg: Word2VecConfig, save_path, step: int,
                          save_fn) -> None:
    """Mid-training checkpoint every cfg.checkpoint_every steps (0 = off).
    Complete loadable model dirs at "<save_path>-step<N>".  ``save_fn`` must
    be collective for engines whose save is (dim/dp/row: every rank calls
    this at the same step because the step loops are lockstep)."""
    if cfg.checkpoint_every <= 0 or not save_path or step == 0:
        return
    if step % cfg.checkpoint_every == 0:
        save_fn(f"{save_path}-step{step}")
        log.info("mid-training checkpoint at step %d", step)


def train_gpu(cfg: Word2VecConfig, vocab: Vocabulary, batches_fn: Callable,
              seed: int, save_path=None, materialize: bool = True,
              init_tables=None) -> Tuple[np.ndarray, np.ndarray]:
    """Dispatch to the right engine: fused single-GPU kernel at world 1,
    dp/dim/row-sharded at world > 1 (one rank per GPU, launched via
    torchrun; every rank calls fit() with the same corpus).
    ``batches_fn()`` yields SentenceBatch objects for one epoch."""
    from . import comm
    import os
    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
    rank, world = comm.init_from_env()
    engine = cfg.engine
    if engine == "auto":
        # shared policy (config.choose_engine, same as bench.py): dp while
        # the delta-allreduce merge stays cheap relative to the step,
        # dim-sharded beyond (the crossover is xGMI merge bandwidth, not
        # HBM capacity — see config.DP_MAX_TABLE_BYTES)
        from ..config import choose_engine
        dtype_bytes = 4 if cfg.dtype == "float32" else 2
        engine = choose_engine(vocab.num_words, cfg.vector_size,
                               dtype_bytes, world)
    if engine in ("dim", "row", "dp") and world >= 1:
        return _train_sharded(cfg, vocab, batches_fn, seed, engine, rank,
                              world, save_path, materialize, init_tables)
    device = torch.device("cuda", torch.cuda.current_device())
    dtype = "bfloat16" if cfg.dtype == "auto" else cfg.dtype
    gs = GpuSgns(vocab.num_words, cfg.vector_size, dtype,
                 device=str(device), seed=seed,
                 syn0_host=None if init_tables is None else init_tables[0],
                 syn1_host=None if init_tables is None else init_tables[1])
    if not cfg.legacy_subsample and cfg.subsample_ratio > 0:
        gs.set_subsample(vocab.counts, vocab.train_words_count,
                         cfg.subsample_ratio)
    gs.set_table(build_unigram_table(vocab.counts, cfg.unigram_table_size,
                                     cfg.unigram_power))
    if cfg.sigmoid_mode == "lut":
        from ..models.sgns import create_exp_table
        gs.set_sigmoid_lut(create_exp_table())
    copy_stream = torch.cuda.Stream(device)
    compute_stream = torch.cuda.current_stream(device)

    total_words = vocab.train_words_count * cfg.num_iterations
    processed = 0
    sent_base = 0
    step = 0
    t0 = time.time()
    for it in range(cfg.num_iterations):
        for batch in batches_fn():
            alpha = cfg.learning_rate * max(1e-4, 1.0 - processed / (total_words + 1))
            with torch.cuda.stream(copy_stream):
                tok = torch.from_numpy(batch.tokens).to(device, non_blocking=True)
                off = torch.from_numpy(batch.offsets).to(device, non_blocking=True)
            compute_stream.wait_stream(copy_stream)
            # tok/off were allocated on copy_stream; tell the caching
            # allocator the compute-stream kernel reads them, so the
            # buffers cannot be recycled for a later H2D copy while a
            # queued kernel still reads them (the host can run many steps
            # ahead when INFO logging is off)
            tok.record_stream(compute_stream)
            off.record_stream(compute_stream)
            gs.train_batch(tok, off, alpha, cfg.window, cfg.n, seed,
                           sent_id_base=sent_base, window_mode=cfg.window_mode,
                           atomic=cfg.resolved_update_mode() != "hogwild",
                           atomic_below=cfg.effective_atomic_below(),
                           atomic_floor=cfg.effective_atomic_floor(
                               vocab.num_words),
                           shared_negatives=cfg.shared_negatives)
            sent_base += batch.num_sentences
            processed += batch.num_tokens
            step += 1
            _maybe_mid_checkpoint(cfg, save_path, step,
                                  lambda p: gs.save_checkpoint(p, cfg, vocab))
            if log.isEnabledFor(logging.INFO):
                st = gs.read_stats(reset=False)
                wps = processed / max(time.time() - t0, 1e-9)
                log.info("iter %d: %d/%d words, alpha=%.5f, %.0f words/s, "
                         "mean_fplus=%.4f", it, processed, total_words, alpha,
                         wps, st.sum_fplus / max(st.positives, 1))
    torch.cuda.synchronize(device)
    st = gs.read_stats()
    dt = time.time() - t0
    log.info("GPU training done: %d words in %.2fs (%.0f words/s), "
             "%d pairs, mean_fplus=%.4f", processed, dt, processed / max(dt, 1e-9),
             st.pairs, st.sum_fplus / max(st.positives, 1))
    if save_path:
        # streamed from HBM — no full-matrix host materialisation
        gs.save_checkpoint(save_path, cfg, vocab,
                           num_shards=cfg.num_shards or 8)
    if not materialize:
        return None, None
    return gs.to_host()


def _train_sharded(cfg: Word2VecConfig, vocab: Vocabulary,
                   batches_fn: Callable, seed: int, engine: str, rank: int,
                   world: int, save_path=None, materialize: bool = True,
                   init_tables=None) -> Tuple[np.ndarray, np.ndarray]:
    """Multi-GPU engines; also usable on CPU (gloo) for tests.  Dim-sharded:
    every rank walks the same data (compute split by dimension).  Row-
    sharded: corpus partitioned by rank (data parallel), rows sharded."""
    import torch
    device = ("cuda:" + str(torch.cuda.current_device())
              if torch.cuda.is_available() else "cpu")
    subsample = 0.0 if cfg.legacy_subsample else cfg.subsample_ratio
    dtype = ("float32" if device == "cpu"
             else ("bfloat16" if cfg.dtype == "auto" else cfg.dtype))
    common = dict(dtype=dtype,
                  device=device, seed=seed, counts=vocab.counts,
                  table_size=cfg.unigram_table_size, subsample=subsample,
                  window_mode=cfg.window_mode,
                  shared_negatives=cfg.shared_negatives)
    total_words = vocab.train_words_count * cfg.num_iterations
    processed = 0
    sent_base = 0
    step = 0
    t0 = time.time()
    if engine == "dim":
        from .dim_sharded import DimShardedSgns
        eng = DimShardedSgns(vocab.num_words, cfg.vector_size,
                             chunk_words=cfg.chunk_words,
                             f_correction=cfg.f_correction,
                             atomic=cfg.resolved_update_mode() != "hogwild",
                             atomic_below=cfg.effective_atomic_below(),
                             atomic_floor=cfg.effective_atomic_floor(vocab.num_words),
                             **common)
        if init_tables is not None:
            eng.load_host(*init_tables)
        for it in range(cfg.num_iterations):
            for batch in batches_fn():
                alpha = cfg.learning_rate * max(
                    1e-4, 1.0 - processed / (total_words + 1))
                tok = torch.from_numpy(batch.tokens).to(eng.device)
                off = torch.from_numpy(batch.offsets).to(eng.device)
                eng.train_step(tok, off, alpha, cfg.window, cfg.n, seed,
                               sent_id_base=sent_base,
                               offsets_host=batch.offsets)
                sent_base += batch.num_sentences
                processed += batch.num_tokens
                step += 1
                _maybe_mid_checkpoint(
                    cfg, save_path, step,
                    lambda p: eng.save_checkpoint(p, cfg, vocab))
    elif engine == "dp":
        from .replicated import ReplicatedSgns
        eng = ReplicatedSgns(vocab.num_words, cfg.vector_size,
                             sync_every=cfg.sync_every,
                             atomic=cfg.resolved_update_mode() != "hogwild",
                             atomic_below=cfg.effective_atomic_below(),
                             atomic_floor=cfg.effective_atomic_floor(vocab.num_words),
                             **common)
        if init_tables is not None:
            eng.load_host(*init_tables)
        empty = (np.zeros(0, dtype=np.int32), np.zeros(1, dtype=np.int32))
        # disjoint per-rank counter-RNG streams: rank in the high bits of
        # the sentence-id space (2^48 sentences per rank before any
        # overlap — unreachable), instead of a fixed decimal stride that a
        # long-running rank could walk past
        sent_base = rank << 48
        for it in range(cfg.num_iterations):
            # data parallel: round-robin whole batches across ranks
            batches = [b for i, b in enumerate(batches_fn())
                       if i % world == rank]
            n_steps = len(batches)
            if world > 1:
                t = torch.tensor([n_steps])
                torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
                n_steps = int(t.item())
            for k in range(n_steps):
                tokens, offsets = ((batches[k].tokens, batches[k].offsets)
                                   if k < len(batches) else empty)
                alpha = cfg.learning_rate * max(
                    1e-4, 1.0 - processed / (total_words // world + 1))
                if eng.is_cuda:
                    tok = torch.from_numpy(tokens).to(eng.device)
                    off = torch.from_numpy(offsets).to(eng.device)
                else:
                    tok, off = tokens, offsets
                eng.train_step(tok, off, alpha, cfg.window, cfg.n, seed,
                               sent_id_base=sent_base)
                sent_base += max(len(offsets) - 1, 0)
                processed += len(tokens)
                step += 1
                _maybe_mid_checkpoint(
                    cfg, save_path, step,
                    lambda p: eng.save_checkpoint(p, cfg, vocab))
    else:  # row
        from .row_sharded import RowShardedSgns
        eng = RowShardedSgns(vocab.num_words, cfg.vector_size,
                             atomic=cfg.resolved_update_mode() == "atomic",
                             atomic_below=cfg.effective_atomic_below(),
                             atomic_floor=cfg.effective_atomic_floor(vocab.num_words),
                             **common)
        if init_tables is not None:
            eng.load_host(*init_tables)
        rng = np.random.default_rng(seed + 17 * rank)
        empty = (np.zeros(0, dtype=np.int32), np.zeros(1, dtype=np.int32))

        def alpha_now():
            return cfg.learning_rate * max(
                1e-4, 1.0 - processed / (total_words // world + 1))

        for it in range(cfg.num_iterations):
            batches = [b for i, b in enumerate(batches_fn())
                       if i % world == rank]
            # every rank must make the same number of collective calls
            n_steps = len(batches)
            if world > 1:
                t = torch.tensor([n_steps])
                torch.distributed.all_reduce(
                    t, op=torch.distributed.ReduceOp.MAX)
                n_steps = int(t.item())

            def batch_at(k):
                return ((batches[k].tokens, batches[k].offsets)
                        if k < len(batches) else empty)

            # Disjoint per-rank sentence-id streams (data parallel).
            sbase = (rank << 48) + sent_base
            if eng.is_cuda and world == 1:
                # fused fast path: the shard IS the full table at kernel
                # stride — one fused launch per step, no plan cycle
                for k in range(n_steps):
                    tokens, offsets = batch_at(k)
                    tok = torch.from_numpy(tokens).to(eng.device)
                    off = torch.from_numpy(offsets).to(eng.device)
                    eng.train_batch_fused(tok, off, alpha_now(), cfg.window,
                                          cfg.n, seed, sent_id_base=sbase)
                    sbase += max(len(offsets) - 1, 0)
                    processed += len(tokens)
                    step += 1
                    _maybe_mid_checkpoint(
                        cfg, save_path, step,
                        lambda p: eng.save_checkpoint(p, cfg, vocab))
                sent_base = sbase - (rank << 48)
                continue
            # world > 1 (or CPU): pipelined pull/train/push — step k+1's
            # pull is issued before step k's train+push, so the alltoallv
            # row fetch overlaps the train kernel (the reference's
            # dotprod/adjust pipeline, mllib:419-429); identical collective
            # order on every rank.
            pending = None   # (state, num_tokens)
            for k in range(n_steps):
                tokens, offsets = batch_at(k)
                if eng.is_cuda:
                    plan = eng.make_plan_counter(
                        tokens, offsets, cfg.window, cfg.n, seed,
                        sent_id_base=sbase)
                else:
                    plan = eng._to_plan_t(eng.make_plan(
                        tokens, offsets, cfg.window, cfg.n, rng))
                sbase += max(len(offsets) - 1, 0)
                state = eng.pull_begin(plan)
                if pending is not None:
                    eng.train_push(pending[0], alpha_now())
                    processed += pending[1]
                    step += 1
                    _maybe_mid_checkpoint(
                        cfg, save_path, step,
                        lambda p: eng.save_checkpoint(p, cfg, vocab))
                pending = (state, len(tokens))
            if pending is not None:
                eng.train_push(pending[0], alpha_now())
                processed += pending[1]
                step += 1
                _maybe_mid_checkpoint(
                    cfg, save_path, step,
                    lambda p: eng.save_checkpoint(p, cfg, vocab))
            sent_base = sbase - (rank << 48)
    st = eng.read_stats()
    dt = time.time() - t0
    log.info("%s-sharded training (rank %d/%d): %d words in %.2fs, %d pairs, "
             "mean_fplus=%.4f", engine, rank, world, processed, dt, st.pairs,
             st.sum_fplus / max(st.positives, 1))
    if save_path:
        if hasattr(eng, "save_checkpoint"):
            # dim/dp honour numParameterServers -> num_shards; the row
            # engine writes one shard per rank by construction
            try:
                eng.save_checkpoint(save_path, cfg, vocab,
                                    num_shards=cfg.num_shards or 8)
            except TypeError:
                eng.save_checkpoint(save_path, cfg, vocab)
        else:
            # defensive fallback (all current engines stream their own
            # checkpoints via save_checkpoint)
            s0, s1 = eng.to_host()
            if rank == 0:
                from ..checkpoint import save_model
                save_model(save_path, cfg, vocab, s0, s1,
                           num_shards=max(world, 1))
            if not materialize:
                return None, None
            return s0, s1
    if not materialize:
        return None, None
    return eng.to_host()
