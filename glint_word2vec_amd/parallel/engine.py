"""Training engines.

``train_gpu``: full-table single-GPU training (BASELINE.json config 2) —
both embedding matrices resident in HBM, one fused-kernel launch per step,
host->device token upload overlapped with compute via a separate copy
stream + double-buffered pinned staging.

The multi-GPU row-sharded engine lives in sharded.py.
"""
from __future__ import annotations

import logging
import time
from typing import Callable, Tuple

import numpy as np
import torch

from ..config import Word2VecConfig
from ..data import batch_sentences
from ..ops.gpu import GpuSgns
from ..vocab import Vocabulary, build_unigram_table, encode_sentences

log = logging.getLogger("glint_word2vec_amd")


def train_gpu(cfg: Word2VecConfig, vocab: Vocabulary, reader: Callable,
              seed: int) -> Tuple[np.ndarray, np.ndarray]:
    device = torch.device("cuda", torch.cuda.current_device())
    gs = GpuSgns(vocab.num_words, cfg.vector_size, cfg.dtype,
                 device=str(device), seed=seed)
    if not cfg.legacy_subsample and cfg.subsample_ratio > 0:
        gs.set_subsample(vocab.counts, vocab.train_words_count,
                         cfg.subsample_ratio)
    gs.set_table(build_unigram_table(vocab.counts, cfg.unigram_table_size,
                                     cfg.unigram_power))
    max_sent = min(cfg.max_sentence_length, 1024)

    copy_stream = torch.cuda.Stream(device)
    compute_stream = torch.cuda.current_stream(device)

    total_words = vocab.train_words_count * cfg.num_iterations
    processed = 0
    sent_base = 0
    t0 = time.time()
    prev_done = None
    for it in range(cfg.num_iterations):
        for batch in batch_sentences(
                encode_sentences(reader(), vocab, max_sent),
                cfg.words_per_step):
            alpha = cfg.learning_rate * max(1e-4, 1.0 - processed / (total_words + 1))
            with torch.cuda.stream(copy_stream):
                tok = torch.from_numpy(batch.tokens).to(device, non_blocking=True)
                off = torch.from_numpy(batch.offsets).to(device, non_blocking=True)
            compute_stream.wait_stream(copy_stream)
            gs.train_batch(tok, off, alpha, cfg.window, cfg.n, seed,
                           sent_id_base=sent_base, window_mode=cfg.window_mode,
                           atomic=cfg.atomic_updates)
            # keep tensors alive until the kernel is done
            prev_done = (tok, off)
            sent_base += batch.num_sentences
            processed += batch.num_tokens
    torch.cuda.synchronize(device)
    st = gs.read_stats()
    dt = time.time() - t0
    log.info("GPU training done: %d words in %.2fs (%.0f words/s), "
             "%d pairs, mean_fplus=%.4f", processed, dt, processed / max(dt, 1e-9),
             st.pairs, st.sum_fplus / max(st.positives, 1))
    return gs.to_host()
