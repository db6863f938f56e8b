#!/usr/bin/env python3
"""Flagship benchmark: SGNS training words/sec (whole node).

BASELINE.json metric: "training words/sec (whole node), vocab=1M dim=300
SGNS, at 1/2/4/8 MI355X".  Synthetic Zipf token stream, random-init weights,
bf16 tables (f32 math in-kernel).

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches via torch.distributed.run with one rank per GPU
(RCCL); rank r reads RANK/LOCAL_RANK/WORLD_SIZE from the env.  Weak scaling:
each GPU trains words_per_step tokens per step on its own corpus partition.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--vocab", type=int, default=1_000_000)
    p.add_argument("--dim", type=int, default=300)
    p.add_argument("--neg", type=int, default=5)
    p.add_argument("--window", type=int, default=5)
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    # 8M words/step measured fastest (204M hogwild vs 195M at 2M): fewer
    # launch gaps, better grid amortization; 16M is flat (r2z/r3a A/B)
    p.add_argument("--words-per-step", type=int, default=8_000_000)
    p.add_argument("--sentence-len", type=int, default=100)
    p.add_argument("--table-size", type=int, default=100_000_000)
    p.add_argument("--subsample", type=float, default=1e-4,
                   help="subsample ratio (drawn in-kernel; 0 disables)")
    p.add_argument("--blocks", type=int, default=0, help="grid blocks override")
    p.add_argument("--updates", choices=["hogwild", "atomic", "hybrid"],
                   default="hybrid",
                   help="row-update mode (same default as fit()): hybrid = "
                        "atomics on the hot Zipf head (rows < hot-rows), "
                        "plain hogwild RMW on the cold tail; hogwild = the "
                        "reference's fire-and-forget adjust semantics "
                        "(mllib:425); atomic = atomics everywhere")
    p.add_argument("--hot-rows", type=int, default=None,
                   help="hybrid mode: rows < K use atomics (default: "
                        "Word2VecConfig.hybrid_hot_rows)")
    p.add_argument("--hot-floor", type=int, default=None,
                   help="hybrid mode: rows < F stay hogwild (ultra-head "
                        "contention escape; default "
                        "Word2VecConfig.hybrid_skip_rows)")
    p.add_argument("--atomic", action="store_true",
                   help="alias for --updates atomic")
    p.add_argument("--atomic-below", type=int, default=None,
                   help="alias for --updates hybrid --hot-rows K")
    p.add_argument("--shared-negatives", action="store_true",
                   help="HogBatch-style shared negatives: one draw set per "
                        "position reused across its contexts (opt-in; cuts "
                        "target-row traffic ~(1+n)/(1+n/2b)-fold at high "
                        "n_neg — the dim-1024 neg-25 config)")
    p.add_argument("--pair-mode", type=int, default=None, choices=[0, 1, 2, 3],
                   help="fused-kernel variant: 0=one pair/wave (64-lane; "
                        "auto default for stride>512), 1=two pairs "
                        "(32-lane halves), 2=four pairs (16-lane), "
                        "3=two pairs software-pipelined 2-deep (auto "
                        "default for stride<=512)")
    p.add_argument("--profile-steps", type=int, default=0,
                   help="run only this many steps, no warmup JSON (rocprof)")
    p.add_argument("--device", choices=["cuda", "cpu"], default="cuda",
                   help="cpu = gloo plumbing test of the distributed path")
    p.add_argument("--engine", choices=["auto", "fused", "dim", "row", "dp"],
                   default="auto",
                   help="auto: fused at world 1; beyond: dp (replicated + "
                        "overlapped delta-allreduce) while tables are "
                        "small, else row-sharded (alltoallv pull/push)")
    p.add_argument("--sync-every", type=int, default=4,
                   help="dp engine: steps between delta merges")
    p.add_argument("--chunk-words", type=int, default=1 << 20,
                   help="dim-sharded feedback chunk size")
    return p.parse_args()


def main():
    args = parse_args()
    import torch
    from glint_word2vec_amd.config import Word2VecConfig, choose_engine

    # resolve the update mode (aliases kept for round-1 scripts)
    if args.atomic:
        args.updates = "atomic"
    elif args.atomic_below is not None:
        args.updates = "hybrid"
        args.hot_rows = args.atomic_below
    if args.hot_rows is None:
        args.hot_rows = Word2VecConfig.hybrid_hot_rows
    if args.hot_floor is None:
        args.hot_floor = (Word2VecConfig.hybrid_skip_rows
                          if args.updates == "hybrid" else 0)
    upd_atomic = args.updates != "hogwild"       # kernel atomic flag
    upd_below = (0 if args.updates == "hogwild" else
                 (2 ** 31 - 1 if args.updates == "atomic" else args.hot_rows))

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1
    use_cpu = args.device == "cpu"
    if distributed:
        import torch.distributed as dist
        if use_cpu:
            dist.init_process_group(backend="gloo")
        else:
            torch.cuda.set_device(local_rank)
            dist.init_process_group(backend="nccl")
    if use_cpu:
        device = torch.device("cpu")
    else:
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)

    from glint_word2vec_amd.data import synthetic_corpus
    from glint_word2vec_amd.vocab import build_unigram_table

    dtype = ("float32" if use_cpu else
             ("bfloat16" if args.dtype == "bf16" else "float32"))
    n_steps = args.profile_steps if args.profile_steps else args.steps
    total_launches = n_steps + (0 if args.profile_steps else args.warmup)

    engine = args.engine
    if engine == "auto":
        if distributed or use_cpu:
            esize = 2 if args.dtype == "bf16" else 4
            engine = choose_engine(args.vocab, args.dim, esize,
                                   world if distributed else 2)
        else:
            engine = "fused"

    # --- synthetic data (Zipf-realistic token stream), reused each step
    # with varying sent_id_base so RNG/negative draws differ per step.
    # dim engine: every rank walks the SAME global batch of
    # world*words_per_step tokens, each computing its dim-slice.
    # dp/row engines: each rank has its OWN words_per_step batch (data
    # parallel).  Either way per-GPU work is fixed as N grows (weak scaling).
    if engine == "dim" and distributed:
        batch = synthetic_corpus(args.vocab, args.words_per_step * world,
                                 sentence_len=args.sentence_len, seed=1234)
    else:
        batch = synthetic_corpus(args.vocab, args.words_per_step,
                                 sentence_len=args.sentence_len,
                                 seed=1234 + (rank if engine in ("dp", "row")
                                              else 0))
    counts = np.bincount(batch.tokens, minlength=args.vocab).astype(np.int64) + 1
    if distributed and engine in ("dp", "row"):
        # table/subsample stats must be identical across ranks
        import torch.distributed as dist
        ct = torch.from_numpy(counts.copy())
        if use_cpu:
            dist.all_reduce(ct)
        else:
            ctd = ct.to(device); dist.all_reduce(ctd); ct = ctd.cpu()
        counts = ct.numpy()
    if engine == "dim":
        from glint_word2vec_amd.parallel.dim_sharded import DimShardedSgns
        trainer = DimShardedSgns(args.vocab, args.dim, dtype=dtype,
                                 device=str(device), seed=1, counts=counts,
                                 table_size=args.table_size,
                                 subsample=args.subsample,
                                 chunk_words=args.chunk_words,
                                 atomic=upd_atomic, atomic_below=upd_below,
                                 atomic_floor=args.hot_floor,
                                 shared_negatives=args.shared_negatives)
        dist_mode = True
    elif engine == "dp":
        from glint_word2vec_amd.parallel.replicated import ReplicatedSgns
        trainer = ReplicatedSgns(args.vocab, args.dim, dtype=dtype,
                                 device=str(device), seed=1, counts=counts,
                                 table_size=args.table_size,
                                 subsample=args.subsample,
                                 sync_every=args.sync_every,
                                 atomic=upd_atomic, atomic_below=upd_below,
                                 atomic_floor=args.hot_floor,
                                 shared_negatives=args.shared_negatives)
        dist_mode = "dp"
    elif engine == "row":
        from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns
        est_unique = min(args.vocab,
                         args.words_per_step * (args.neg + 1) * 6)
        cache_gb = 8 * est_unique * (args.dim + 64) * \
            (2 if args.dtype == "bf16" else 4) / 2 ** 30
        if cache_gb > 200 and rank == 0:
            print(f"# warning: row-engine pull caches may need ~"
                  f"{cache_gb:.0f} GB/GPU at this vocab/step size; "
                  f"reduce --words-per-step", file=sys.stderr)
        trainer = RowShardedSgns(args.vocab, args.dim, dtype=dtype,
                                 device=str(device), seed=1, counts=counts,
                                 table_size=args.table_size,
                                 subsample=args.subsample,
                                 atomic=args.updates == "atomic",
                                 atomic_below=upd_below,
                                 atomic_floor=args.hot_floor,
                                 shared_negatives=args.shared_negatives)
        dist_mode = "row"
    else:
        from glint_word2vec_amd.ops.gpu import GpuSgns
        gs = GpuSgns(args.vocab, args.dim, dtype=dtype, device=str(device), seed=1)
        table = build_unigram_table(counts, args.table_size)
        gs.set_table(table)
        if args.subsample > 0:
            gs.set_subsample(counts, int(counts.sum()), args.subsample)
        trainer = gs
        dist_mode = False

    tok = torch.from_numpy(batch.tokens).to(device)
    off = torch.from_numpy(batch.offsets).to(device)
    alpha = 0.01875
    nsent = batch.num_sentences

    row_rng = np.random.default_rng(99 + rank)
    row_pending = []   # pipelined row engine: one pull in flight

    def step(i):
        if dist_mode == "dp":
            trainer.train_step(tok, off, alpha, args.window, args.neg,
                               seed=99 + rank, sent_id_base=i * nsent)
        elif dist_mode == "row":
            if world == 1 and trainer.is_cuda:
                trainer.train_batch_fused(tok, off, alpha, args.window,
                                          args.neg, 99 + rank,
                                          sent_id_base=i * nsent)
                return
            if trainer.is_cuda:
                plan = trainer.make_plan_counter(
                    tok, off, args.window, args.neg, 99 + rank,
                    sent_id_base=i * nsent)
            else:
                plan = trainer._to_plan_t(trainer.make_plan(
                    batch.tokens, batch.offsets, args.window, args.neg,
                    row_rng))
            # pipelined: issue this step's pull, then train+push the
            # previous step (pull k+1 overlaps train k — row_sharded.py)
            st = trainer.pull_begin(plan)
            if row_pending:
                trainer.train_push(row_pending.pop(), alpha)
            row_pending.append(st)
        elif dist_mode:
            trainer.train_step(tok, off, alpha, args.window, args.neg,
                               seed=99, sent_id_base=i * nsent,
                               offsets_host=batch.offsets)
        else:
            trainer.train_batch(tok, off, alpha, args.window, args.neg, 99,
                                sent_id_base=i * nsent,
                                atomic=upd_atomic,
                                atomic_below=(None if args.updates == "atomic"
                                              else upd_below),
                                atomic_floor=args.hot_floor,
                                shared_negatives=args.shared_negatives,
                                blocks=args.blocks or None,
                                **({} if args.pair_mode is None
                                   else {"pair_mode": args.pair_mode}))

    def barrier_sync():
        if row_pending:
            trainer.train_push(row_pending.pop(), alpha)
        if not use_cpu:
            torch.cuda.synchronize(device)
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        if not use_cpu:
            torch.cuda.synchronize(device)

    if args.profile_steps:
        for i in range(n_steps):
            step(i)
        if not use_cpu:
            torch.cuda.synchronize(device)
        if rank == 0:
            st = trainer.read_stats()
            print(f"profiled {n_steps} steps: {st}")
        return

    for i in range(args.warmup):
        step(i)
    barrier_sync()
    trainer.read_stats()  # reset
    t0 = time.time()
    for i in range(args.steps):
        step(args.warmup + i)
    barrier_sync()
    elapsed = time.time() - t0

    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    st = trainer.read_stats()
    words = args.words_per_step * args.steps * world
    wps = words / elapsed
    if rank == 0:
        out = {
            "metric": "training words/sec (whole node), vocab=1M dim=300 SGNS",
            "value": wps,
            "unit": "words/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("fp32" if use_cpu else
                      ("bf16" if args.dtype == "bf16" else "fp32")),
            "data": "synthetic (Zipf 1.05 token stream, random-init weights)",
            "config": {
                "model": f"sgns vocab={args.vocab} dim={args.dim} "
                         f"neg={args.neg} window={args.window}",
                "global_batch": args.words_per_step * world,
                "seq_len": args.sentence_len,
                "parallelism": (f"{engine}shard-rccl-x{world}" if dist_mode
                                else "hogwild-1gpu"),
                "updates": (args.updates if args.updates != "hybrid" else
                            f"hybrid(atomic rows "
                            f"{args.hot_floor}..{args.hot_rows})"),
                "shared_negatives": bool(args.shared_negatives),
            },
            "pairs_per_step": st.pairs / max(args.steps, 1),
            "mean_fplus": st.sum_fplus / max(st.positives, 1),
        }
        print(json.dumps(out))
    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
