#!/usr/bin/env python3
"""Row-engine pull-path stage timings (VERDICT round-1 weak #2).

Forces the pull/train/push cycle at world 1 (use_direct=False) — the
per-GPU cost structure of the world-8 path with the collectives as device
copies — and times each stage with CUDA events:

    plan (counter walker) | unique | pull (gather) | train | delta+push

Run: python benchmarks/row_pull_probe.py --vocab 80000000 [--steps 5]
"""
import argparse
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import json
import time

import numpy as np
import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--vocab", type=int, default=80_000_000)
    p.add_argument("--dim", type=int, default=300)
    p.add_argument("--neg", type=int, default=5)
    p.add_argument("--window", type=int, default=5)
    p.add_argument("--words", type=int, default=2_000_000)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--atomic", action="store_true")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    args = p.parse_args()

    from glint_word2vec_amd.data import synthetic_corpus
    from glint_word2vec_amd.parallel.row_sharded import RowShardedSgns

    dev = torch.device("cuda", 0)
    torch.cuda.set_device(dev)
    batch = synthetic_corpus(args.vocab, args.words, sentence_len=100,
                             seed=1234)
    counts = np.bincount(batch.tokens, minlength=args.vocab).astype(
        np.int64) + 1
    eng = RowShardedSgns(args.vocab, args.dim,
                         dtype="bfloat16" if args.dtype == "bf16"
                         else "float32",
                         device=str(dev), seed=1, counts=counts,
                         table_size=100_000_000, subsample=1e-4,
                         atomic=args.atomic)
    eng.use_direct = False
    tok = torch.from_numpy(batch.tokens).to(dev)
    off = torch.from_numpy(batch.offsets).to(dev)
    nsent = batch.num_sentences

    stages = ["plan", "unique_route", "pull", "train", "push"]
    acc = {s: 0.0 for s in stages}

    def run_step(i, record):
        # replicate the world-1 pull_begin internals so each sub-stage gets
        # its own event pair (keep in sync with row_sharded.pull_begin)
        evs = [torch.cuda.Event(enable_timing=True) for _ in range(8)]
        evs[0].record()
        plan = eng.make_plan_counter(tok, off, args.window, args.neg, 99,
                                     sent_id_base=i * nsent)
        evs[1].record()
        uc, inv_c = torch.unique(plan.group_center, return_inverse=True)
        ut, inv_t = torch.unique(plan.pair_target, return_inverse=True)
        evs[2].record()
        cache0 = eng._ws("c0", uc.numel(), eng.store_stride, eng.syn0.dtype)
        cache1 = eng._ws("c1", ut.numel(), eng.store_stride, eng.syn0.dtype)
        eng._gather_native(eng.syn0, uc.int().contiguous(), cache0)
        eng._gather_native(eng.syn1, ut.int().contiguous(), cache1)
        evs[3].record()
        st = {"plan": plan, "gc": inv_c.int().contiguous(),
              "gt": inv_t.int().contiguous(), "cache0": cache0,
              "cache1": cache1, "orig0": None, "orig1": None,
              "uc_sorted": uc, "ut_sorted": ut}
        evs[4].record()
        eng.train_push(st, 0.01875)
        evs[5].record()
        torch.cuda.synchronize(dev)
        if record:
            acc["plan"] += evs[0].elapsed_time(evs[1])
            acc["unique_route"] += evs[1].elapsed_time(evs[2])
            acc["pull"] += evs[2].elapsed_time(evs[3])
            acc["train"] += evs[4].elapsed_time(evs[5])
            acc["push"] += evs[3].elapsed_time(evs[4])
        return plan

    plan = run_step(0, False)  # warmup
    pairs = plan.num_pairs
    uc = int(torch.unique(plan.group_center).numel())
    ut = int(torch.unique(plan.pair_target).numel())
    t0 = time.time()
    for i in range(args.steps):
        run_step(i + 1, True)
    wall = time.time() - t0
    wps = args.words * args.steps / wall
    out = {
        "vocab": args.vocab, "dim": args.dim, "dtype": args.dtype,
        "atomic": args.atomic, "words_per_step": args.words,
        "pairs_per_step": pairs, "unique_centers": uc, "unique_targets": ut,
        "words_per_sec": round(wps),
        "ms_per_step": round(wall / args.steps * 1e3, 2),
        "stage_ms": {k: round(v / args.steps, 2) for k, v in acc.items()},
        "hbm_gb": round(torch.cuda.max_memory_allocated(dev) / 2**30, 1),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
