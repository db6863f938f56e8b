#!/usr/bin/env python3
"""RCCL process-group validation — run under torchrun on real GPUs:

  python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nproc-per-node N scripts/rccl_check.py

Initialises the nccl(=RCCL) backend and exercises every collective the
engines use (broadcast, fp32/bf16 allreduce, all_gather, all_to_all,
all_to_all_single with uneven splits, barrier), printing per-op status and
effective bandwidth.  At world 1 this still creates the communicator and
runs each op through RCCL — the same code path an 8-GPU launch takes — so
a 1-GPU lease de-risks the whole distributed stack.
Output is kept under profiles/ as multi-GPU readiness evidence.
"""
import json
import os
import sys
import time

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
    backend = "nccl" if use_cuda else "gloo"
    t0 = time.time()
    dist.init_process_group(backend=backend)
    init_s = time.time() - t0
    dev = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    results = {"backend": backend, "world": world, "init_s": round(init_s, 3),
               "torch": torch.__version__,
               "device": (torch.cuda.get_device_name(local_rank)
                          if use_cuda else "cpu")}

    def timed(name, fn, nbytes=0, iters=5):
        fn()  # warmup
        if use_cuda:
            torch.cuda.synchronize(dev)
        t = time.time()
        for _ in range(iters):
            fn()
        if use_cuda:
            torch.cuda.synchronize(dev)
        dt = (time.time() - t) / iters
        results[name] = {"ms": round(dt * 1e3, 3)}
        if nbytes:
            results[name]["GB_s"] = round(nbytes / dt / 1e9, 1)

    n = 16 << 20  # 16M elements
    x32 = torch.ones(n, dtype=torch.float32, device=dev)
    x16 = torch.ones(n, dtype=torch.bfloat16, device=dev)
    timed("broadcast_fp32_64MB", lambda: dist.broadcast(x32, src=0),
          nbytes=4 * n)
    timed("allreduce_fp32_64MB", lambda: dist.all_reduce(x32), nbytes=4 * n)
    try:
        timed("allreduce_bf16_32MB", lambda: dist.all_reduce(x16),
              nbytes=2 * n)
    except RuntimeError as e:
        results["allreduce_bf16_32MB"] = {"error": str(e)}
    g = [torch.empty_like(x16) for _ in range(world)]
    timed("all_gather_bf16", lambda: dist.all_gather(g, x16), nbytes=2 * n)
    a2a_out = [torch.empty(n // world, dtype=torch.bfloat16, device=dev)
               for _ in range(world)]
    a2a_in = list(torch.ones(n, dtype=torch.bfloat16, device=dev)
                  .chunk(world))
    timed("all_to_all_bf16", lambda: dist.all_to_all(a2a_out, a2a_in),
          nbytes=2 * n)
    # uneven alltoallv (the row-engine pull shape)
    in_splits = [(r + 1 + rank) % world + 1 for r in range(world)]
    out_splits = [(rank + 1 + s) % world + 1 for s in range(world)]
    flat_in = torch.arange(sum(in_splits), dtype=torch.float32, device=dev)
    flat_out = torch.empty(sum(out_splits), dtype=torch.float32, device=dev)
    timed("all_to_all_single_uneven",
          lambda: dist.all_to_all_single(
              flat_out, flat_in, output_split_sizes=out_splits,
              input_split_sizes=in_splits))
    timed("barrier", dist.barrier)

    # the engines' comm helpers, through the production module
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from glint_word2vec_amd.parallel import comm
    y = torch.ones(1 << 20, dtype=torch.float32, device=dev)
    timed("comm.all_reduce_sum_compressed",
          lambda: comm.all_reduce_sum_compressed(y), nbytes=4 * (1 << 20))
    ok = bool(torch.isfinite(x32).all() and torch.isfinite(flat_out).all())
    results["finite"] = ok
    if rank == 0:
        print(json.dumps(results, indent=2))
    dist.destroy_process_group()
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
