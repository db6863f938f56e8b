"""torch.distributed helpers: RCCL (backend "nccl" on ROCm) in production,
gloo for CPU multi-process tests.

alltoallv: RCCL has dist.all_to_all_single; gloo does not, so the fallback
implements the same contract with isend/irecv pairs (tests only).
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist


def init_from_env(backend: Optional[str] = None) -> tuple[int, int]:
    """Initialise the default process group from torchrun env vars.
    Returns (rank, world).

    A world-1 run under torchrun (MASTER_ADDR set) still initialises the
    group: RCCL communicator creation and every world-1 collective then
    exercise the exact code path an 8-GPU launch uses — the production
    path is the tested path, not a bypass."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world == 1 and "MASTER_ADDR" not in os.environ:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=300))
    return dist.get_rank(), dist.get_world_size()


def backend_is_gloo() -> bool:
    return dist.is_initialized() and dist.get_backend() == "gloo"


def all_to_all_v(output_chunks: List[torch.Tensor],
                 input_chunks: List[torch.Tensor]) -> None:
    """Exchange variable-size chunks: input_chunks[d] goes to rank d;
    output_chunks[s] receives from rank s.  All tensors pre-allocated with
    correct sizes (sizes agreed beforehand via all_to_all of counts)."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        output_chunks[0].copy_(input_chunks[0])
        return
    if backend_is_gloo():
        rank = dist.get_rank()
        world = dist.get_world_size()
        reqs = []
        for peer in range(world):
            if peer == rank:
                output_chunks[rank].copy_(input_chunks[rank])
                continue
            if input_chunks[peer].numel() > 0:
                reqs.append(dist.isend(input_chunks[peer].contiguous(), peer))
            if output_chunks[peer].numel() > 0:
                reqs.append(dist.irecv(output_chunks[peer], peer))
        for r in reqs:
            r.wait()
    else:
        dist.all_to_all(output_chunks, list(input_chunks))


def all_to_all_single_v(output: torch.Tensor, input: torch.Tensor,
                        out_splits, in_splits) -> None:
    """Flat-buffer alltoallv: ``input`` rows [sum(in_splits), ...] where the
    d-th contiguous segment goes to rank d; ``output`` receives rank s's
    segment at the s-th position.  One RCCL op instead of world tensors —
    the preferred form for the row-sharded pull/push exchanges."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        output.copy_(input)
        return
    if backend_is_gloo():
        rank = dist.get_rank()
        world = dist.get_world_size()
        ob = [0]
        ib = [0]
        for s in range(world):
            ob.append(ob[-1] + int(out_splits[s]))
            ib.append(ib[-1] + int(in_splits[s]))
        reqs = []
        for peer in range(world):
            if peer == rank:
                output[ob[rank]:ob[rank + 1]].copy_(
                    input[ib[rank]:ib[rank + 1]])
                continue
            if int(in_splits[peer]) > 0:
                reqs.append(dist.isend(
                    input[ib[peer]:ib[peer + 1]].contiguous(), peer))
            if int(out_splits[peer]) > 0:
                reqs.append(dist.irecv(output[ob[peer]:ob[peer + 1]], peer))
        for r in reqs:
            r.wait()
    else:
        dist.all_to_all_single(output, input,
                               output_split_sizes=[int(x) for x in out_splits],
                               input_split_sizes=[int(x) for x in in_splits])


def exchange_counts(counts: torch.Tensor) -> torch.Tensor:
    """counts[d] = number of items this rank sends to rank d.
    Returns recv_counts[s] = number of items rank s sends to this rank."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    if world == 1:
        return counts.clone()
    gathered = [torch.zeros_like(counts) for _ in range(world)]
    dist.all_gather(gathered, counts)
    rank = dist.get_rank()
    return torch.stack([g[rank] for g in gathered])


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


_BF16_OK = True


def bf16_collectives_ok() -> bool:
    """bf16 tensors over collectives: RCCL yes, gloo no; sticky-disabled if
    a bf16 allreduce ever fails at runtime."""
    return (_BF16_OK and dist.is_initialized()
            and dist.get_backend() == "nccl")


def all_reduce_sum_compressed(t: torch.Tensor) -> torch.Tensor:
    """Sum-allreduce with bf16 transport when the backend supports it
    (halves xGMI bytes; ~0.4% relative rounding on the summand — fine for
    SGNS dot partials and table deltas)."""
    if not (dist.is_initialized() and dist.get_world_size() > 1):
        return t
    if t.dtype == torch.float32 and bf16_collectives_ok():
        try:
            c = t.bfloat16()
            dist.all_reduce(c, op=dist.ReduceOp.SUM)
            t.copy_(c.float())
            return t
        except RuntimeError:
            # dtype unsupported by this RCCL build: raises uniformly on all
            # ranks before any transfer — safe to retry uncompressed
            global _BF16_OK
            _BF16_OK = False
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def broadcast_(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.broadcast(t, src=src)
    return t


def barrier() -> None:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.barrier()
