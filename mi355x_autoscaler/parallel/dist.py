"""torch.distributed plumbing for the sharded controller.

On MI355X nodes the backend is "nccl" (RCCL over xGMI: 7 p2p links x
~153 GB/s per GPU); on CPU test rigs it is "gloo". The per-cycle
traffic is tiny and latency-bound (segment key/count rows, the
per-pool decision vector, per-node maintain partials), which is
exactly the regime SURVEY §2.2 row G describes — a single ring is
fine; there is nothing bandwidth-bound here except snapshot
rebalancing (row H, chunked).
"""

from __future__ import annotations

import os


import numpy as np
import torch
import torch.distributed as dist


def init_from_env() -> bool:
    """Initialize the default process group from torchrun env vars.
    Returns True if distributed is active (world > 1 or explicit)."""
    if dist.is_available() and dist.is_initialized():
        return True
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return False
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return True


def rank() -> int:
    return dist.get_rank() if dist.is_available() and dist.is_initialized() else 0


def world_size() -> int:
    return dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1


def barrier():
    if world_size() > 1:
        dist.barrier()


def allgather_rows(rows: np.ndarray, device=None) -> np.ndarray:
    """All-gather a variable-length [G_local, C] int64 row matrix; the
    result is the concatenation over ranks (rank order). Collective #1
    of the decision agreement (segment keys + counts)."""
    if world_size() == 1:
        return rows
    assert rows.ndim == 2
    C = rows.shape[1]
    # device-side gather only when the backend actually supports CUDA
    # tensors (nccl/RCCL); gloo with GPU engines gathers on the host
    use_cuda = (device is not None and str(device).startswith("cuda")
                and backend() == "nccl")
    dev = device if use_cuda else torch.device("cpu")
    t = torch.from_numpy(np.ascontiguousarray(rows)).to(dev)

    n_local = torch.tensor([t.shape[0]], dtype=torch.int64, device=dev)
    sizes = [torch.zeros(1, dtype=torch.int64, device=dev) for _ in range(world_size())]
    dist.all_gather(sizes, n_local)
    max_n = int(max(s.item() for s in sizes))
    if max_n == 0:
        return np.zeros((0, C), np.int64)
    pad = torch.zeros((max_n, C), dtype=torch.int64, device=dev)
    pad[: t.shape[0]] = t
    outs = [torch.zeros_like(pad) for _ in range(world_size())]
    dist.all_gather(outs, pad)
    chunks = [o[: int(s.item())].cpu().numpy() for o, s in zip(outs, sizes)]
    return np.concatenate(chunks, axis=0)


def backend() -> str:
    """Active torch.distributed backend name ('' when uninitialized)."""
    try:
        return dist.get_backend() if dist.is_initialized() else ""
    except Exception:
        return ""


def allgather_rows_device(rows: torch.Tensor) -> torch.Tensor:
    """Device-resident variant of allgather_rows: rows stay on the
    GPU end to end (one scalar MAX all-reduce for the pad size, one
    all_gather). Padded rows carry count 0 and are skipped by the
    rows_hash kernel, so no per-rank size readback is needed."""
    if world_size() == 1:
        return rows
    n = torch.tensor([rows.shape[0]], dtype=torch.int64, device=rows.device)
    dist.all_reduce(n, op=dist.ReduceOp.MAX)
    max_n = int(n.item())
    if max_n == 0:
        return rows[:0]
    pad = torch.zeros((max_n, rows.shape[1]), dtype=torch.int64,
                      device=rows.device)
    pad[: rows.shape[0]] = rows
    outs = [torch.zeros_like(pad) for _ in range(world_size())]
    dist.all_gather(outs, pad)
    return torch.cat(outs, dim=0)


def allgather_fixed_device(t: torch.Tensor) -> torch.Tensor:
    """All-gather equal-shaped device tensors — ZERO host syncs (no
    size exchange; padding carries count 0, overflow rides the poison
    row). Returns the rank-order concatenation."""
    if world_size() == 1:
        return t
    outs = [torch.zeros_like(t) for _ in range(world_size())]
    dist.all_gather(outs, t.contiguous())
    return torch.cat(outs, dim=0)


def allreduce_tensor(t: torch.Tensor, op: str = "sum") -> torch.Tensor:
    """In-place all-reduce; returns t. ops: sum | max | min. CUDA
    tensors bounce through the host when the backend can't take them
    (gloo test rigs driving real GPU engines)."""
    if world_size() == 1:
        return t
    red = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX, "min": dist.ReduceOp.MIN}[op]
    if t.is_cuda and backend() != "nccl":
        h = t.cpu()
        dist.all_reduce(h, op=red)
        t.copy_(h)
        return t
    dist.all_reduce(t, op=red)
    return t
