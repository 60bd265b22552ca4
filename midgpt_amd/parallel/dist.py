"""Distributed substrate: one process per GPU over RCCL
(torch.distributed backend "nccl" IS RCCL on ROCm), gloo on CPU.

Replaces the reference's jax.distributed + GSPMD-inserted collectives
(reference src/sharding.py, launch.py:22-23) with explicit calls:
  C1 param all-gather, C2 grad reduce-scatter, C3/C4 all-reduce,
  C6 init + barrier.

On gloo (CPU tests), reduce_scatter_tensor / all_gather_into_tensor are
unavailable; functional fallbacks keep multi-process CPU tests running on
the same code path shape.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist


def init_distributed(device_type: str | None = None) -> tuple[int, int, torch.device]:
    """Initialize the default process group from torchrun env vars.
    Returns (rank, world_size, device). Safe to call in single-process mode
    (no env vars -> world 1, no process group)."""
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if not dist.is_initialized():
            dist.init_process_group(backend=backend)
        rank = dist.get_rank()
        world = dist.get_world_size()
    else:
        rank, world = 0, 1
    if device_type is None:
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
    if device_type == "cuda":
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    return rank, world, device


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def is_main() -> bool:
    return get_rank() == 0


def barrier():
    if dist.is_initialized():
        dist.barrier()


def _is_gloo() -> bool:
    return dist.get_backend() == "gloo"


def broadcast_str(s: str, src: int = 0, max_len: int = 1024) -> str:
    """Broadcast a short string from ``src`` to all ranks (used for the
    generated rundir name; reference multihost runs prespecify it)."""
    if not dist.is_initialized():
        return s
    buf = torch.zeros(max_len, dtype=torch.uint8)
    if get_rank() == src:
        raw = s.encode()[:max_len]
        buf[:len(raw)] = torch.frombuffer(bytearray(raw), dtype=torch.uint8)
    if torch.cuda.is_available() and dist.get_backend() == "nccl":
        buf = buf.cuda()
    dist.broadcast(buf, src=src)
    raw = bytes(buf.cpu().tolist())
    return raw.rstrip(b"\x00").decode()


def all_reduce_(t: torch.Tensor, op=dist.ReduceOp.SUM if dist.is_available() else None):
    if dist.is_initialized():
        dist.all_reduce(t, op=op)
    return t


def reduce_scatter_flat(flat: torch.Tensor, shard_out: torch.Tensor):
    """SUM-reduce ``flat`` (world*S elements) across ranks, each rank
    receiving its S-element shard into ``shard_out``. Mutates ``flat`` on
    the gloo fallback."""
    if not dist.is_initialized():
        shard_out.copy_(flat[: shard_out.numel()])
        return
    if _is_gloo():
        dist.all_reduce(flat)
        r = dist.get_rank()
        s = shard_out.numel()
        shard_out.copy_(flat[r * s:(r + 1) * s])
    else:
        dist.reduce_scatter_tensor(shard_out, flat)


def all_gather_flat(flat_out: torch.Tensor, shard: torch.Tensor):
    """All-gather per-rank S-element shards into ``flat_out`` (world*S)."""
    if not dist.is_initialized():
        flat_out[: shard.numel()].copy_(shard)
        return
    if _is_gloo():
        w = dist.get_world_size()
        s = shard.numel()
        chunks = [flat_out[i * s:(i + 1) * s] for i in range(w)]
        dist.all_gather(chunks, shard.contiguous())
    else:
        dist.all_gather_into_tensor(flat_out, shard)
