"""Process-group plumbing: one process per GPU over RCCL (xGMI), gloo on CPU.

Replaces the reference's HTTPS transport: dispatch becomes
an object broadcast, result return a gather of image tensors, model sync a
weight broadcast, the liveness probe a device memory query + allgather.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist

from ..utils import get_logger

log = get_logger("parallel")


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def init_group(
    backend: Optional[str] = None,
    timeout_s: float = 300.0,
    init_method: Optional[str] = None,
) -> bool:
    """Initialise torch.distributed from torchrun env vars. Returns True if
    a multi-rank group exists. backend default: nccl(=RCCL) with GPUs, gloo
    otherwise."""
    if dist.is_initialized():
        return dist.get_world_size() > 1
    world = env_world_size()
    if world <= 1 and init_method is None:
        return False
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        torch.cuda.set_device(env_local_rank())
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    dist.init_process_group(
        backend=backend,
        init_method=init_method,
        rank=env_rank(),
        world_size=world,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    log.info(
        "rank %d/%d joined %s group", dist.get_rank(), world, backend
    )
    return world > 1


def barrier() -> None:
    if dist.is_initialized() and dist.get_world_size() > 1:
        if dist.get_backend() == "nccl":
            dist.barrier(device_ids=[torch.cuda.current_device()])
        else:
            dist.barrier()


def broadcast_object(obj, src: int = 0):
    """Broadcast one picklable object from src; returns the object."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return obj
    box = [obj]
    dist.broadcast_object_list(box, src=src)
    return box[0]


def allgather_object(obj) -> List:
    """All-gather one small picklable object per rank (infotext lists,
    (model, digest) pairs). Returns a world_size-long list in rank order."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return [obj]
    out: List = [None] * dist.get_world_size()
    dist.all_gather_object(out, obj)
    return out


def allgather_floats(values: List[float], device) -> List[List[float]]:
    """All-gather a small per-rank float vector (ipm, elapsed, flags)."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return [values]
    t = torch.tensor(values, dtype=torch.float32, device=device)
    out = [torch.empty_like(t) for _ in range(dist.get_world_size())]
    dist.all_gather(out, t)
    return [o.cpu().tolist() for o in out]


def sync_weights(module: torch.nn.Module, src: int = 0) -> None:
    """Broadcast every parameter/buffer from src (ref C13 model sync,
    POST /options -> one coalesced weight broadcast at load)."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return
    tensors = [p.data for p in module.parameters()] + [
        b.data for b in module.buffers()
    ]
    if not tensors:
        return
    try:
        # coalesced: fewer, larger RCCL broadcasts (xGMI-friendly)
        dist._broadcast_coalesced(
            dist.group.WORLD, tensors, 256 * 1024 * 1024, src
        )
    except (AttributeError, TypeError):  # pragma: no cover - older torch
        for t in tensors:
            dist.broadcast(t, src=src)


def gather_images(
    shard: torch.Tensor, shard_sizes: List[int], device
) -> Optional[torch.Tensor]:
    """All-gather variably-sized uint8 image shards [n_i, H, W, 3].

    Shards are padded to the max shard size so one all_gather moves
    everything (few MB; one hop per xGMI link, no ring serialisation on a
    single large message). Returns the concatenated [sum(n_i), H, W, 3]
    tensor on every rank (rank 0 uses it; others may drop it).
    """
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return shard
    world = dist.get_world_size()
    # Agree on the padded H/W first: a rank with an EMPTY shard (idle or
    # failed) sized its placeholder from request math, while generating
    # ranks' output size is latent-derived (H//f*f, floored hires scale) —
    # with fractional hr_scale the two can differ and all_gather would
    # error/hang on mismatched shapes. The consensus shape comes from any
    # rank that actually produced images.
    local = list(shard.shape) + [0] * (4 - shard.dim())
    shp = torch.tensor(local[:4], dtype=torch.int64, device=device)
    shapes = [torch.empty_like(shp) for _ in range(world)]
    dist.all_gather(shapes, shp)
    rows = [s.cpu().tolist() for s in shapes]
    tail = next(
        (tuple(r[1:]) for r in rows if r[0] > 0), tuple(shard.shape[1:])
    )
    if shard.shape[0] > 0 and tuple(shard.shape[1:]) != tail:
        # never raise on a subset of ranks mid-collective (the others
        # would hang in the all_gather): crop/pad to the consensus shape
        # and scream — the gallery shows the defect, the job completes
        log.error(
            "image shard shape mismatch: local %s vs consensus %s; "
            "cropping to keep the gather alive",
            tuple(shard.shape[1:]), tail,
        )
        fixed = torch.zeros(
            (shard.shape[0], *tail), dtype=shard.dtype, device=shard.device
        )
        h = min(shard.shape[1], tail[0])
        w = min(shard.shape[2], tail[1])
        fixed[:, :h, :w] = shard[:, :h, :w]
        shard = fixed
    max_n = max(max(shard_sizes), 1)
    padded = torch.zeros((max_n, *tail), dtype=shard.dtype, device=device)
    if shard.shape[0] > 0:
        padded[: shard.shape[0]] = shard.to(device)
    out = [torch.empty_like(padded) for _ in range(world)]
    dist.all_gather(out, padded)
    parts = [o[: shard_sizes[r]] for r, o in enumerate(out)]
    return torch.cat(parts, dim=0)


def destroy_group() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
