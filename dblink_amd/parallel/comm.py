"""Distributed communication layer: one process per GPU over RCCL.

Replaces the reference's Spark shuffle / broadcast / accumulators
(``SURVEY.md §2.3``) with torch.distributed collectives:

- summary reduction  -> all_reduce on a small packed tensor (RCCL ring is fine)
- cluster migration  -> all_to_all_v over xGMI (direct pairwise on RCCL)
- theta              -> recomputed redundantly on every rank from the reduced
                        counts with the same Philox stream (no broadcast needed)

Backend: "nccl" (= RCCL on ROCm) when CUDA devices are available, else
"gloo" for CPU multi-process tests. gloo has no all_to_all, so the wrapper
falls back to batched isend/irecv there.
"""

from __future__ import annotations

import datetime
import os

import numpy as np
import torch
import torch.distributed as dist


def init_from_env(backend=None):
    """Initialize the default process group from torchrun env vars.

    Returns (rank, world_size, device). Safe to call in single-process mode
    (no env) — returns (0, 1, cpu/cuda:0) without creating a group.
    """
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        if torch.cuda.is_available():
            return 0, 1, torch.device("cuda", 0)
        return 0, 1, torch.device("cpu")
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = os.environ.get("DBLINK_BACKEND")
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, timeout=datetime.timedelta(minutes=10))
    if torch.cuda.is_available():
        # ranks may oversubscribe devices in shared-GPU test setups
        dev_idx = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(dev_idx)
        device = torch.device("cuda", dev_idx)
    else:
        device = torch.device("cpu")
    return rank, world, device


def rank_world():
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def is_distributed():
    return dist.is_available() and dist.is_initialized()


def barrier():
    if is_distributed():
        dist.barrier()


def all_reduce_sum_(t: torch.Tensor):
    if is_distributed():
        src, home = _to_comm(t)
        dist.all_reduce(src, op=dist.ReduceOp.SUM)
        if home is not None:
            t.copy_(src.to(home))
    return t


_summary_group = None


def summary_group():
    """A second process group so the small summary all-reduce can overlap
    with the migration all-to-all on the default group (separate
    communicator -> separate stream on RCCL)."""
    global _summary_group
    if not is_distributed():
        return None
    if _summary_group is None:
        _summary_group = dist.new_group(ranks=list(range(dist.get_world_size())))
    return _summary_group


def all_reduce_sum_async(t: torch.Tensor):
    """Start an async SUM all-reduce on the summary group; returns the work
    handle (or None when single-process)."""
    if not is_distributed():
        return None
    return dist.all_reduce(t, op=dist.ReduceOp.SUM, group=summary_group(), async_op=True)


def all_gather_object(obj):
    if not is_distributed():
        return [obj]
    out = [None] * dist.get_world_size()
    dist.all_gather_object(out, obj)
    return out


def _backend_supports_all_to_all():
    try:
        return dist.get_backend() == "nccl"
    except Exception:
        return False


def _comm_device():
    """Device collectives must run on: cuda for RCCL, cpu for gloo."""
    try:
        backend = dist.get_backend()
    except Exception:
        return None
    if backend == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    if backend == "gloo":
        return torch.device("cpu")
    return None


def _to_comm(t):
    dev = _comm_device()
    if dev is not None and t.device != dev:
        return t.to(dev), t.device
    return t, None


def exchange_counts(*send_count_lists):
    """Exchange K count vectors in ONE all_to_all (one sync instead of K).

    Each argument is a length-world list; returns the matching recv lists.
    """
    if not is_distributed():
        return tuple([int(c) for c in lst] for lst in send_count_lists)
    world = dist.get_world_size()
    K = len(send_count_lists)
    sc = torch.tensor(
        [[int(lst[p]) for lst in send_count_lists] for p in range(world)],
        dtype=torch.int64,
    ).reshape(-1)
    dev = _comm_device()
    if dev is not None:
        sc = sc.to(dev)
    rc = torch.empty_like(sc)
    dist.all_to_all_single(rc, sc)
    rc = rc.view(world, K).cpu()
    return tuple([int(rc[p, k]) for p in range(world)] for k in range(K))


def _fallback_exchange(tensor, send_counts, recv_counts, out):
    """Pairwise isend/irecv exchange (gloo); chunks bridged to the comm
    device (gloo needs CPU buffers even when state lives on the GPU)."""
    rank = dist.get_rank()
    world = dist.get_world_size()
    send_offsets = np.concatenate([[0], np.cumsum(send_counts)])
    recv_offsets = np.concatenate([[0], np.cumsum(recv_counts)])
    reqs = []
    src, _ = _to_comm(tensor.contiguous())
    for peer in range(world):
        if peer == rank:
            continue
        chunk = src[send_offsets[peer] : send_offsets[peer + 1]]
        if chunk.numel():
            reqs.append(dist.isend(chunk.clone(), dst=peer))
    for peer in range(world):
        if peer == rank:
            out[recv_offsets[peer] : recv_offsets[peer + 1]] = tensor[
                send_offsets[peer] : send_offsets[peer + 1]
            ]
            continue
        dst = out[recv_offsets[peer] : recv_offsets[peer + 1]]
        if dst.numel():
            buf = torch.empty(dst.shape, dtype=dst.dtype, device=src.device)
            dist.recv(buf, src=peer)
            dst.copy_(buf.to(dst.device))
    for r in reqs:
        r.wait()
    return out


def all_to_all_payload(tensor: torch.Tensor, send_counts, recv_counts):
    """Row-slice exchange with counts already known (no counts round-trip)."""
    if not is_distributed():
        return tensor
    world = dist.get_world_size()
    out_shape = (sum(recv_counts),) + tuple(tensor.shape[1:])
    if _backend_supports_all_to_all():
        src, home = _to_comm(tensor.contiguous())
        out = torch.empty(out_shape, dtype=tensor.dtype, device=src.device)
        dist.all_to_all_single(out, src, recv_counts, send_counts)
        return out.to(home) if home is not None else out
    out = torch.empty(out_shape, dtype=tensor.dtype, device=tensor.device)
    return _fallback_exchange(tensor, send_counts, recv_counts, out)


def all_to_all_v(tensor: torch.Tensor, send_counts, device=None):
    """Exchange row-slices of ``tensor`` between all ranks.

    ``send_counts``: int list/array of length world, rows destined per rank
    (rows must already be grouped by destination rank, ascending).
    Returns (received tensor, recv_counts list).
    """
    if not is_distributed():
        return tensor, [int(send_counts[0])]
    world = dist.get_world_size()
    send_counts = [int(c) for c in send_counts]
    comm_dev = _comm_device()
    sc = torch.tensor(send_counts, dtype=torch.int64,
                      device=comm_dev if comm_dev is not None else tensor.device)
    rc = torch.empty_like(sc)
    dist.all_to_all_single(rc, sc)
    recv_counts = [int(x) for x in rc.cpu()]
    out_shape = (sum(recv_counts),) + tuple(tensor.shape[1:])
    if _backend_supports_all_to_all():
        src, home = _to_comm(tensor.contiguous())
        out = torch.empty(out_shape, dtype=tensor.dtype, device=src.device)
        dist.all_to_all_single(out, src, recv_counts, send_counts)
        return (out.to(home) if home is not None else out), recv_counts
    out = torch.empty(out_shape, dtype=tensor.dtype, device=tensor.device)
    _fallback_exchange(tensor, send_counts, recv_counts, out)
    return out, recv_counts
