"""Cross-rank migration of entity clusters after a sweep.

Replaces the reference's Spark shuffle (``GibbsUpdates.scala:144-150`` +
``util/HardPartitioner.scala``): partition id -> owning rank is
``pid % world_size`` (the HardPartitioner contract), and each entity travels
together with its linked records.

Implementation: rows are grouped by destination rank with one stable argsort,
payloads are packed into two int32 matrices, and a migration costs one
counts exchange plus two all-to-all-v calls. On RCCL this is a direct
pairwise all-to-all over xGMI; on gloo (CPU tests) it falls back to
isend/irecv.
"""

from __future__ import annotations

import numpy as np
import torch

from . import comm


def owner_of_partition(pid, world_size):
    return pid % world_size


def migrate(state, world_size: int, device=None):
    """Exchange clusters so every entity lives on ``ent_part % world`` rank.

    ``state`` is a ChainState with numpy arrays (CPU engine) — the GPU engine
    has its own tensor-resident variant. Afterwards the state is re-sorted by
    partition id.
    """
    if world_size <= 1 or not comm.is_distributed():
        state.sort_by_partition()
        return state

    dest_e = state.ent_part % world_size  # [E]
    order_e = np.argsort(dest_e, kind="stable")
    inv_e = np.empty_like(order_e)
    inv_e[order_e] = np.arange(order_e.size)
    send_counts_e = np.bincount(dest_e, minlength=world_size)

    # records follow their entity
    dest_r = dest_e[state.rec_ent]
    order_r = np.argsort(dest_r, kind="stable")
    send_counts_r = np.bincount(dest_r, minlength=world_size)

    # entity-local index within its destination block, so receivers can
    # rebuild rec_ent: for each entity, its position among same-destination
    # entities; receiver offsets by its own entity-block boundaries.
    pos_in_dest = inv_e - np.concatenate([[0], np.cumsum(send_counts_e)])[dest_e]

    def a2a(arr_np, counts_list):
        t = torch.from_numpy(np.ascontiguousarray(arr_np))
        out, counts = comm.all_to_all_v(t, counts_list, device)
        return out.numpy(), counts

    counts_e = [int(c) for c in send_counts_e]
    new_ent_values, recv_counts_e = a2a(state.ent_values[order_e], counts_e)
    new_ent_part, _ = a2a(state.ent_part[order_e], counts_e)

    counts_r = [int(c) for c in send_counts_r]
    rec_ent_local = pos_in_dest[state.rec_ent][order_r].astype(np.int64)
    new_rec_entlocal, recv_counts_r = a2a(rec_ent_local, counts_r)
    new_rec_values, _ = a2a(state.rec_values[order_r], counts_r)
    new_rec_file, _ = a2a(state.rec_file[order_r], counts_r)
    new_rec_dist, _ = a2a(state.rec_dist[order_r], counts_r)
    new_rec_gid, _ = a2a(state.rec_gid[order_r], counts_r)

    # rebuild rec_ent: received records from rank s refer to entity block s
    ent_block_offsets = np.concatenate([[0], np.cumsum(recv_counts_e)])
    rec_block_offsets = np.concatenate([[0], np.cumsum(recv_counts_r)])
    new_rec_ent = np.empty(new_rec_entlocal.shape[0], dtype=np.int64)
    for s in range(len(recv_counts_r)):
        r0, r1 = rec_block_offsets[s], rec_block_offsets[s + 1]
        new_rec_ent[r0:r1] = new_rec_entlocal[r0:r1] + ent_block_offsets[s]

    state.ent_values = new_ent_values
    state.ent_part = new_ent_part
    state.rec_values = new_rec_values
    state.rec_file = new_rec_file
    state.rec_dist = new_rec_dist
    state.rec_gid = new_rec_gid
    state.rec_ent = new_rec_ent
    state.sort_by_partition()
    return state


def migrate_and_sort_tensors(gs, world_size: int):
    """Tensor-resident variant of ``migrate`` used by the GPU engine.

    ``gs`` carries torch tensors: ent_values, ent_part, rec_values, rec_file,
    rec_dist, rec_gid, rec_ent (+ rec_part, rebuilt here). Works on CPU
    (gloo tests) and GPU (RCCL all-to-all over xGMI) alike.

    Payloads are packed into two int32 matrices (entities: values|part|nrec;
    records: values|file|dist-bitmask|gid_lo|gid_hi) so a migration costs one
    counts exchange + two all-to-all-v calls. Records of one source arrive
    grouped behind their entity in order, so rec_ent is rebuilt from the
    per-entity record counts.
    """
    import torch

    device = gs.ent_part.device
    A = gs.ent_values.shape[1]
    assert A <= 16, "distortion bitmask packing supports at most 16 attributes"
    if world_size > 1 and comm.is_distributed():
        dest_e = gs.ent_part.to(torch.int64) % world_size
        order_e = torch.argsort(dest_e, stable=True)
        send_e = torch.bincount(dest_e, minlength=world_size)
        inv_e = torch.empty_like(order_e)
        inv_e[order_e] = torch.arange(order_e.numel(), device=device)
        base = torch.cumsum(
            torch.cat([torch.zeros(1, dtype=torch.int64, device=device), send_e[:-1]]), 0
        )
        pos_in_dest = inv_e - base[dest_e]  # entity position within its dest block
        dest_r = dest_e[gs.rec_ent]
        # group records behind their entity's position so receivers can rebuild
        # rec_ent from per-entity counts (rec_ent need not be sorted here)
        E_all = gs.ent_values.shape[0]
        order_r = torch.argsort(dest_r * max(E_all, 1) + pos_in_dest[gs.rec_ent], stable=True)
        send_r = torch.bincount(dest_r, minlength=world_size)

        E, R = gs.ent_values.shape[0], gs.rec_values.shape[0]
        nrec = torch.zeros(E, dtype=torch.int64, device=device)
        nrec.scatter_add_(0, gs.rec_ent, torch.ones_like(gs.rec_ent))

        ent_pack = torch.empty((E, A + 2), dtype=torch.int32, device=device)
        ent_pack[:, :A] = gs.ent_values[order_e]
        ent_pack[:, A] = gs.ent_part[order_e]
        ent_pack[:, A + 1] = nrec[order_e].to(torch.int32)

        weights = (1 << torch.arange(A, device=device, dtype=torch.int32))
        distbits = (gs.rec_dist.to(torch.int32) * weights.view(1, A)).sum(dim=1)
        rec_pack = torch.empty((R, A + 4), dtype=torch.int32, device=device)
        rec_pack[:, :A] = gs.rec_values[order_r]
        rec_pack[:, A] = gs.rec_file[order_r]
        rec_pack[:, A + 1] = distbits[order_r]
        gid = gs.rec_gid[order_r]
        rec_pack[:, A + 2] = (gid & 0xFFFFFFFF).to(torch.int32)
        rec_pack[:, A + 3] = (gid >> 32).to(torch.int32)

        send_e_l = [int(x) for x in send_e.cpu()]
        send_r_l = [int(x) for x in send_r.cpu()]
        recv_e, recv_r = comm.exchange_counts(send_e_l, send_r_l)
        new_ep = comm.all_to_all_payload(ent_pack.contiguous(), send_e_l, recv_e)
        new_rp = comm.all_to_all_payload(rec_pack.contiguous(), send_r_l, recv_r)

        gs.ent_values = new_ep[:, :A].contiguous()
        gs.ent_part = new_ep[:, A].contiguous()
        new_nrec = new_ep[:, A + 1].to(torch.int64)
        gs.rec_values = new_rp[:, :A].contiguous()
        gs.rec_file = new_rp[:, A].contiguous()
        db = new_rp[:, A + 1].to(torch.int32)
        gs.rec_dist = (
            (db.view(-1, 1) >> torch.arange(A, device=device, dtype=torch.int32).view(1, A)) & 1
        ).to(torch.uint8).contiguous()
        lo = new_rp[:, A + 2].to(torch.int64) & 0xFFFFFFFF
        hi = new_rp[:, A + 3].to(torch.int64)
        gs.rec_gid = (lo | (hi << 32)).contiguous()
        # records from each source arrive grouped behind their entities in
        # entity order, and entity blocks concatenate in source order
        gs.rec_ent = torch.repeat_interleave(
            torch.arange(new_ep.shape[0], device=device, dtype=torch.int64), new_nrec
        )

    # local re-sort by partition id (stable)
    order = torch.argsort(gs.ent_part.to(torch.int64), stable=True)
    inv = torch.empty_like(order)
    inv[order] = torch.arange(order.numel(), device=device)
    gs.ent_values = gs.ent_values[order].contiguous()
    gs.ent_part = gs.ent_part[order].contiguous()
    new_rec_ent = inv[gs.rec_ent]
    rorder = torch.argsort(new_rec_ent, stable=True)
    gs.rec_ent = new_rec_ent[rorder].contiguous()
    gs.rec_values = gs.rec_values[rorder].contiguous()
    gs.rec_file = gs.rec_file[rorder].contiguous()
    gs.rec_dist = gs.rec_dist[rorder].contiguous()
    gs.rec_gid = gs.rec_gid[rorder].contiguous()
    gs.rec_part = gs.ent_part[gs.rec_ent].contiguous()
    return gs
