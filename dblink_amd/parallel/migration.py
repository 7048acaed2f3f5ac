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


def migrate_overlapped(gs, world_size: int, rank: int, during_flight=None,
                       hist_add=None):
    """Overlapped, migrant-only migration (the per-sweep RCCL all-to-all over
    xGMI): only entities whose destination rank changed —
    with their records — travel; the counts exchange and both payload
    exchanges run async on the communicator's stream while the compute
    stream packs payloads, sorts the stay-home block, and (via ``hist_add``)
    pre-builds the NEXT sweep's posting histogram; ``during_flight`` runs on
    the host while payloads are in the air (the engine uses it for the
    summary readback that feeds the next theta draw).

    Bitwise-identical post-state to ``migrate_and_sort_tensors``: the merged
    pre-sort entity order is [arrivals from ranks < rank | stay-home |
    arrivals from ranks > rank] (what the dense all-to-all would produce),
    followed by the same stable sort by partition id.

    Replaces the reference's Spark shuffle (GibbsUpdates.scala:144-150).
    """
    import torch
    import torch.distributed as dist

    device = gs.ent_part.device
    A = gs.ent_values.shape[1]
    assert A <= 16, "distortion bitmask packing supports at most 16 attributes"
    if world_size <= 1 or not comm.is_distributed():
        return migrate_and_sort_tensors(gs, world_size)

    E, R = gs.ent_values.shape[0], gs.rec_values.shape[0]
    dest_e = gs.ent_part.to(torch.int64) % world_size
    mig_mask = dest_e != rank
    mig_idx = torch.nonzero(mig_mask).squeeze(1)       # original order
    stay_idx = torch.nonzero(~mig_mask).squeeze(1)
    mdest = dest_e[mig_idx]
    morder = torch.argsort(mdest, stable=True)
    mig_sorted = mig_idx[morder]                       # grouped by dest
    send_e = torch.bincount(mdest, minlength=world_size)

    dest_r = dest_e[gs.rec_ent]
    rmig_idx = torch.nonzero(dest_r != rank).squeeze(1)
    send_r = torch.bincount(dest_r[rmig_idx], minlength=world_size)

    # ---- async counts exchange (equal splits, stays on the comm device) ----
    comm_dev = comm._comm_device() or device
    sc = torch.stack([send_e, send_r], dim=1).reshape(-1).to(comm_dev)
    rc = torch.empty_like(sc)
    counts_work = dist.all_to_all_single(rc, sc, async_op=True)

    # ---- pack migrant payloads on the compute stream ------------------------
    nrec = torch.zeros(E, dtype=torch.int64, device=device)
    nrec.scatter_add_(0, gs.rec_ent, torch.ones_like(gs.rec_ent))

    n_mig = mig_sorted.numel()
    ent_pack = torch.empty((n_mig, A + 2), dtype=torch.int32, device=device)
    ent_pack[:, :A] = gs.ent_values[mig_sorted]
    ent_pack[:, A] = gs.ent_part[mig_sorted]
    ent_pack[:, A + 1] = nrec[mig_sorted].to(torch.int32)

    # migrant records sorted by (dest, entity position within dest block) so
    # receivers can rebuild rec_ent from the per-entity record counts
    base = torch.cumsum(
        torch.cat([torch.zeros(1, dtype=torch.int64, device=device), send_e[:-1]]), 0
    )
    pos_of_ent = torch.empty(E, dtype=torch.int64, device=device)
    pos_of_ent[mig_sorted] = (
        torch.arange(n_mig, device=device, dtype=torch.int64) - base[mdest[morder]]
    )
    rkey = dest_r[rmig_idx] * max(E, 1) + pos_of_ent[gs.rec_ent[rmig_idx]]
    rmig_sorted = rmig_idx[torch.argsort(rkey, stable=True)]

    n_rmig = rmig_sorted.numel()
    weights = 1 << torch.arange(A, device=device, dtype=torch.int32)
    rec_pack = torch.empty((n_rmig, A + 4), dtype=torch.int32, device=device)
    rec_pack[:, :A] = gs.rec_values[rmig_sorted]
    rec_pack[:, A] = gs.rec_file[rmig_sorted]
    rec_pack[:, A + 1] = (
        gs.rec_dist[rmig_sorted].to(torch.int32) * weights.view(1, A)
    ).sum(dim=1)
    gid = gs.rec_gid[rmig_sorted]
    rec_pack[:, A + 2] = (gid & 0xFFFFFFFF).to(torch.int32)
    rec_pack[:, A + 3] = (gid >> 32).to(torch.int32)

    # stay-home block (original relative order, like the dense path's self
    # block); posting histogram for the next sweep can start right away
    stay_vals = gs.ent_values[stay_idx].contiguous()
    stay_part = gs.ent_part[stay_idx].contiguous()
    if hist_add is not None:
        hist_add(stay_part, stay_vals, reset=True)

    # ---- counts land: allocate and launch the payload exchanges -------------
    counts_work.wait()
    both = torch.stack([sc, rc]).cpu()  # ONE small D2H for all count lists
    send_el = [int(x) for x in both[0, 0::2]]
    send_rl = [int(x) for x in both[0, 1::2]]
    recv_el = [int(x) for x in both[1, 0::2]]
    recv_rl = [int(x) for x in both[1, 1::2]]

    ep_src, ep_home = comm._to_comm(ent_pack)
    rp_src, _ = comm._to_comm(rec_pack)
    arr_ep = torch.empty((sum(recv_el), A + 2), dtype=torch.int32, device=ep_src.device)
    arr_rp = torch.empty((sum(recv_rl), A + 4), dtype=torch.int32, device=rp_src.device)
    work_e = dist.all_to_all_single(arr_ep, ep_src, recv_el, send_el, async_op=True)
    work_r = dist.all_to_all_single(arr_rp, rp_src, recv_rl, send_rl, async_op=True)

    if during_flight is not None:
        during_flight()

    work_e.wait()
    work_r.wait()
    if ep_home is not None:  # gloo bridging for CUDA-resident tests
        arr_ep = arr_ep.to(ep_home)
        arr_rp = arr_rp.to(ep_home)

    # ---- merge: [arrivals < rank | stay | arrivals > rank], stable sort -----
    n_arr = arr_ep.shape[0]
    n_stay = stay_idx.numel()
    arr_before = int(sum(recv_el[:rank]))  # arrivals from lower ranks
    arr_vals = arr_ep[:, :A]
    arr_part = arr_ep[:, A]
    arr_nrec = arr_ep[:, A + 1].to(torch.int64)
    if hist_add is not None:
        hist_add(arr_part.contiguous(), arr_vals.contiguous(), reset=False)

    E_new = n_stay + n_arr
    merged_vals = torch.empty((E_new, A), dtype=torch.int32, device=device)
    merged_part = torch.empty(E_new, dtype=torch.int32, device=device)
    merged_vals[:arr_before] = arr_vals[:arr_before]
    merged_part[:arr_before] = arr_part[:arr_before]
    merged_vals[arr_before : arr_before + n_stay] = stay_vals
    merged_part[arr_before : arr_before + n_stay] = stay_part
    merged_vals[arr_before + n_stay :] = arr_vals[arr_before:]
    merged_part[arr_before + n_stay :] = arr_part[arr_before:]

    order = torch.argsort(merged_part.to(torch.int64), stable=True)
    inv = torch.empty_like(order)
    inv[order] = torch.arange(E_new, device=device)
    gs.ent_values = merged_vals[order].contiguous()
    gs.ent_part = merged_part[order].contiguous()

    # ---- records: concat [stay | arrivals], final entity ids, stable sort ---
    stay_r = torch.nonzero(dest_r == rank).squeeze(1)
    # stay entity -> merged pre-sort index
    stay_pos = torch.empty(E, dtype=torch.int64, device=device)
    stay_pos[stay_idx] = torch.arange(n_stay, device=device, dtype=torch.int64)
    p_stay = arr_before + stay_pos[gs.rec_ent[stay_r]]
    # arrival entity k (global, source-ascending) -> merged pre-sort index
    k = torch.arange(n_arr, device=device, dtype=torch.int64)
    p_of_arr_ent = torch.where(k < arr_before, k, k + n_stay)
    p_arr = torch.repeat_interleave(p_of_arr_ent, arr_nrec)

    new_rec_ent = inv[torch.cat([p_stay, p_arr])]
    rorder = torch.argsort(new_rec_ent, stable=True)
    n_rstay = stay_r.numel()
    ro_stay = rorder < n_rstay

    def merge_rec(stay_col, arr_col, dtype):
        out = torch.empty((rorder.numel(),) + tuple(stay_col.shape[1:]),
                          dtype=dtype, device=device)
        out[ro_stay] = stay_col[rorder[ro_stay]].to(dtype)
        out[~ro_stay] = arr_col[rorder[~ro_stay] - n_rstay].to(dtype)
        return out.contiguous()

    db = arr_rp[:, A + 1]
    arr_dist = (
        (db.view(-1, 1) >> torch.arange(A, device=device, dtype=torch.int32).view(1, A)) & 1
    ).to(torch.uint8)
    arr_gid = (arr_rp[:, A + 2].to(torch.int64) & 0xFFFFFFFF) | (
        arr_rp[:, A + 3].to(torch.int64) << 32
    )
    gs.rec_ent = new_rec_ent[rorder].contiguous()
    gs.rec_values = merge_rec(gs.rec_values[stay_r], arr_rp[:, :A], torch.int32)
    gs.rec_file = merge_rec(gs.rec_file[stay_r], arr_rp[:, A], torch.int32)
    gs.rec_dist = merge_rec(gs.rec_dist[stay_r], arr_dist, torch.uint8)
    gs.rec_gid = merge_rec(gs.rec_gid[stay_r], arr_gid, torch.int64)
    gs.rec_part = gs.ent_part[gs.rec_ent].contiguous()
    return gs


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
