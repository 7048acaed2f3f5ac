"""Cross-rank migration of entity clusters after a sweep.

Replaces the reference's Spark shuffle (``GibbsUpdates.scala:144-150`` +
``util/HardPartitioner.scala``): partition id -> owning rank is
``pid % world_size`` (the HardPartitioner contract), and each entity travels
together with its linked records.

Implementation: rows are grouped by destination rank with one stable argsort,
then exchanged with six all_to_all_v calls (entity meta/values, record
values/file/dist/gid). On RCCL this is a direct pairwise all-to-all over
xGMI; on gloo (CPU tests) it falls back to isend/irecv.
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

    def a2a(arr_np, dtype):
        t = torch.from_numpy(np.ascontiguousarray(arr_np))
        out, counts = comm.all_to_all_v(t, send_counts_list, device)
        return out.numpy(), counts

    send_counts_list = [int(c) for c in send_counts_e]
    ent_sorted_vals = state.ent_values[order_e]
    ent_sorted_part = state.ent_part[order_e]
    new_ent_values, recv_counts_e = a2a(ent_sorted_vals, np.int32)
    new_ent_part, _ = a2a(ent_sorted_part, np.int32)

    send_counts_list = [int(c) for c in send_counts_r]
    rec_ent_local = pos_in_dest[state.rec_ent][order_r].astype(np.int64)
    new_rec_entlocal, recv_counts_r = a2a(rec_ent_local, np.int64)
    new_rec_values, _ = a2a(state.rec_values[order_r], np.int32)
    new_rec_file, _ = a2a(state.rec_file[order_r], np.int32)
    new_rec_dist, _ = a2a(state.rec_dist[order_r], np.uint8)
    new_rec_gid, _ = a2a(state.rec_gid[order_r], np.int64)

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
