"""Posterior linkage-chain queries (parity: ``LinkageChain.scala:27-212``).

Host-side numpy/pyarrow implementations — these run once per project over the
saved chain, not per iteration, so they are not performance-critical.
"""

from __future__ import annotations

import os
from collections import Counter, defaultdict

import numpy as np

from ..engine.writers import read_linkage_chain


def _iter_clusters(table):
    """Yield (iteration, partitionId, clusters) from a linkage-chain table."""
    iters = table["iteration"].to_pylist()
    pids = table["partitionId"].to_pylist()
    structs = table["linkageStructure"].to_pylist()
    yield from zip(iters, pids, structs)


def most_probable_clusters(table):
    """{record_id -> (cluster frozenset, frequency)} (LinkageChain.scala:52-64)."""
    iterations = set(table["iteration"].to_pylist())
    num_samples = len(iterations)
    freq = Counter()
    for _, _, clusters in _iter_clusters(table):
        for cluster in clusters:
            if cluster:
                freq[frozenset(cluster)] += 1
    best = {}
    for cluster, count in freq.items():
        f = count / num_samples
        for rid in cluster:
            cur = best.get(rid)
            if cur is None or f > cur[1]:
                best[rid] = (cluster, f)
    return best


def shared_most_probable_clusters(table):
    """Steorts et al. sMPC point estimate -> list of clusters (sets of record
    ids) (LinkageChain.scala:75-109)."""
    mpc = most_probable_clusters(table)
    agg = defaultdict(set)
    for rid, (cluster, _) in mpc.items():
        agg[cluster].add(rid)
    return [set(v) for v in agg.values()]


def partition_sizes(table):
    """{iteration -> {pid -> number of clusters}} (LinkageChain.scala:118-128)."""
    out = defaultdict(dict)
    for it, pid, clusters in _iter_clusters(table):
        out[it][pid] = len(clusters)
    return dict(out)


def cluster_size_distribution(table):
    """{iteration -> {size -> count}} (LinkageChain.scala:137-154)."""
    out = defaultdict(Counter)
    for it, _, clusters in _iter_clusters(table):
        for cluster in clusters:
            out[it][len(cluster)] += 1
    return {it: dict(c) for it, c in out.items()}


def save_cluster_size_distribution(dist, output_path):
    """CSV: header `iteration,0..maxSize` (LinkageChain.scala:162-185)."""
    path = os.path.join(output_path, "cluster-size-distribution.csv")
    max_size = max((max(c) for c in dist.values() if c), default=0)
    with open(path, "w", encoding="utf-8") as f:
        f.write("iteration," + ",".join(str(k) for k in range(max_size + 1)) + "\n")
        for it in sorted(dist):
            row = [str(dist[it].get(k, 0)) for k in range(max_size + 1)]
            f.write(f"{it}," + ",".join(row) + "\n")
    return path


def save_partition_sizes(sizes, output_path):
    """CSV: header `iteration,<pid...>` (LinkageChain.scala:193-211)."""
    path = os.path.join(output_path, "partition-sizes.csv")
    pids = sorted({p for m in sizes.values() for p in m})
    with open(path, "w", encoding="utf-8") as f:
        f.write("iteration," + ",".join(str(p) for p in pids) + "\n")
        for it in sorted(sizes):
            f.write(f"{it}," + ",".join(str(sizes[it].get(p, 0)) for p in pids) + "\n")
    return path


def save_clusters_csv(clusters, path):
    """One cluster per line, comma-separated record ids (analysis/package.scala:99-108)."""
    with open(path, "w", encoding="utf-8") as f:
        for cluster in clusters:
            f.write(", ".join(sorted(cluster)) + "\n")
    return path


def read_clusters_csv(path):
    out = []
    with open(path, "r", encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if line:
                out.append({x.strip() for x in line.split(",")})
    return out


def load_chain(output_path, lower_iteration_cutoff=0):
    d = os.path.join(output_path, "linkage-chain.parquet")
    if not os.path.isdir(d):
        return None
    table = read_linkage_chain(output_path, lower_iteration_cutoff)
    if table is None or table.num_rows == 0:
        return None
    return table


def _flatten_chain(table):
    """Flatten the nested linkage column -> (record_ids, cluster_gidx,
    iteration_per_cluster, cluster_sizes). cluster_gidx is a dense global
    cluster-instance index across all rows."""
    import pyarrow as pa

    col = table["linkageStructure"].combine_chunks()
    if isinstance(col, pa.ChunkedArray):
        col = col.combine_chunks()
    iters = np.asarray(table["iteration"].to_numpy())
    # outer level: clusters per (iteration, pid) row
    outer = col
    outer_offsets = np.asarray(outer.offsets)
    clusters = outer.flatten()  # list<string> per cluster
    inner_offsets = np.asarray(clusters.offsets)
    record_ids = clusters.flatten().to_numpy(zero_copy_only=False)
    n_clusters = len(clusters)
    cluster_sizes = np.diff(inner_offsets)
    # iteration of each cluster: repeat row iteration by clusters-per-row
    clusters_per_row = np.diff(outer_offsets)
    iter_per_cluster = np.repeat(iters, clusters_per_row)
    cluster_gidx = np.repeat(np.arange(n_clusters, dtype=np.int64), cluster_sizes)
    return record_ids, cluster_gidx, iter_per_cluster, cluster_sizes, inner_offsets


# widest (code | count | index) packing that fits a signed int64 composite
# sort key; beyond it _mpc_core falls back to a 3-key lexsort
_PACK_BITS = 62


def _native_mpc():
    """The OpenMP MPC helpers, or None (numpy fallback). Their integer
    arithmetic is bitwise-identical to the numpy expressions they stand
    in for (DBLINK_NATIVE_MPC=0 forces the numpy path)."""
    import os

    if os.environ.get("DBLINK_NATIVE_MPC", "1") == "0":
        return None
    try:
        from .. import ops

        if ops.have_native() and hasattr(ops.native(), "mpc_cluster_keys"):
            return ops.native()
    except Exception:
        pass
    return None


def _mpc_keys_numpy(codes, off64, cluster_sizes):
    """Order-independent cluster content keys: two splitmix-derived per-code
    hashes summed per cluster mod 2^64 (zero-prefixed cumsum + segment
    diff), combined with the cluster size. mpc_cluster_keys computes the
    exact same integers natively."""

    def mix(x):  # splitmix64 finalizer
        x = (x + np.uint64(0x9E3779B97F4A7C15))
        x ^= x >> np.uint64(30)
        x *= np.uint64(0xBF58476D1CE4E5B9)
        x ^= x >> np.uint64(27)
        x *= np.uint64(0x94D049BB133111EB)
        x ^= x >> np.uint64(31)
        return x

    h1 = mix(codes)
    # second independent sum: a nonlinear remix of h1 (h1 is already
    # well-mixed, so one multiply round suffices for pair independence)
    h2 = (h1 ^ (h1 >> np.uint64(29))) * np.uint64(0xD6E8FEB86659FD93)
    h2 ^= h2 >> np.uint64(32)
    n_flat = len(h1)
    cc1 = np.empty(n_flat + 1, dtype=np.uint64)
    cc1[0] = np.uint64(0)
    np.cumsum(h1, out=cc1[1:])
    cc2 = np.empty(n_flat + 1, dtype=np.uint64)
    cc2[0] = np.uint64(0)
    np.cumsum(h2, out=cc2[1:])
    s1 = cc1[off64[1:]] - cc1[off64[:-1]]
    s2 = cc2[off64[1:]] - cc2[off64[:-1]]
    return (s1 ^ (s2 * np.uint64(0x9E3779B97F4A7C15))) + cluster_sizes.astype(
        np.uint64)


def _mpc_core(table):
    """Vectorized MPC internals over large chains: clusters are identified
    by an order-independent composite hash of their member record ids (two
    independent 64-bit mixes of the records' dictionary codes + size —
    collisions are negligible). All grouping runs on integer codes with
    contiguous-segment reductions (clusters are contiguous in the flattened
    layout), so no per-entry string hashing or string groupbys. Returns
    (best_codes, best_keys, best_freq, first_idx, inner_offsets, codes_i,
    uniq_rids, kcode)."""
    import pandas as pd
    import pyarrow as pa

    # flatten in Arrow and dictionary-encode the member ids natively: no
    # 50M-element python-object string arrays, no string factorize
    col = table["linkageStructure"].combine_chunks()
    if isinstance(col, pa.ChunkedArray):
        col = col.combine_chunks()
    clusters = col.flatten()
    inner_offsets = np.asarray(clusters.offsets)
    cluster_sizes = np.diff(inner_offsets)
    members = clusters.flatten()
    if isinstance(members, pa.ChunkedArray):
        members = members.combine_chunks()
    enc = members.dictionary_encode()
    codes_i = np.asarray(enc.indices)
    uniq_rids = np.asarray(enc.dictionary.to_numpy(zero_copy_only=False))
    num_samples = len(np.unique(np.asarray(table["iteration"].to_numpy())))
    codes = codes_i.astype(np.uint64)
    off64 = inner_offsets.astype(np.int64)
    nat = _native_mpc()
    if nat is not None:
        import torch

        c32 = np.ascontiguousarray(codes_i, dtype=np.int32)
        if not c32.flags.writeable:  # arrow buffers are read-only
            c32 = c32.copy()
        codes32 = torch.from_numpy(c32)
        off_t = torch.from_numpy(np.ascontiguousarray(off64))
        key = nat.mpc_cluster_keys(codes32, off_t).numpy().view(np.uint64)
    else:
        with np.errstate(over="ignore"):
            key = _mpc_keys_numpy(codes, off64, cluster_sizes)
    # frequency of each distinct cluster content
    kcode, _ = pd.factorize(key)
    kcounts = np.bincount(kcode)
    freq_per_cluster = kcounts[kcode] / num_samples
    # one representative cluster instance per distinct key: first occurrence
    if nat is not None:
        first_idx = nat.first_occurrence(
            __import__("torch").from_numpy(
                np.ascontiguousarray(kcode, dtype=np.int64)),
            int(len(kcounts))).numpy()
    else:
        # reversed fancy assignment leaves the FIRST occurrence (no sort)
        first_idx = np.zeros(len(kcounts), dtype=np.int64)
        first_idx[kcode[::-1]] = np.arange(len(kcode) - 1, -1, -1,
                                           dtype=np.int64)
    # per record-code: best (max count, first entry on ties) cluster entry
    # via ONE composite int64 sort: code | count | inverted entry index
    n_ent = len(codes)
    idx_bits = max(1, int(n_ent - 1).bit_length())
    cnt_bits = max(1, int(num_samples).bit_length())
    code_bits = max(1, int(len(uniq_rids) - 1).bit_length())
    if code_bits + cnt_bits + idx_bits <= _PACK_BITS:
        if nat is not None:
            import torch

            combo = nat.mpc_combo(
                codes32, off_t,
                torch.from_numpy(kcounts[kcode].astype(np.int64)),
                int(cnt_bits), int(idx_bits)).numpy()
        else:
            entry_counts = np.repeat(kcounts[kcode].astype(np.int64),
                                     cluster_sizes)
            inv_idx = (np.int64(n_ent - 1) - np.arange(n_ent, dtype=np.int64))
            combo = ((codes.astype(np.int64) << np.int64(cnt_bits + idx_bits))
                     | (entry_counts << np.int64(idx_bits)) | inv_idx)
        combo.sort()
        dec_code = combo >> np.int64(cnt_bits + idx_bits)
        last = np.flatnonzero(np.r_[dec_code[1:] != dec_code[:-1], True])
        best_entries = (np.int64(n_ent - 1)
                        - (combo[last]
                           & ((np.int64(1) << np.int64(idx_bits)) - 1)))
        best_codes = dec_code[last]
    else:  # astronomically large chains: 3-key lexsort instead of packing
        entry_counts = np.repeat(kcounts[kcode].astype(np.int64),
                                 cluster_sizes)
        order = np.lexsort((-np.arange(n_ent, dtype=np.int64), entry_counts,
                            codes.astype(np.int64)))
        c_sorted = codes[order].astype(np.int64)
        last = np.flatnonzero(np.r_[c_sorted[1:] != c_sorted[:-1], True])
        best_entries = order[last]
        best_codes = c_sorted[last]
    # cluster of each winning entry: searchsorted over the offsets (only
    # ~n_records winners, so no 50M-element repeat of cluster ids needed)
    best_cluster = np.searchsorted(off64, best_entries, side="right") - 1
    best_keys = kcode[best_cluster]
    best_freq = freq_per_cluster[best_cluster]
    return (best_codes, best_keys, best_freq, first_idx, inner_offsets,
            codes_i, uniq_rids, kcode)


def most_probable_clusters_fast(table):
    """{record_id -> (most-probable cluster frozenset, frequency)} — the
    per-record dict view over _mpc_core (LinkageChain.scala:52-64)."""
    (best_codes, best_keys, best_freq, first_idx, inner_offsets, codes_i,
     uniq_rids, _kcode) = _mpc_core(table)
    uniq_list = uniq_rids.tolist()
    bc = best_codes.tolist()
    bk = best_keys.tolist()
    bf = best_freq.tolist()
    out = {}
    fs_cache = {}
    for rc, k, f in zip(bc, bk, bf):
        fs = fs_cache.get(k)
        if fs is None:
            ci = first_idx[k]  # representative cluster instance for this key
            lo, hi = inner_offsets[ci], inner_offsets[ci + 1]
            fs = frozenset(uniq_rids[codes_i[lo:hi]].tolist())
            fs_cache[k] = fs
        out[uniq_list[rc]] = (fs, float(f))
    return out


def shared_most_probable_clusters_fast(table):
    """Vectorized sMPC (LinkageChain.scala:75-109 semantics): records
    grouped by the content key of their most-probable cluster — built
    straight from the integer core, no per-record dict or frozensets."""
    (best_codes, best_keys, _bf, _fi, _io, _ci, uniq_rids, _kc) = (
        _mpc_core(table))
    order = np.argsort(best_keys, kind="stable")
    sk = best_keys[order]
    bounds = np.flatnonzero(np.r_[True, sk[1:] != sk[:-1]])
    bounds = np.r_[bounds, len(sk)]
    members = uniq_rids[best_codes[order]]
    return [set(members[bounds[i]:bounds[i + 1]].tolist())
            for i in range(len(bounds) - 1)]
