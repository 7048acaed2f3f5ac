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


def most_probable_clusters_fast(table):
    """Vectorized MPC over large chains: clusters are identified by an
    order-independent composite hash of their member record ids (two
    independent 64-bit hash aggregates + size — collisions are negligible).
    Returns {record_id -> (cluster member tuple, frequency)}."""
    import pandas as pd

    record_ids, cluster_gidx, iter_per_cluster, cluster_sizes, inner_offsets = (
        _flatten_chain(table)
    )
    num_samples = len(np.unique(np.asarray(table["iteration"].to_numpy())))
    rid_ser = pd.Series(record_ids)
    h1 = pd.util.hash_pandas_object(rid_ser, index=False).to_numpy()
    h2 = pd.util.hash_pandas_object(rid_ser + "#2", index=False).to_numpy()
    df = pd.DataFrame({
        "c": cluster_gidx,
        "h1": h1.astype(np.uint64),
        "h2": h2.astype(np.uint64),
    })
    agg = df.groupby("c", sort=False).agg(
        s1=("h1", "sum"), s2=("h2", "sum"), n=("h1", "size")
    )
    # composite cluster-content key
    key = (agg["s1"].to_numpy() ^ (agg["s2"].to_numpy() * np.uint64(0x9E3779B97F4A7C15))
           ) + agg["n"].to_numpy().astype(np.uint64)
    # frequency of each distinct cluster content
    kdf = pd.DataFrame({"key": key})
    counts = kdf.groupby("key", sort=False).size()
    freq_per_cluster = counts.loc[kdf["key"]].to_numpy() / num_samples
    # one representative instance per distinct key (first occurrence)
    first_idx = kdf.drop_duplicates("key").index.to_numpy()
    rep_of_key = dict(zip(kdf["key"].to_numpy()[first_idx], first_idx))
    # per record: best (max freq) cluster among those containing it
    rec_df = pd.DataFrame({
        "rid": record_ids,
        "key": kdf["key"].to_numpy()[cluster_gidx],
        "freq": freq_per_cluster[cluster_gidx],
    })
    best = rec_df.loc[rec_df.groupby("rid", sort=False)["freq"].idxmax()]
    out = {}
    fs_cache = {}
    for rid, k, f in zip(best["rid"], best["key"], best["freq"]):
        fs = fs_cache.get(k)
        if fs is None:
            ci = rep_of_key[k]
            lo, hi = inner_offsets[ci], inner_offsets[ci + 1]
            fs = frozenset(record_ids[lo:hi].tolist())
            fs_cache[k] = fs
        out[rid] = (fs, float(f))
    return out


def shared_most_probable_clusters_fast(table):
    """Vectorized sMPC (LinkageChain.scala:75-109 semantics)."""
    mpc = most_probable_clusters_fast(table)
    agg = defaultdict(set)
    for rid, (cluster, _) in mpc.items():
        agg[cluster].add(rid)
    return [set(v) for v in agg.values()]
