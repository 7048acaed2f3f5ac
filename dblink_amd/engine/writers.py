"""Output writers: Parquet linkage chain + CSV diagnostics.

File contracts mirror the reference:
- ``linkage-chain.parquet``: one row per (iteration, partitionId) with the
  partition's linkage structure as list<list<record id>> — the Parquet form
  of ``LinkageState`` (``package.scala:94-96``), written buffered/appending
  (``util/BufferedRDDWriter.scala:30-75``).
- ``diagnostics.csv``: schema ``iteration, systemTime-ms, numObservedEntities,
  logLikelihood, popSize, aggDist-<attr>..., recDistortion-0..A``
  (``DiagnosticsWriter.scala:39-72``).
"""

from __future__ import annotations

import os
import time

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq


class LinkageChainWriter:
    """Buffered Parquet writer for linkage-structure samples.

    Each rank writes its own part files under ``linkage-chain.parquet/``;
    flushes every ``buffer_size`` appended iterations
    (reference default writeBufferSize=10, ``Sampler.scala:57``).
    """

    SCHEMA = pa.schema(
        [
            ("iteration", pa.int64()),
            ("partitionId", pa.int32()),
            ("linkageStructure", pa.list_(pa.list_(pa.string()))),
        ]
    )

    def __init__(self, output_path, rank=0, buffer_size=10, append=False):
        self.dir = os.path.join(output_path, "linkage-chain.parquet")
        os.makedirs(self.dir, exist_ok=True)
        self.rank = rank
        self.buffer_size = buffer_size
        self._rows = {"iteration": [], "partitionId": [], "linkageStructure": []}
        self._buffered = 0
        existing = [f for f in os.listdir(self.dir) if f.startswith(f"part-r{rank:05d}-")]
        if not append:
            for f in existing:
                os.remove(os.path.join(self.dir, f))
            self._file_ctr = 0
        else:
            self._file_ctr = len(existing)

    def append(self, iteration, partition_clusters):
        """partition_clusters: {pid -> list of clusters, each a list of record-id strings}."""
        for pid, clusters in sorted(partition_clusters.items()):
            self._rows["iteration"].append(int(iteration))
            self._rows["partitionId"].append(int(pid))
            self._rows["linkageStructure"].append(clusters)
        self._buffered += 1
        if self._buffered >= self.buffer_size:
            self.flush()

    def append_arrays(self, iteration, pid_list, pid_offsets, cluster_offsets,
                      record_ids, id_dictionary=None):
        """Vectorized append: one (iteration, pid) row per partition with the
        linkage structure built directly as Arrow nested lists (no per-cluster
        Python objects). ``record_ids`` is either a string array aligned with
        the flattened cluster layout, or — when ``id_dictionary`` is given —
        an int index array into that dictionary (stored dictionary-encoded:
        no per-sample string materialization at all)."""
        if id_dictionary is not None:
            if not hasattr(self, "_dict_arr") or self._dict_arr is None:
                self._dict_arr = pa.array(id_dictionary, type=pa.string())
            values = pa.DictionaryArray.from_arrays(
                pa.array(np.asarray(record_ids, dtype=np.int32)), self._dict_arr
            ).cast(pa.string())
        else:
            values = pa.array(record_ids, type=pa.string())
        inner = pa.ListArray.from_arrays(
            pa.array(cluster_offsets, type=pa.int32()), values
        )
        outer = pa.ListArray.from_arrays(pa.array(pid_offsets, type=pa.int32()), inner)
        batch = pa.table(
            {
                "iteration": pa.array([int(iteration)] * len(pid_list), type=pa.int64()),
                "partitionId": pa.array(pid_list, type=pa.int32()),
                "linkageStructure": outer,
            },
            schema=self.SCHEMA,
        )
        self._batches = getattr(self, "_batches", [])
        self._batches.append(batch)
        self._buffered += 1
        if self._buffered >= self.buffer_size:
            self.flush()

    def flush(self):
        batches = getattr(self, "_batches", [])
        if not self._rows["iteration"] and not batches:
            self._buffered = 0
            return
        tables = list(batches)
        if self._rows["iteration"]:
            tables.append(pa.table(self._rows, schema=self.SCHEMA))
        table = pa.concat_tables(tables) if len(tables) > 1 else tables[0]
        path = os.path.join(self.dir, f"part-r{self.rank:05d}-{self._file_ctr:05d}.parquet")
        pq.write_table(table, path)
        self._file_ctr += 1
        self._rows = {"iteration": [], "partitionId": [], "linkageStructure": []}
        self._batches = []
        self._buffered = 0

    def close(self):
        self.flush()


def read_linkage_chain(output_path, lower_iteration_cutoff=0):
    """Read all linkage-chain part files -> pyarrow Table."""
    d = os.path.join(output_path, "linkage-chain.parquet")
    files = sorted(
        os.path.join(d, f) for f in os.listdir(d) if f.endswith(".parquet")
    )
    if not files:
        return None
    tables = [pq.read_table(f) for f in files]
    table = pa.concat_tables(tables)
    if lower_iteration_cutoff > 0:
        import pyarrow.compute as pc

        table = table.filter(pc.greater_equal(table["iteration"], lower_iteration_cutoff))
    return table


class DiagnosticsWriter:
    """CSV diagnostics writer (DiagnosticsWriter.scala:32-80). Rank 0 only."""

    def __init__(self, output_path, cache, continue_chain=False):
        os.makedirs(output_path, exist_ok=True)
        self.path = os.path.join(output_path, "diagnostics.csv")
        self.cache = cache
        mode = "a" if continue_chain and os.path.exists(self.path) else "w"
        self._f = open(self.path, mode, encoding="utf-8")
        self._wrote_header = mode == "a"

    def _header(self):
        names = [ia.name for ia in self.cache.indexed_attributes]
        agg = ",".join(f"aggDist-{n}" for n in names)
        rec = ",".join(f"recDistortion-{k}" for k in range(len(names) + 1))
        return f"iteration,systemTime-ms,numObservedEntities,logLikelihood,popSize,{agg},{rec}\n"

    def write_row(self, state):
        if not self._wrote_header:
            self._f.write(self._header())
            self._wrote_header = True
        s = state.summary
        agg_per_attr = s.agg_distortions.sum(axis=1)
        row = [
            str(state.iteration),
            str(int(time.time() * 1000)),
            str(state.population_size - s.num_isolates),
            f"{s.log_likelihood:.9e}",
            str(state.population_size),
        ]
        row += [str(int(x)) for x in agg_per_attr]
        row += [str(int(x)) for x in s.rec_distortions]
        self._f.write(",".join(row) + "\n")
        self._f.flush()

    def close(self):
        self._f.close()
