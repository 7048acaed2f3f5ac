"""Markov-chain state: flat structure-of-arrays over entities and records.

The reference keeps an RDD of per-partition ``EntRecCluster`` case classes
(``State.scala:56-68``, ``package.scala:80-88``). The MI355X design keeps a
rank-local flat SoA, sorted by partition id, so the whole sweep runs as
batched kernels over contiguous partition segments:

- ent_values  int32 [E, A]   latent entity attribute value ids
- ent_part    int32 [E]      partition id of each entity (sorted ascending)
- rec_values  int32 [R, A]   record attribute value ids (-1 = missing)
- rec_file    int32 [R]      dense file id
- rec_ent     int64 [R]      local index of the linked entity
- rec_dist    uint8 [R, A]   distortion indicators
- rec_gid     int64 [R]      global record row (names resolved at write time)

Records are stored grouped behind their entity's partition (recomputed after
each migration). Entity clusters (entity + its linked records) migrate
together between ranks, like the reference's Spark shuffle
(``GibbsUpdates.scala:144-150``).
"""

from __future__ import annotations

import os
import pickle
from dataclasses import dataclass

import numpy as np


@dataclass
class SummaryVars:
    """Parity: ``package.scala:116-119``."""

    num_isolates: int = 0
    log_likelihood: float = 0.0
    agg_distortions: np.ndarray = None  # int64 [A, F]
    rec_distortions: np.ndarray = None  # int64 [A+1] histogram

    @classmethod
    def zeros(cls, num_attributes, num_files):
        return cls(
            0,
            0.0,
            np.zeros((num_attributes, num_files), dtype=np.int64),
            np.zeros(num_attributes + 1, dtype=np.int64),
        )


def _shard_rank(fname):
    """Rank number of a partitions-state shard filename, or None."""
    prefix, suffix = "partitions-state-rank", ".npz"
    if fname.startswith(prefix) and fname.endswith(suffix):
        try:
            return int(fname[len(prefix):-len(suffix)])
        except ValueError:
            return None
    return None


@dataclass
class ChainState:
    iteration: int
    ent_values: np.ndarray
    ent_part: np.ndarray
    rec_values: np.ndarray
    rec_file: np.ndarray
    rec_ent: np.ndarray
    rec_dist: np.ndarray
    rec_gid: np.ndarray
    dist_probs: "object"  # DistortionProbs
    population_size: int  # GLOBAL population size (across ranks)
    start_seed: int
    current_seed: int
    summary: SummaryVars = None
    rng_state: object = None  # host RNG (theta draws), rank 0 only

    @property
    def num_entities(self):
        return int(self.ent_values.shape[0])

    @property
    def num_records(self):
        return int(self.rec_values.shape[0])

    @property
    def num_attributes(self):
        return int(self.rec_values.shape[1])

    def sort_by_partition(self):
        """Re-sort entities by partition id and records behind their entity.

        Keeps rec_ent consistent. Stable, so within-partition order is
        preserved (matters for reproducibility of per-partition RNG streams).
        """
        self.cpu_sorted = True
        from .cpu_fast import _stable_argsort

        order = _stable_argsort(self.ent_part, int(self.ent_part.max()) + 1
                                if self.ent_part.size else 1)
        inv = np.empty_like(order)
        inv[order] = np.arange(order.size)
        self.ent_values = np.ascontiguousarray(self.ent_values[order])
        self.ent_part = np.ascontiguousarray(self.ent_part[order])
        new_rec_ent = inv[self.rec_ent]
        rec_order = _stable_argsort(new_rec_ent, order.size)
        self.rec_ent = np.ascontiguousarray(new_rec_ent[rec_order])
        self.rec_values = np.ascontiguousarray(self.rec_values[rec_order])
        self.rec_file = np.ascontiguousarray(self.rec_file[rec_order])
        self.rec_dist = np.ascontiguousarray(self.rec_dist[rec_order])
        self.rec_gid = np.ascontiguousarray(self.rec_gid[rec_order])

    def partition_offsets(self, num_partitions):
        """(ent_ptr[P+1], rec_ptr[P+1]) over the sorted arrays."""
        ent_ptr = np.searchsorted(self.ent_part, np.arange(num_partitions + 1))
        rec_part = self.ent_part[self.rec_ent]
        rec_ptr = np.searchsorted(rec_part, np.arange(num_partitions + 1))
        return ent_ptr.astype(np.int64), rec_ptr.astype(np.int64)

    # ---- persistence (two-file contract, State.scala:122-193) ----------------

    def save(self, output_path, rank=0, world_size=None, extra=None):
        os.makedirs(output_path, exist_ok=True)
        driver = {
            "iteration": self.iteration,
            "dist_probs": self.dist_probs.probs,
            "population_size": self.population_size,
            "start_seed": self.start_seed,
            "current_seed": self.current_seed,
            "summary": {
                "num_isolates": self.summary.num_isolates,
                "log_likelihood": self.summary.log_likelihood,
                "agg_distortions": self.summary.agg_distortions,
                "rec_distortions": self.summary.rec_distortions,
            },
            "rng_state": self.rng_state,
        }
        if world_size is not None:
            driver["num_shards"] = int(world_size)
        if extra:
            driver.update(extra)
        if rank == 0:
            with open(os.path.join(output_path, "driver-state"), "wb") as f:
                pickle.dump(driver, f)
            if world_size is not None:
                # a previous run in the same outputPath may have used a larger
                # world size; its extra shard files must not leak into a resume
                for fname in list(os.listdir(output_path)):
                    r = _shard_rank(fname)
                    if r is not None and r >= world_size:
                        os.remove(os.path.join(output_path, fname))
        np.savez(
            os.path.join(output_path, f"partitions-state-rank{rank:05d}.npz"),
            ent_values=self.ent_values,
            ent_part=self.ent_part,
            rec_values=self.rec_values,
            rec_file=self.rec_file,
            rec_ent=self.rec_ent,
            rec_dist=self.rec_dist,
            rec_gid=self.rec_gid,
        )

    @classmethod
    def exists(cls, output_path):
        return os.path.exists(os.path.join(output_path, "driver-state")) and any(
            f.startswith("partitions-state-rank") for f in os.listdir(output_path)
        ) if os.path.isdir(output_path) else False

    @classmethod
    def load(cls, output_path, rank=0, world_size=1):
        from ..models.distortion import DistortionProbs

        with open(os.path.join(output_path, "driver-state"), "rb") as f:
            driver = pickle.load(f)
        shards = sorted(
            f for f in os.listdir(output_path) if _shard_rank(f) is not None
        )
        num_shards = driver.get("num_shards")
        if num_shards is not None:
            # only read the shards the saving run actually wrote; anything
            # beyond is stale debris from an earlier, wider run
            shards = [s for s in shards if _shard_rank(s) < num_shards]
            if len(shards) != num_shards:
                raise FileNotFoundError(
                    f"saved state in {output_path} expects {num_shards} partition "
                    f"shard(s) but found {len(shards)}"
                )
        # Re-shard if world size changed: each rank takes every k-th shard and
        # re-sorts; partition ownership is re-established by the next migration.
        mine = [s for i, s in enumerate(shards) if i % world_size == rank]
        arrays = {k: [] for k in ["ent_values", "ent_part", "rec_values", "rec_file", "rec_ent", "rec_dist", "rec_gid"]}
        ent_base = 0
        for s in mine:
            z = np.load(os.path.join(output_path, s))
            for k in arrays:
                if k == "rec_ent":
                    arrays[k].append(z[k] + ent_base)
                else:
                    arrays[k].append(z[k])
            ent_base += z["ent_values"].shape[0]
        def cat(k, dtype, ncol=None):
            if arrays[k]:
                return np.concatenate(arrays[k])
            if ncol is not None:
                return np.empty((0, ncol), dtype=dtype)
            return np.empty(0, dtype=dtype)
        A = driver["dist_probs"].shape[0]
        state = cls(
            iteration=driver["iteration"],
            ent_values=cat("ent_values", np.int32, A),
            ent_part=cat("ent_part", np.int32),
            rec_values=cat("rec_values", np.int32, A),
            rec_file=cat("rec_file", np.int32),
            rec_ent=cat("rec_ent", np.int64),
            rec_dist=cat("rec_dist", np.uint8),
            rec_gid=cat("rec_gid", np.int64),
            dist_probs=DistortionProbs(driver["dist_probs"]),
            population_size=driver["population_size"],
            start_seed=driver["start_seed"],
            current_seed=driver["current_seed"],
            summary=SummaryVars(
                driver["summary"]["num_isolates"],
                driver["summary"]["log_likelihood"],
                driver["summary"]["agg_distortions"],
                driver["summary"]["rec_distortions"],
            ),
            rng_state=driver.get("rng_state"),
        )
        # the fitted partition function travels with the driver state
        # (parity: State.scala broadcasts/saves the PartitionFunction)
        state.saved_partitioner = driver.get("partitioner")
        # A single-rank resume must reproduce the saved engine order verbatim
        # (the GPU engine keeps records in stable identity order; re-sorting
        # would make a resumed chain diverge bitwise from a continuing one).
        # Only a world-size change needs re-sorting: concatenated shards are
        # not globally partition-sorted.
        if len(mine) != 1 or world_size != len(shards):
            state.sort_by_partition()
        return state
