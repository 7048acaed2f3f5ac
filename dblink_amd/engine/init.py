"""Deterministic initial state (parity: ``State.scala:205-334``).

- entity count per rank via the reference's bin-packing heuristic
  (``State.scala:230-250``), with ranks playing the role of input partitions
- record i (rank-local) links to entity ``i % numEntities``; entity values
  copied from the first linked record, missing values drawn from phi
  (``State.scala:269-293``)
- distortion indicators: observed and differing from the entity value
- leftover entities are isolates with fully random values (``:295-303``)
- KD-tree fitted on the initial entity values, then clusters are placed on
  their partitions (``:308-317``)
- distortion probabilities initialised at the prior mean (``:320``)
"""

from __future__ import annotations

import numpy as np

from ..models.distortion import DistortionProbs
from ..parallel import comm, migration
from .state import ChainState


def entities_per_rank(num_recs_per_rank, population_size):
    """Bin-packing heuristic (State.scala:230-250)."""
    counts = dict(enumerate(num_recs_per_rank))
    total_recs = sum(num_recs_per_rank)
    if population_size is None:
        population_size = total_recs
    n_ranks = len(counts)
    assert population_size >= n_ranks, "Too few entities. Need at least one entity per rank"
    extra = population_size - total_recs
    keys = list(counts.keys())
    i = 0
    while extra != 0:
        k = keys[i % n_ranks]
        i += 1
        if extra > 0:
            counts[k] += 1
            extra -= 1
        elif counts[k] > 1:
            counts[k] -= 1
            extra += 1
    return [counts[r] for r in range(n_ranks)], population_size


def deterministic_init(
    rec_values,
    rec_file,
    rec_gid,
    cache,
    partitioner,
    seed,
    population_size=None,
    rank=0,
    world_size=1,
):
    """Build the initial ChainState for this rank's slice of records."""
    R, A = rec_values.shape
    counts = comm.all_gather_object(R) if world_size > 1 else [R]
    ents_per_rank, pop_size = entities_per_rank(counts, population_size)
    E = ents_per_rank[rank]

    rng = np.random.Generator(np.random.Philox(key=seed + rank))

    # Entity e is created by its first linked record i = e (record i links to
    # entity i % E, State.scala:269-281); vectorized over entities.
    rec_ent = (np.arange(R, dtype=np.int64) % E) if E > 0 else np.empty(0, np.int64)
    k = min(E, R)
    ent_values = np.empty((E, A), dtype=np.int32)
    ent_values[:k] = rec_values[:k]
    for a in range(A):
        dist = cache.indexed_attributes[a].index.distribution
        miss = np.flatnonzero(ent_values[:k, a] < 0)
        if miss.size:
            ent_values[miss, a] = dist.sample(rng, miss.size)
        if E > k:  # isolated entities: fully random values
            ent_values[k:, a] = dist.sample(rng, E - k)
    rec_dist = (
        (rec_values >= 0) & (rec_values != ent_values[rec_ent])
    ).astype(np.uint8)

    new_seed = seed + world_size

    # Fit the partitioner on the GLOBAL entity values (KD-tree needs global
    # value counts). Values are gathered as numpy arrays; for the dataset
    # sizes this framework targets (<=10M entities x <=8 attrs of int32) this
    # is a few hundred MB once at startup.
    if world_size > 1:
        all_vals = comm.all_gather_object(ent_values)
        fit_vals = np.concatenate(all_vals)
    else:
        fit_vals = ent_values
    partitioner.fit(fit_vals)

    ent_part = partitioner.get_partition_ids(ent_values).astype(np.int32)

    state = ChainState(
        iteration=0,
        ent_values=ent_values,
        ent_part=ent_part,
        rec_values=np.ascontiguousarray(rec_values.astype(np.int32)),
        rec_file=np.ascontiguousarray(rec_file.astype(np.int32)),
        rec_ent=rec_ent,
        rec_dist=rec_dist,
        rec_gid=np.ascontiguousarray(np.asarray(rec_gid, dtype=np.int64)),
        dist_probs=DistortionProbs.from_prior_mean(
            [ia.distortion_prior for ia in cache.indexed_attributes], cache.num_files
        ),
        population_size=pop_size,
        start_seed=seed,
        current_seed=new_seed,
        summary=None,
    )
    # place clusters on their partitions' owner ranks
    migration.migrate(state, world_size)
    return state
