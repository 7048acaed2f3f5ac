"""CPU reference implementation of the partitioned Gibbs sweep.

This is the numerical ORACLE for the HIP kernels and the engine used for
CPU-only runs (BASELINE config #1). It reproduces, update for update, the
reference's per-partition sweep (``GibbsUpdates.scala``):

  1. link update per record          (:363-466, :473-530)
  2. entity-value update per (e, a)  (:533-727, :731-755)
  3. new partition id per entity + distortion update per linked record (:156-211, :324-359)

plus the summary-variable pass (:219-301) and the host-side theta update
(:305-320).

Four sampler variants, selected exactly as in ``ProjectStep.scala:53-58``:
PCG-I (collapsed values), PCG-II (collapsed ids + values), Gibbs, and
Gibbs-Sequential (no inverted index).

Per-partition RNG streams: ``numpy.random.Philox`` keyed by
(partition_id + current_seed), mirroring the reference's
``MersenneTwister(index + currentRandomSeed)`` scheme
(``GibbsUpdates.scala:139-147``); the seed advances by the GLOBAL number of
partitions per sweep, so chains are reproducible for a given seed and
partition count regardless of how partitions are spread over ranks.
"""

from __future__ import annotations

import math

import numpy as np

from ..models.distortion import DistortionProbs
from .state import ChainState, SummaryVars

LOG = math.log


class SamplerFlags:
    def __init__(self, collapsed_entity_ids=False, collapsed_entity_values=True, sequential=False):
        self.collapsed_entity_ids = collapsed_entity_ids
        self.collapsed_entity_values = collapsed_entity_values
        self.sequential = sequential

    @classmethod
    def for_sampler(cls, name):
        return {
            "PCG-I": cls(False, True, False),
            "PCG-II": cls(True, True, False),
            "Gibbs": cls(False, False, False),
            "Gibbs-Sequential": cls(False, False, True),
        }[name]


def _sample_weights(rng, weights):
    """Categorical draw from unnormalized float64 weights (list)."""
    total = 0.0
    for w in weights:
        total += w
    if total <= 0.0 or not math.isfinite(total):
        raise ValueError("zero or non-finite probability mass")
    u = rng.random() * total
    acc = 0.0
    for i, w in enumerate(weights):
        acc += w
        if u < acc:
            return i
    return len(weights) - 1


def _sample_dict(rng, weights_dict):
    total = sum(weights_dict.values())
    if total <= 0.0 or not math.isfinite(total):
        raise ValueError("zero or non-finite probability mass")
    u = rng.random() * total
    acc = 0.0
    last = None
    for k, w in weights_dict.items():
        acc += w
        last = k
        if u < acc:
            return k
    return last


def _alias_draw(rng, table):
    return int(table.sample(rng))


class _PartitionData:
    """One partition's working copy (mirrors updatePartition's buffers)."""

    __slots__ = (
        "ent_values", "rec_values", "rec_file", "rec_dist", "rec_gid",
        "rec_ent", "ent_recs",
    )


def _build_inverted_index(ent_values):
    """(attr, value) -> ascending array of local entity ids."""
    inv = {}
    E, A = ent_values.shape
    for a in range(A):
        col = ent_values[:, a]
        order = np.argsort(col, kind="stable")
        sv = col[order]
        bounds = np.flatnonzero(np.r_[True, sv[1:] != sv[:-1]])
        for bi, b in enumerate(bounds):
            e_end = bounds[bi + 1] if bi + 1 < len(bounds) else len(sv)
            inv[(a, int(sv[b]))] = np.sort(order[b:e_end])
    return inv


def _get_possible_entities(rec_vals, rec_dist, inv_index, num_entities, const_mask,
                           ent_values=None):
    """Set intersection over observed, non-distorted attributes
    (GibbsUpdates.scala:473-530). Returns (candidate ndarray, obs-distorted attr list).

    With ``ent_values`` the intersection enumerates the SMALLEST posting list
    and filters by direct value comparison — the posting list of (a, v) is
    exactly the entities whose a equals v, so membership needs no set ops.
    Yields the identical candidate array (same set, same order) as the
    isin-based intersection, just ~10x cheaper.
    """
    obs_dist = []
    sets = []
    attrs = []
    A = len(rec_vals)
    for a in range(A):
        v = rec_vals[a]
        if v >= 0:
            if not rec_dist[a]:
                sets.append(inv_index.get((a, int(v)), _EMPTY_I64))
                attrs.append(a)
            else:
                obs_dist.append(a)
    if not sets:
        return np.arange(num_entities), obs_dist
    if len(sets) == 1:
        return sets[0], obs_dist
    if ent_values is not None:
        base = min(range(len(sets)), key=lambda i: len(sets[i]))
        result = sets[base]
        for i, a in enumerate(attrs):
            if i == base or result.size == 0:
                continue
            result = result[ent_values[result, a] == rec_vals[a]]
        return result, obs_dist
    sets.sort(key=len)
    result = sets[0]
    for s in sets[1:]:
        result = result[np.isin(result, s, assume_unique=True)]
        if result.size == 0:
            break
    return result, obs_dist


_EMPTY_I64 = np.empty(0, dtype=np.int64)
_FAST_VALUE = True


def _cpu_fast_enabled():
    import os

    return os.environ.get("DBLINK_CPU_FAST", "1") != "0"


import os as _os  # noqa: E402

_PHASE_TIMERS = {} if _os.environ.get("DBLINK_PHASE_TIMERS", "") == "1" else None


def sweep_partition(
    rng,
    part,  # _PartitionData
    cache,
    dist_probs: DistortionProbs,
    flags: SamplerFlags,
):
    """Run link + value + distortion updates for one partition, in place.

    Returns nothing; ``part`` is mutated. New partition assignment is
    computed by the caller (needs the partitioner).
    """
    attrs = cache.indexed_attributes
    A = len(attrs)
    E = part.ent_values.shape[0]
    R = part.rec_values.shape[0]

    # --- phase 1: link updates (entity values fixed) --------------------------
    if not flags.sequential:
        inv_index = _build_inverted_index(part.ent_values)
    rec_ent = np.empty(R, dtype=np.int64)
    for r in range(R):
        rv = part.rec_values[r]
        rd = part.rec_dist[r]
        if flags.sequential:
            rec_ent[r] = _update_entity_id_seq(rng, rv, rd, part.ent_values, attrs)
        elif flags.collapsed_entity_ids:
            rec_ent[r] = _update_entity_id_collapsed(
                rng, rv, part.ent_values, attrs, dist_probs, int(part.rec_file[r])
            )
        else:
            rec_ent[r] = _update_entity_id(rng, rv, rd, part.ent_values, inv_index, attrs)
    part.rec_ent = rec_ent
    # entity -> linked record rows (insertion order = record row order,
    # matching LinksIndex.addLink which appends in record order)
    ent_recs = [[] for _ in range(E)]
    for r in range(R):
        ent_recs[rec_ent[r]].append(r)
    part.ent_recs = ent_recs

    # --- phase 2: entity value updates (links fixed) --------------------------
    for e in range(E):
        linked = ent_recs[e]
        new_vals = np.empty(A, dtype=np.int32)
        for a in range(A):
            ia = attrs[a]
            if flags.sequential:
                new_vals[a] = _update_entity_value_seq(rng, a, ia, part, linked)
            elif flags.collapsed_entity_values:
                new_vals[a] = _update_entity_value_collapsed(
                    rng, a, ia, part, linked, dist_probs
                )
            else:
                new_vals[a] = _update_entity_value(rng, a, ia, part, linked)
        part.ent_values[e] = new_vals

    # --- phase 3: distortion updates (links + values fixed) -------------------
    # Reference order: per entity, per linked record (GibbsUpdates.scala:205-209).
    for e in range(E):
        y = part.ent_values[e]
        for r in ent_recs[e]:
            _update_distortions(rng, part, r, y, attrs, dist_probs)


def _update_entity_id(rng, rv, rd, ent_values, inv_index, attrs):
    """Non-collapsed indexed link update (GibbsUpdates.scala:398-430)."""
    cands, obs_dist = _get_possible_entities(rv, rd, inv_index, ent_values.shape[0], None,
                                             ent_values=ent_values)
    if len(cands) == 0:
        raise RuntimeError("empty candidate set: state invariant violated")
    if not obs_dist:
        return int(cands[int(rng.integers(0, len(cands)))])
    weights = np.ones(len(cands), dtype=np.float64)
    for a in obs_dist:
        ia = attrs[a]
        x = int(rv[a])
        px = ia.index.probability_of(x)
        if ia.is_constant:
            weights *= px
        else:
            y = ent_values[cands, a]
            norms = ia.index.sim_norms[y]
            es = ia.index.exp_sim_many(x, y)
            weights *= norms * es * px
    idx = _sample_weights(rng, weights)
    return int(cands[idx])


def _update_entity_id_collapsed(rng, rv, ent_values, attrs, dist_probs, file_id):
    """PCG-II dense link update, distortions integrated out
    (GibbsUpdates.scala:363-395)."""
    E = ent_values.shape[0]
    weights = np.ones(E, dtype=np.float64)
    for a in range(len(attrs)):
        x = int(rv[a])
        if x < 0:
            continue
        ia = attrs[a]
        theta = dist_probs(a, file_id)
        px = ia.index.probability_of(x)
        y = ent_values[:, a]
        agree = (y == x).astype(np.float64) * (1.0 - theta)
        if ia.is_constant:
            weights *= agree + theta * px
        else:
            norms = ia.index.sim_norms[y]
            es = ia.index.exp_sim_many(x, y)
            weights *= agree + theta * px * norms * es
    return _sample_weights(rng, weights)


def _update_entity_id_seq(rng, rv, rd, ent_values, attrs):
    """Brute-force link update without index (GibbsUpdates.scala:434-466)."""
    E = ent_values.shape[0]
    weights = np.empty(E, dtype=np.float64)
    for e in range(E):
        w = 1.0
        for a in range(len(attrs)):
            x = int(rv[a])
            if x < 0:
                continue
            y = int(ent_values[e, a])
            if not rd[a]:
                if x != y:
                    w = 0.0
                    break
            else:
                ia = attrs[a]
                if ia.is_constant:
                    w *= ia.index.probability_of(x)
                else:
                    w *= (
                        ia.index.sim_norms[y]
                        * ia.index.exp_sim_of(x, y)
                        * ia.index.probability_of(x)
                    )
        weights[e] = w
    return _sample_weights(rng, weights)


def _observed_linked(part, linked, a):
    return [r for r in linked if part.rec_values[r, a] >= 0]


def _base_distribution(ia, k):
    """baseDistribution selection (GibbsUpdates.scala:584-586, 612-614)."""
    if (not ia.is_constant) and k > 0:
        return ia.index.sim_norm_dist(k), ("power", k)
    return ia.index.distribution, ("phi", 0)


def _base_prob(ia, kind, value_id):
    which, k = kind
    if which == "power":
        return ia.index.sim_norm_prob(value_id, k)
    return ia.index.probability_of(value_id)


def _update_entity_value_collapsed(rng, a, ia, part, linked, dist_probs):
    """PCG value update with distortions collapsed (GibbsUpdates.scala:576-599)."""
    obs = _observed_linked(part, linked, a)
    base, kind = _base_distribution(ia, len(obs))
    if not obs:
        return _alias_draw(rng, base)
    if (_FAST_VALUE and len(obs) == 1 and not ia.is_constant
            and ia.index.sim_row_len(int(part.rec_values[obs[0], a])) >= 128):
        # vectorized single-record case (the common one): one sim row, same
        # arithmetic and draw sequence as the dict path below (bitwise-equal;
        # _FAST_VALUE exists so tests can force the reference dict path)
        r = obs[0]
        theta = dist_probs(a, int(part.rec_file[r]))
        x = int(part.rec_values[r, a])
        px = ia.index.probability_of(x)
        normx = ia.index.sim_norms[x]
        cols, sims = ia.index.sim_index.row(x)
        w = sims.astype(np.float64, copy=True)
        pos = int(np.searchsorted(cols, x))
        if pos < len(cols) and cols[pos] == x:
            w[pos] = sims[pos] + (1.0 / theta - 1.0) / (px * normx)
        k = kind[1]
        weights = (ia.index.probs[cols] * ia.index.sim_norms[cols] ** k
                   / ia.index.sim_norm_total(k)) * (w - 1.0)
        cum = np.cumsum(weights)
        total = float(cum[-1]) if weights.size else 0.0
        if rng.random() < 1.0 / (1.0 + total):
            return _alias_draw(rng, base)
        if total <= 0.0 or not math.isfinite(total):
            raise ValueError("zero or non-finite probability mass")
        u = rng.random() * total
        i = int(np.searchsorted(cum, u, side="right"))
        return int(cols[min(i, len(cols) - 1)])
    vw = {}
    for r in obs:
        theta = dist_probs(a, int(part.rec_file[r]))
        x = int(part.rec_values[r, a])
        px = ia.index.probability_of(x)
        if ia.is_constant:
            w = 1.0 + (1.0 / theta - 1.0) / px
            vw[x] = w * vw.get(x, 1.0)
        else:
            normx = ia.index.sim_norms[x]
            cols, sims = ia.index.sim_index.row(x)
            for v, es in zip(cols.tolist(), sims.tolist()):
                w = es + (1.0 / theta - 1.0) / (px * normx) if v == x else es
                vw[v] = w * vw.get(v, 1.0)
    for v in vw:
        vw[v] = _base_prob(ia, kind, v) * (vw[v] - 1.0)
    total = sum(vw.values())
    if rng.random() < 1.0 / (1.0 + total):
        return _alias_draw(rng, base)
    return _sample_dict(rng, vw)


def _update_entity_value(rng, a, ia, part, linked):
    """Non-collapsed value update (GibbsUpdates.scala:605-646)."""
    obs = _observed_linked(part, linked, a)
    base, kind = _base_distribution(ia, len(obs))
    if not obs:
        return _alias_draw(rng, base)
    for r in obs:
        if not part.rec_dist[r, a]:
            return int(part.rec_values[r, a])  # deterministic copy
    if ia.is_constant:
        return _alias_draw(rng, base)
    vw = {}
    for r in obs:
        x = int(part.rec_values[r, a])
        cols, sims = ia.index.sim_index.row(x)
        for v, es in zip(cols.tolist(), sims.tolist()):
            vw[v] = es * vw.get(v, 1.0)
    for v in vw:
        vw[v] = _base_prob(ia, kind, v) * (vw[v] - 1.0)
    total = sum(vw.values())
    if rng.random() < 1.0 / (1.0 + total):
        return _alias_draw(rng, base)
    return _sample_dict(rng, vw)


def _update_entity_value_seq(rng, a, ia, part, linked):
    """Brute-force value update over the whole domain (GibbsUpdates.scala:652-698)."""
    obs = _observed_linked(part, linked, a)
    if not obs:
        return _alias_draw(rng, ia.index.distribution)
    for r in obs:
        if not part.rec_dist[r, a]:
            return int(part.rec_values[r, a])
    if ia.is_constant:
        return _alias_draw(rng, ia.index.distribution)
    V = ia.index.num_values
    weights = ia.index.probs.copy()
    for r in obs:
        x = int(part.rec_values[r, a])
        px = ia.index.probability_of(x)
        es = np.ones(V)
        cols, sims = ia.index.sim_index.row(x)
        es[cols] = sims
        weights *= es * ia.index.sim_norms * px
    return _sample_weights(rng, weights)


def _update_distortions(rng, part, r, y, attrs, dist_probs):
    """Per-record distortion resample (GibbsUpdates.scala:324-359)."""
    f = int(part.rec_file[r])
    for a in range(len(attrs)):
        x = int(part.rec_values[r, a])
        theta = dist_probs(a, f)
        if x < 0:
            part.rec_dist[r, a] = rng.random() < theta
        elif x == int(y[a]):
            # self_mass = phi(x) [* norm(x) * expsim(x, x)] — static per index
            pr1 = theta * attrs[a].index.self_mass[x]
            pr0 = 1.0 - theta
            p = pr1 / (pr1 + pr0) if (pr1 + pr0) != 0.0 else 0.0
            part.rec_dist[r, a] = rng.random() < p
        else:
            part.rec_dist[r, a] = True


def _exp_sim_pairs(index, xs, ys):
    """Vectorized exp_sim over elementwise (x, y) pairs (sparse CSR lookup;
    values not in x's similarity row have exp-similarity 1)."""
    import scipy.sparse as sp

    si = index.sim_index
    mat = getattr(index, "_scipy_csr", None)
    if mat is None:
        V = index.num_values
        mat = sp.csr_matrix((si.expsim, si.col, si.row_ptr), shape=(V, V))
        index._scipy_csr = mat
    es = np.asarray(mat[np.asarray(xs), np.asarray(ys)]).ravel()
    return np.where(es > 0.0, es, 1.0)


def _summary_native(state: ChainState, cache):
    """OpenMP summary reduction (summary_cpu). Integer outputs are exact;
    the log-likelihood terms match the numpy path but the float summation
    ORDER differs (OMP blocks vs numpy pairwise), so the value can differ
    in ulps — it is a diagnostic series everywhere. Returns None when the
    native extension is unavailable or DBLINK_NATIVE_SUMMARY=0."""
    import os as _os

    if _os.environ.get("DBLINK_NATIVE_SUMMARY", "1") == "0":
        return None
    from .. import ops

    if not ops.have_native() or not hasattr(ops.native(), "summary_cpu"):
        return None
    import torch

    cat = getattr(cache, "_summary_cat", None)
    if cat is None:
        attrs = cache.indexed_attributes
        voff = np.zeros(len(attrs) + 1, dtype=np.int64)
        np.cumsum([ia.index.num_values for ia in attrs], out=voff[1:])
        probs = np.concatenate([ia.index.probs for ia in attrs])
        log_probs = np.concatenate([ia.index.log_probs for ia in attrs])
        sim_norms = np.concatenate([ia.index.sim_norms for ia in attrs])
        rp_cat = np.zeros(voff[-1] + 1, dtype=np.int64)
        col_parts, es_parts = [], []
        base = 0
        for a, ia in enumerate(attrs):
            if ia.is_constant:
                rp_cat[voff[a] + 1:voff[a + 1] + 1] = base
            else:
                si = ia.index.sim_index
                rp_cat[voff[a] + 1:voff[a + 1] + 1] = base + si.row_ptr[1:]
                col_parts.append(si.col.astype(np.int32))
                es_parts.append(si.expsim.astype(np.float64))
                base += len(si.col)
        col_cat = (np.concatenate(col_parts) if col_parts
                   else np.zeros(0, np.int32))
        es_cat = (np.concatenate(es_parts) if es_parts
                  else np.zeros(0, np.float64))
        cst = np.array([1 if ia.is_constant else 0 for ia in attrs],
                       dtype=np.uint8)
        cat = tuple(
            torch.from_numpy(np.ascontiguousarray(x))
            for x in (probs, log_probs, sim_norms, voff, rp_cat, col_cat,
                      es_cat, cst))
        cache._summary_cat = cat
    probs_t, lp_t, sn_t, voff_t, rp_t, col_t, es_t, cst_t = cat
    ll, iso, agg, hist = ops.native().summary_cpu(
        torch.from_numpy(np.ascontiguousarray(state.rec_values)),
        torch.from_numpy(np.ascontiguousarray(state.rec_dist)),
        torch.from_numpy(np.ascontiguousarray(state.rec_file)),
        torch.from_numpy(np.ascontiguousarray(state.rec_ent)),
        torch.from_numpy(np.ascontiguousarray(state.ent_values)),
        probs_t, lp_t, sn_t, voff_t, rp_t, col_t, es_t, cst_t,
        int(cache.num_files))
    s = SummaryVars.zeros(len(cache.indexed_attributes), cache.num_files)
    s.log_likelihood = float(ll)
    s.num_isolates = int(iso)
    s.agg_distortions = agg.numpy()
    s.rec_distortions = hist.numpy()
    return s


def compute_summary(state: ChainState, cache, dist_probs) -> SummaryVars:
    """Summary variables (GibbsUpdates.scala:219-301) for the LOCAL shard.

    The Beta-prior terms (driver-side in the reference) are added by the
    caller after the cross-rank reduction — see ``add_prior_terms``.
    """
    native = _summary_native(state, cache)
    if native is not None:
        return native
    attrs = cache.indexed_attributes
    A = len(attrs)
    F = cache.num_files
    s = SummaryVars.zeros(A, F)

    loglik = 0.0
    # entity prior terms, all entities (gathering the cached log table is
    # bitwise-identical to logging the gathered probs)
    for a in range(A):
        log_probs = attrs[a].index.log_probs
        loglik += float(np.sum(log_probs[state.ent_values[:, a]]))
    # isolates
    linked_counts = np.bincount(state.rec_ent, minlength=state.num_entities)
    s.num_isolates = int(np.sum(linked_counts == 0))
    # record distortion terms
    rec_ndist = np.zeros(state.num_records, dtype=np.int64)
    for a in range(A):
        ia = attrs[a]
        d = state.rec_dist[:, a].astype(bool)
        rec_ndist += d
        if not np.any(d):
            continue
        files = state.rec_file[d]
        np.add.at(s.agg_distortions[a], files, 1)
        x = state.rec_values[d, a]
        obs = x >= 0
        if np.any(obs):
            xo = x[obs]
            if ia.is_constant:
                loglik += float(np.sum(ia.index.log_probs[xo]))
            else:
                y = state.ent_values[state.rec_ent[d], a][obs]
                es = _exp_sim_pairs(ia.index, xo, y)
                loglik += float(
                    np.sum(np.log(ia.index.probs[xo] * ia.index.sim_norms[y] * es))
                )
    np.add.at(s.rec_distortions, rec_ndist, 1)
    s.log_likelihood = loglik
    return s


def add_prior_terms(summary: SummaryVars, cache, dist_probs) -> None:
    """Driver-side Beta-prior log-likelihood terms (GibbsUpdates.scala:286-293)."""
    priors = [a.distortion_prior for a in cache.indexed_attributes]
    file_sizes = np.array([cache.file_sizes[f] for f in cache.file_ids], dtype=np.float64)
    for a, p in enumerate(priors):
        for f in range(cache.num_files):
            theta = dist_probs(a, f)
            nd = float(summary.agg_distortions[a, f])
            summary.log_likelihood += (p.alpha + nd - 1.0) * LOG(theta) + (
                p.beta + file_sizes[f] - nd - 1.0
            ) * LOG(1.0 - theta)


def sweep(
    state: ChainState,
    cache,
    partitioner,
    flags: SamplerFlags,
    num_partitions: int,
):
    """One full local sweep: per-partition updates + new partition ids.

    Does NOT exchange clusters between ranks (see parallel.migration) and does
    not compute summaries. Advances ``current_seed`` by num_partitions.

    The indexed variants (PCG-I and plain Gibbs) run through the vectorized
    whole-rank sweep (``cpu_fast.sweep_fast``) unless DBLINK_CPU_FAST=0; the
    per-record loops below remain the numerical oracle and serve PCG-II and
    Gibbs-Sequential.
    """
    if _cpu_fast_enabled() and not flags.sequential:
        dense_ok = True
        if flags.collapsed_entity_ids:
            # PCG-II is dense over each partition's entities (like the
            # reference); only vectorize when the largest R_p x E_p block
            # is affordable, else fall through to the per-record loops
            ec = np.bincount(state.ent_part, minlength=num_partitions)
            rc = np.bincount(state.ent_part[state.rec_ent], minlength=num_partitions)
            dense_ok = (flags.collapsed_entity_values
                        and int((ec.astype(np.int64) * rc).max()) <= 30_000_000)
        if not flags.collapsed_entity_ids or dense_ok:
            from ..parallel import comm
            from .cpu_fast import sweep_fast

            rank = comm.rank_world()[0] if comm.is_distributed() else 0
            sweep_fast(state, cache, partitioner, num_partitions, rank=rank,
                       timers=_PHASE_TIMERS,
                       collapsed=flags.collapsed_entity_values,
                       collapsed_ids=flags.collapsed_entity_ids)
            return
    if state.num_entities and int(state.ent_part.max()) >= num_partitions:
        raise RuntimeError(
            "state has partition ids beyond the partitioner's range "
            "(resuming with an unfitted partitioner?)"
        )
    ent_ptr, rec_ptr = state.partition_offsets(num_partitions)
    new_rec_ent = np.empty_like(state.rec_ent)
    for p in range(num_partitions):
        e0, e1 = int(ent_ptr[p]), int(ent_ptr[p + 1])
        r0, r1 = int(rec_ptr[p]), int(rec_ptr[p + 1])
        if e1 == e0 and r1 == r0:
            continue
        if e1 == e0:
            raise RuntimeError(f"partition {p} has records but no entities")
        rng = np.random.Generator(np.random.Philox(key=state.current_seed + p))
        part = _PartitionData()
        part.ent_values = state.ent_values[e0:e1]
        part.rec_values = state.rec_values[r0:r1]
        part.rec_file = state.rec_file[r0:r1]
        part.rec_dist = state.rec_dist[r0:r1]
        part.rec_gid = state.rec_gid[r0:r1]
        sweep_partition(rng, part, cache, state.dist_probs, flags)
        new_rec_ent[r0:r1] = part.rec_ent + e0
    state.rec_ent = new_rec_ent
    # new partition assignment from the freshly sampled values
    state.ent_part = partitioner.get_partition_ids(state.ent_values).astype(np.int32)
    state.current_seed += num_partitions
    state.iteration += 1


class CpuEngine:
    """CPU reference engine: one Markov transition per ``step`` call.

    Transition order parity (``State.nextState``, State.scala:78-99):
    theta update from the previous summary -> per-partition sweep ->
    cluster migration -> summary recomputation.

    Theta is recomputed redundantly on every rank from the all-reduced
    distortion counts with an iteration-keyed Philox stream, so no broadcast
    is needed (SURVEY.md §2.3 note).
    """

    def __init__(self, cache, partitioner, world_size=1, rank=0):
        self.cache = cache
        self.partitioner = partitioner
        self.world_size = world_size
        self.rank = rank
        self.num_partitions = partitioner.num_partitions
        self.phase_timers = _PHASE_TIMERS is not None

    # -- theta ---------------------------------------------------------------

    def _update_dist_probs(self, state):
        from ..models.distortion import update_dist_probs

        rng = np.random.Generator(
            np.random.Philox(key=(state.start_seed << 20) + state.iteration)
        )
        file_sizes = np.array(
            [self.cache.file_sizes[f] for f in self.cache.file_ids], dtype=np.int64
        )
        state.dist_probs = update_dist_probs(
            state.summary.agg_distortions,
            [ia.distortion_prior for ia in self.cache.indexed_attributes],
            file_sizes,
            rng,
        )

    # -- summary reduction ---------------------------------------------------

    def _reduce_summary(self, state, local: SummaryVars) -> SummaryVars:
        import torch

        from ..parallel import comm

        A, F = local.agg_distortions.shape
        packed = np.concatenate(
            [
                np.array([local.log_likelihood], dtype=np.float64),
                np.array([local.num_isolates], dtype=np.float64),
                local.agg_distortions.reshape(-1).astype(np.float64),
                local.rec_distortions.astype(np.float64),
            ]
        )
        t = torch.from_numpy(packed)
        comm.all_reduce_sum_(t)
        packed = t.numpy()
        out = SummaryVars(
            num_isolates=int(round(packed[1])),
            log_likelihood=float(packed[0]),
            agg_distortions=packed[2 : 2 + A * F].reshape(A, F).astype(np.int64),
            rec_distortions=packed[2 + A * F :].astype(np.int64),
        )
        return out

    # -- transition ----------------------------------------------------------

    def step(self, state: ChainState, flags: SamplerFlags):
        from ..parallel import migration

        # The GPU engine keeps records in stable identity order; the CPU
        # sweep slices positionally by partition, so canonicalize once when
        # picking up such a state (idempotent for CPU-produced states).
        if not getattr(state, "cpu_sorted", False):
            state.sort_by_partition()
        self._update_dist_probs(state)
        sweep(state, self.cache, self.partitioner, flags, self.num_partitions)
        migration.migrate(state, self.world_size)
        local = compute_summary(state, self.cache, state.dist_probs)
        summary = self._reduce_summary(state, local)
        add_prior_terms(summary, self.cache, state.dist_probs)
        state.summary = summary
        return state

    def phase_times(self):
        """Cumulative {phase: ms} from the vectorized sweep
        (DBLINK_PHASE_TIMERS=1; same contract as GpuEngine.phase_times)."""
        return dict(_PHASE_TIMERS or {})

    def initial_summary(self, state: ChainState):
        local = compute_summary(state, self.cache, state.dist_probs)
        summary = self._reduce_summary(state, local)
        add_prior_terms(summary, self.cache, state.dist_probs)
        state.summary = summary

    # -- linkage structure (State.getLinkageStructure) -----------------------

    def linkage_arrays(self, state: ChainState):
        """Vectorized linkage structure for this rank: returns
        (partition_ids int32 [P_local], clusters-per-partition offsets
        int64 [P_local+1], records-per-cluster offsets int64 [C+1],
        record gids int64 [R]) — Arrow-ready, no per-cluster Python loops."""
        pid = state.ent_part[state.rec_ent]
        order = np.lexsort((state.rec_ent, pid))
        sorted_pid = pid[order]
        sorted_ent = state.rec_ent[order]
        gids = state.rec_gid[order]
        cb = np.flatnonzero(np.r_[True, sorted_ent[1:] != sorted_ent[:-1]])
        cluster_offsets = np.r_[cb, len(sorted_ent)].astype(np.int64)
        cluster_pid = sorted_pid[cb] if len(cb) else np.empty(0, np.int32)
        # every owned partition (one with entities) gets a row, even when all
        # its entities are isolated — the reference's getLinkageStructure
        # emits an empty cluster list per partition (State.scala:102-112)
        owned = np.unique(state.ent_part)
        counts = np.zeros(owned.size, dtype=np.int64)
        if cluster_pid.size:
            uniq, ccounts = np.unique(cluster_pid, return_counts=True)
            counts[np.searchsorted(owned, uniq)] = ccounts
        pid_offsets = np.r_[0, np.cumsum(counts)].astype(np.int64)
        return owned.astype(np.int32), pid_offsets, cluster_offsets, gids

    def linkage_structure(self, state: ChainState, rec_id_of=None):
        """{pid -> list of clusters (lists of record-id strings)} for this rank
        (dict form, used by tests and small runs)."""
        if rec_id_of is None:
            rec_id_of = getattr(self, "rec_id_of", None) or (lambda gid: str(gid))
        pid_list, pid_offsets, cluster_offsets, gids = self.linkage_arrays(state)
        out = {int(p): [] for p in np.unique(state.ent_part)}
        for pi, p in enumerate(pid_list):
            for ci in range(int(pid_offsets[pi]), int(pid_offsets[pi + 1])):
                lo, hi = int(cluster_offsets[ci]), int(cluster_offsets[ci + 1])
                out[int(p)].append([rec_id_of(int(g)) for g in gids[lo:hi]])
        return out
