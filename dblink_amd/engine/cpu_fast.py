"""Vectorized CPU sweeps for PCG-I (flagship), plain Gibbs and PCG-II.

Mirrors the GPU engine's phase structure in numpy — whole-rank flat arrays,
counter-based Philox draws keyed by (seed, iteration, phase, element id) —
instead of the per-record python loops of ``cpu_engine.sweep_partition``.
The per-record implementation remains the numerical oracle (and serves
Gibbs-Sequential plus oversized PCG-II partitions); this path produces the
same conditional distributions (checked by the cross-variant
posterior-consistency and fast-vs-reference band tests plus per-phase
conditional TV tests) but a different — still deterministic — draw
sequence.

Parity: indexed link update ``GibbsUpdates.scala:398-430, 473-530``; dense
collapsed link ``:363-395``; value updates ``:576-599`` (collapsed) /
``:605-646`` (plain); distortion update ``:324-359``.
"""

from __future__ import annotations

import numpy as np

_PH_LINK = np.uint32(1)
_PH_DIST = np.uint32(2)
_PH_VAL = np.uint32(3)

_M0 = np.uint64(0xD2511F53)
_M1 = np.uint64(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)


# Dense PCG-II blocks larger than this switch from expression arithmetic
# (fast while glibc serves the temporaries from brk) to a cached in-place
# workspace (above ~50 MB per temporary glibc mmaps/munmaps every
# allocation and page-table churn dominates the sweep).
_DENSE_WS_THRESHOLD = 8_000_000

# Attribute domains up to this size get dense [V, V] log link-weight tables
# (see _FastModel); larger domains use searchsorted over flat sim keys.
# The two paths are bitwise-identical (log_norms[y] + log(expsim) either way).
_DENSE_LOGSIM_MAX_V = 2048


def _stable_argsort(keys, k=None):
    """Stable argsort of non-negative int keys. Small key ranges route to the
    native O(n + k) counting sort (identical permutation to numpy's stable
    mergesort, which measured ~25% of the real-RLdata10000 stationary
    sweep); anything else falls back to np.argsort(kind="stable")."""
    if k is not None and keys.size and k <= 8 * keys.size + 65536:
        from .. import ops

        if ops.have_native() and hasattr(ops.native(), "counting_argsort_cpu"):
            import torch

            keys64 = np.ascontiguousarray(keys, dtype=np.int64)
            return ops.native().counting_argsort_cpu(
                torch.from_numpy(keys64), int(k)).numpy()
    return np.argsort(keys, kind="stable")


def _philox_uniform4(seed, iteration, phase, ids, draw, rank=0):
    """Philox4x32-10 keyed uniforms in (0, 1): FOUR independent streams per
    id (the four 32-bit output words) for the price of one keyed evaluation.
    State lives in masked uint64 lanes with preallocated scratch — the 10
    rounds run allocation-free."""
    ids = np.ascontiguousarray(ids, dtype=np.uint64)
    n = len(ids)
    M32 = np.uint64(0xFFFFFFFF)
    c0 = ids & M32
    c1 = ids >> np.uint64(32)
    c2 = np.full(n, np.uint64(np.uint32(draw)))
    c3 = np.full(n, np.uint64(np.uint32(iteration)))
    k0 = np.uint64(np.uint64(seed) & M32)
    k1 = np.uint64(
        int((np.uint64(seed) >> np.uint64(32)) & M32)
        ^ ((int(phase) * 0x9E3779B1) & 0xFFFFFFFF)
        ^ ((rank * 0x85EBCA6B) & 0xFFFFFFFF)
    )
    W0, W1 = np.uint64(0x9E3779B9), np.uint64(0xBB67AE85)
    M0, M1 = _M0, _M1
    p0 = np.empty(n, dtype=np.uint64)
    p1 = np.empty(n, dtype=np.uint64)
    t = np.empty(n, dtype=np.uint64)
    for _ in range(10):
        np.multiply(c0, M0, out=p0)
        np.multiply(c2, M1, out=p1)
        # c0' = hi1 ^ c1 ^ k0 ; c1' = lo1 ; c2' = hi0 ^ c3 ^ k1 ; c3' = lo0
        np.right_shift(p1, np.uint64(32), out=t)
        np.bitwise_xor(t, c1, out=t)
        np.bitwise_xor(t, k0, out=t)
        np.bitwise_and(p1, M32, out=c1)
        c0, t = t, c0
        np.right_shift(p0, np.uint64(32), out=t)
        np.bitwise_xor(t, c3, out=t)
        np.bitwise_xor(t, k1, out=t)
        np.bitwise_and(p0, M32, out=c3)
        c2, t = t, c2
        k0 = (k0 + W0) & M32
        k1 = (k1 + W1) & M32
    to_u = lambda c: ((c & M32).astype(np.float64) + 0.5) * (2.0 ** -32)  # noqa: E731
    return to_u(c0), to_u(c1), to_u(c2), to_u(c3)


def _philox_uniform(seed, iteration, phase, ids, draw, rank=0):
    return _philox_uniform4(seed, iteration, phase, ids, draw, rank)[0]


def _philox_dense(seed, iteration, phase, n, rank=0):
    """n iid uniforms keyed by dense position: one Philox evaluation yields
    four output words, so positions are packed four per counter."""
    m = (n + 3) // 4
    w = _philox_uniform4(seed, iteration, phase, np.arange(m, dtype=np.int64), 0, rank)
    return np.stack(w, axis=1).reshape(-1)[:n]


def _alias_draw_vec(u1, u2, prob, alias):
    n = len(prob)
    idx = np.minimum((u1 * n).astype(np.int64), n - 1)
    return np.where(u2 < prob[idx], idx, alias[idx])


def _ragged_expand(lengths):
    """(flat->group index, group offsets) for ragged group lengths."""
    off = np.zeros(len(lengths) + 1, dtype=np.int64)
    np.cumsum(lengths, out=off[1:])
    grp = np.repeat(np.arange(len(lengths)), lengths)
    return grp, off


class _FastModel:
    """Per-cache static arrays for the vectorized sweep (built once)."""

    def __init__(self, cache):
        attrs = cache.indexed_attributes
        self.attrs = attrs
        self.A = len(attrs)
        self.Vmax = max(ia.index.num_values for ia in attrs)
        self.log_norms = [
            None if ia.is_constant else np.log(ia.index.sim_norms) for ia in attrs
        ]
        # flat sorted (x * V + y) keys with log(expsim) values: pair lookups
        # become one searchsorted over the nnz array (the scipy fancy-index
        # path builds intermediate matrices and costs ~3x more)
        self.sim_keys = []
        self.sim_logvals = []
        self.num_values = [ia.index.num_values for ia in attrs]
        for ia in attrs:
            if ia.is_constant:
                self.sim_keys.append(None)
                self.sim_logvals.append(None)
            else:
                si = ia.index.sim_index
                V = ia.index.num_values
                rows = np.repeat(np.arange(V, dtype=np.int64),
                                 np.diff(si.row_ptr))
                keys = rows * V + si.col.astype(np.int64)
                order = np.argsort(keys)
                self.sim_keys.append(keys[order])
                self.sim_logvals.append(
                    np.log(si.expsim.astype(np.float64))[order])
        # dense [V, V] log link-weight tables, log_norms[y] folded in, for
        # small domains (the common RLdata-like case): one gather replaces
        # a searchsorted per (record, candidate) pair
        self.sim_logdense = []
        for ia in attrs:
            if ia.is_constant or ia.index.num_values > _DENSE_LOGSIM_MAX_V:
                self.sim_logdense.append(None)
            else:
                si = ia.index.sim_index
                V = ia.index.num_values
                tbl = np.tile(np.log(ia.index.sim_norms), (V, 1))
                rows = np.repeat(np.arange(V, dtype=np.int64),
                                 np.diff(si.row_ptr))
                tbl[rows, si.col] += np.log(si.expsim.astype(np.float64))
                self.sim_logdense.append(tbl)
        self.phi_tables = [ia.index.distribution for ia in attrs]
        # cached power-dist normalizers Z_k, k = 0..kmax (Z_0 unused)
        self.kmax = max(1, min(16, max(
            getattr(ia.index, "_max_cached_power", 1) for ia in attrs
        )))
        self.pow_totals = []
        for ia in attrs:
            if ia.is_constant:
                self.pow_totals.append(None)
            else:
                self.pow_totals.append(np.array(
                    [1.0] + [ia.index.sim_norm_total(kv)
                             for kv in range(1, self.kmax + 1)]
                ))


def get_fast_model(cache):
    got = getattr(cache, "_fast_model", None)
    if got is None:
        got = _FastModel(cache)
        cache._fast_model = got
    return got


def _pcg2_native_model(fm):
    """Concatenated f64 model arrays (voff layout) for the native PCG-II
    kernel; built once per cache."""
    got = getattr(fm, "_pcg2_native", None)
    if got is not None:
        return got
    attrs, A = fm.attrs, fm.A
    voff = np.zeros(A + 1, dtype=np.int64)
    phi, norm, cols, sims = [], [], [], []
    row_ptr = [np.zeros(1, dtype=np.int64)]
    nnz = 0
    for a, ia in enumerate(attrs):
        V = ia.index.num_values
        voff[a + 1] = voff[a] + V
        phi.append(ia.index.probs)
        if ia.is_constant:
            norm.append(np.ones(V))
            row_ptr.append(np.full(V, nnz, dtype=np.int64))
        else:
            si = ia.index.sim_index
            norm.append(ia.index.sim_norms)
            row_ptr.append(nnz + si.row_ptr[1:])
            cols.append(si.col)
            sims.append(si.expsim)
            nnz += int(si.row_ptr[-1])
    norm_cat = np.concatenate(norm)
    got = fm._pcg2_native = {
        "voff": voff,
        "phi": np.ascontiguousarray(np.concatenate(phi)),
        "norm": np.ascontiguousarray(norm_cat),
        "log_norm": np.ascontiguousarray(np.log(norm_cat)),
        "row_ptr": np.ascontiguousarray(np.concatenate(row_ptr)),
        "col": np.ascontiguousarray(
            np.concatenate(cols) if cols else np.empty(0, np.int32)),
        "expsim": np.ascontiguousarray(
            np.concatenate(sims) if sims else np.empty(0)),
        "const": np.array([1 if ia.is_constant else 0 for ia in attrs],
                          dtype=np.uint8),
    }
    with np.errstate(divide="ignore"):
        got["log_expsim"] = np.ascontiguousarray(np.log(got["expsim"]))
    return got


def _distortion_native(state, fm, seed, it, rank):
    """OpenMP distortion resample (distortion_update_cpu): bitwise-identical
    to the numpy phase-3 block (same packed Philox stream, same f64
    expressions). Returns the [R, A] uint8 array or None when unavailable."""
    import os as _os

    if _os.environ.get("DBLINK_NATIVE_DIST", "1") == "0":
        return None
    from .. import ops

    if not ops.have_native() or not hasattr(ops.native(), "distortion_update_cpu"):
        return None
    import torch

    sm = getattr(fm, "_self_mass_cat", None)
    if sm is None:
        sm = torch.from_numpy(np.ascontiguousarray(np.concatenate(
            [ia.index.self_mass for ia in fm.attrs])))
        voff = np.zeros(fm.A + 1, dtype=np.int64)
        np.cumsum([ia.index.num_values for ia in fm.attrs], out=voff[1:])
        fm._self_mass_cat = sm
        fm._voff_cat = torch.from_numpy(voff)
    return ops.native().distortion_update_cpu(
        torch.from_numpy(np.ascontiguousarray(state.rec_values)),
        torch.from_numpy(np.ascontiguousarray(state.rec_file)),
        torch.from_numpy(np.ascontiguousarray(state.rec_ent)),
        torch.from_numpy(np.ascontiguousarray(state.ent_values)),
        torch.from_numpy(np.ascontiguousarray(state.dist_probs.probs)),
        sm, fm._voff_cat, int(seed), int(it), int(rank),
    ).numpy()


def _value_native(state, fm, collapsed, seed, it, rank, rorder_all):
    """OpenMP entity-value update (value_update_cpu): same Philox streams and
    f64 op order as the numpy block below (draws agree except ulp-boundary
    CDF ties). Returns (new_ev, fallback pair list) or None when the native
    extension is unavailable or disabled."""
    import os as _os

    if _os.environ.get("DBLINK_NATIVE_VALUE", "1") == "0":
        return None
    from .. import ops

    if not ops.have_native() or not hasattr(ops.native(), "value_update_cpu"):
        return None
    import torch

    C = ops.native()
    nm = _pcg2_native_model(fm)
    vt = getattr(fm, "_value_native_tabs", None)
    if vt is None:
        attrs = fm.attrs
        phi_prob, phi_alias, pow_prob, pow_alias = [], [], [], []
        ptot = np.ones((fm.A, fm.kmax + 1))
        for a, ia in enumerate(attrs):
            t = fm.phi_tables[a]
            phi_prob.append(torch.from_numpy(
                np.ascontiguousarray(t.prob.astype(np.float64))))
            phi_alias.append(torch.from_numpy(
                np.ascontiguousarray(t.alias.astype(np.int64))))
            if ia.is_constant:
                pow_prob.append(torch.empty(0, dtype=torch.float64))
                pow_alias.append(torch.empty(0, dtype=torch.int64))
            else:
                pp = np.concatenate([
                    ia.index.sim_norm_dist(kv).prob.astype(np.float64)
                    for kv in range(1, fm.kmax + 1)])
                pa = np.concatenate([
                    ia.index.sim_norm_dist(kv).alias.astype(np.int64)
                    for kv in range(1, fm.kmax + 1)])
                pow_prob.append(torch.from_numpy(np.ascontiguousarray(pp)))
                pow_alias.append(torch.from_numpy(np.ascontiguousarray(pa)))
                ptot[a, :] = fm.pow_totals[a][: fm.kmax + 1]
        vt = fm._value_native_tabs = (
            phi_prob, phi_alias, pow_prob, pow_alias,
            torch.from_numpy(np.ascontiguousarray(ptot)))
    phi_prob, phi_alias, pow_prob, pow_alias, ptot_t = vt
    starts = np.searchsorted(
        state.rec_ent[rorder_all], np.arange(state.num_entities + 1)
    ).astype(np.int64)
    t = torch.from_numpy
    new_ev, fb = C.value_update_cpu(
        t(np.ascontiguousarray(state.rec_values)),
        t(np.ascontiguousarray(state.rec_file.astype(np.int32))),
        t(np.ascontiguousarray(state.rec_dist)),
        t(np.ascontiguousarray(state.rec_ent)),
        t(np.ascontiguousarray(rorder_all.astype(np.int64))), t(starts),
        t(np.ascontiguousarray(state.ent_values)),
        t(np.ascontiguousarray(state.dist_probs.probs)),
        t(nm["phi"]), t(nm["norm"]), t(nm["voff"]), t(nm["row_ptr"]),
        t(nm["col"]), t(nm["expsim"]), t(nm["const"]),
        phi_prob, phi_alias, pow_prob, pow_alias, ptot_t,
        int(fm.kmax), 1 if collapsed else 0, int(seed), int(it), int(rank),
    )
    A = fm.A
    fallback = [(int(p) // A, int(p) % A) for p in fb.numpy()]
    return new_ev.numpy(), fallback


def _link_indexed_native(state, fm, num_partitions, seed, it):
    """OpenMP indexed PCG-I/Gibbs link update (pcg1_link_cpu); None when the
    native extension is unavailable or disabled."""
    import os as _os

    if _os.environ.get("DBLINK_NATIVE_PCG1", "1") == "0":
        return None
    from .. import ops

    if not ops.have_native() or not hasattr(ops.native(), "pcg1_link_cpu"):
        return None
    import torch

    C = ops.native()
    nm = _pcg2_native_model(fm)
    ent_ptr = np.searchsorted(
        state.ent_part, np.arange(num_partitions + 1)).astype(np.int64)
    perms, ptrs = [], []
    for a, ia in enumerate(fm.attrs):
        Va = ia.index.num_values
        keys = state.ent_part.astype(np.int64) * Va + state.ent_values[:, a]
        perm = _stable_argsort(keys, num_partitions * Va).astype(np.int32)
        cnt = np.bincount(keys, minlength=num_partitions * Va)
        ptr = np.zeros(num_partitions * Va + 1, np.int64)
        ptr[1:] = np.cumsum(cnt)
        perms.append(torch.from_numpy(perm))
        ptrs.append(torch.from_numpy(ptr))
    rec_part = state.ent_part[state.rec_ent].astype(np.int32)
    t = torch.from_numpy
    out, n_empty = C.pcg1_link_cpu(
        t(np.ascontiguousarray(state.rec_values)),
        t(np.ascontiguousarray(state.rec_dist)),
        t(np.ascontiguousarray(rec_part)),
        t(np.ascontiguousarray(state.rec_gid)),
        t(np.ascontiguousarray(state.rec_ent)),
        t(np.ascontiguousarray(state.ent_values)),
        t(ent_ptr), t(nm["log_norm"]), t(nm["voff"]), t(nm["row_ptr"]),
        t(nm["col"]), t(nm["log_expsim"]), t(nm["const"]),
        perms, ptrs, int(seed), int(it),
    )
    if n_empty:
        raise RuntimeError(
            f"{n_empty} empty candidate sets in link update (invariant violated)"
        )
    return out.numpy()


def _link_dense_collapsed_native(state, fm, num_partitions, seed, it, rank):
    """OpenMP f64 log-space PCG-II link update (link_dense_cpu.cpp); returns
    None when the native extension is unavailable or disabled."""
    import os as _os

    if _os.environ.get("DBLINK_NATIVE_PCG2", "1") == "0":
        return None
    from .. import ops

    if not ops.have_native() or not hasattr(ops.native(), "pcg2_link_cpu"):
        return None
    import torch

    C = ops.native()
    nm = _pcg2_native_model(fm)
    attrs = fm.attrs
    ent_ptr = np.searchsorted(
        state.ent_part, np.arange(num_partitions + 1)).astype(np.int64)
    perms, ptrs = [], []
    for a, ia in enumerate(attrs):
        if ia.is_constant:
            continue
        Va = ia.index.num_values
        keys = state.ent_part.astype(np.int64) * Va + state.ent_values[:, a]
        perm = _stable_argsort(keys, num_partitions * Va).astype(np.int32)
        cnt = np.bincount(keys, minlength=num_partitions * Va)
        ptr = np.zeros(num_partitions * Va + 1, np.int64)
        ptr[1:] = np.cumsum(cnt)
        perms.append(torch.from_numpy(perm))
        ptrs.append(torch.from_numpy(ptr))
    u_rec = _philox_dense(seed, it, _PH_LINK, state.num_records, rank)
    rec_part = state.ent_part[state.rec_ent].astype(np.int32)
    t = torch.from_numpy
    out = C.pcg2_link_cpu(
        t(np.ascontiguousarray(state.rec_values)),
        t(np.ascontiguousarray(state.rec_file.astype(np.int32))),
        t(np.ascontiguousarray(rec_part)),
        t(np.ascontiguousarray(state.ent_values)),
        t(ent_ptr), t(np.ascontiguousarray(state.dist_probs.probs)),
        t(nm["phi"]), t(nm["norm"]), t(nm["log_norm"]), t(nm["voff"]),
        t(nm["row_ptr"]), t(nm["col"]), t(nm["expsim"]), t(nm["const"]),
        perms, ptrs, t(np.ascontiguousarray(u_rec)),
    )
    return out.numpy()


def _link_dense_collapsed(state, fm, theta_ra, num_partitions, seed, it, rank):
    """PCG-II link update: distortions integrated out, dense over each
    partition's entities (GibbsUpdates.scala:363-395). Quadratic like the
    reference; the caller size-gates it. When the native extension is built,
    the whole update runs as the threaded f64 log-space kernel of
    ``link_dense_cpu.cpp`` (~20x the numpy path, no underflow handling
    needed); the numpy path below remains the fallback.

    Works entirely in per-shape CACHED workspaces with in-place updates —
    at R_p x E_p ~ 25M elements, per-sweep 100 MB temporaries previously
    spent more time in the allocator (madvise churn) than in compute.
    Weight products stay in f32 (matching the GPU kernels' precision; f64
    beyond 8 attributes for underflow headroom); the agreement bonus is
    applied by sparse scatter instead of a dense mask product."""
    native = _link_dense_collapsed_native(state, fm, num_partitions, seed, it, rank)
    if native is not None:
        return native
    attrs, A = fm.attrs, fm.A
    ev = state.ent_values
    rv = state.rec_values
    ent_ptr = np.searchsorted(state.ent_part, np.arange(num_partitions + 1))
    rec_part = state.ent_part[state.rec_ent]
    rec_ptr = np.searchsorted(rec_part, np.arange(num_partitions + 1))
    u_rec = _philox_dense(seed, it, _PH_LINK, state.num_records, rank)
    new_rec_ent = np.empty(state.num_records, dtype=np.int64)
    wdt = np.float32 if A <= 8 else np.float64
    ws = getattr(fm, "_dense_ws", None)
    if ws is None:
        ws = fm._dense_ws = {}
    for pid in range(num_partitions):
        r0, r1 = int(rec_ptr[pid]), int(rec_ptr[pid + 1])
        e0, e1 = int(ent_ptr[pid]), int(ent_ptr[pid + 1])
        Rp, Ep = r1 - r0, e1 - e0
        if Rp == 0:
            continue
        # Two arithmetic styles by block size: plain numpy expressions give
        # the allocator brk-reusable ~25 MB temporaries and run fastest for
        # mid-size partitions, but at >= ~50 MB per temporary glibc switches
        # to mmap/munmap per allocation and the sweep drowns in page-table
        # churn — there the cached in-place workspace wins by an order of
        # magnitude.
        big = Rp * Ep > _DENSE_WS_THRESHOLD
        if big:
            key = (Rp, Ep)
            if key not in ws:
                ws[key] = (np.empty((Rp, Ep), dtype=wdt),   # weight product
                           np.empty((Rp, Ep), dtype=wdt),   # per-attr work
                           np.empty((Rp, Ep), dtype=wdt),   # agree scratch
                           np.empty((Rp, Ep), dtype=bool),  # agreement mask
                           np.empty((Rp, Ep), dtype=np.float64))  # CDF
            wprod, work, scr, eq, cum = ws[key]
            wprod.fill(wdt(1.0))
        else:
            wprod = np.ones((Rp, Ep), dtype=wdt)
        for a in range(A):
            ia = attrs[a]
            xo = rv[r0:r1, a]
            obs = xo >= 0
            if not obs.any():
                continue
            y = ev[e0:e1, a]
            th = theta_ra[r0:r1, a].astype(wdt)
            px = ia.index.probs[np.maximum(xo, 0)].astype(wdt)
            base_r = th * px
            if not ia.is_constant:
                si = ia.index.sim_index
                # dense exp-sim block: ragged fill of each record's sim row
                # onto the partition's entities, grouped by entity value
                eorder = np.argsort(y, kind="stable")
                ys = y[eorder]
                lo = si.row_ptr[np.maximum(xo, 0)]
                ln = np.where(obs, si.row_ptr[np.maximum(xo, 0) + 1] - lo, 0)
                grp, goff = _ragged_expand(ln)
                fi = lo[grp] + (np.arange(goff[-1]) - goff[grp])
                fcol = si.col[fi]
                fval = si.expsim[fi]
                a_lo = np.searchsorted(ys, fcol)
                cnt = np.searchsorted(ys, fcol, side="right") - a_lo
                g2, off2 = _ragged_expand(cnt)
                epos = eorder[a_lo[g2] + (np.arange(off2[-1]) - off2[g2])]
                norms_w = ia.index.sim_norms[y].astype(wdt)
            if big:
                if ia.is_constant:
                    np.copyto(work, base_r[:, None])
                else:
                    work.fill(wdt(1.0))
                    work[grp[g2], epos] = fval[g2]
                    work *= norms_w[None, :]
                    work *= base_r[:, None]
                np.equal(xo[:, None], y[None, :], out=eq)
                np.multiply(eq, (wdt(1.0) - th)[:, None], out=scr)
                work += scr
            else:
                agree = (xo[:, None] == y[None, :]) * (wdt(1.0) - th)[:, None]
                if ia.is_constant:
                    work = agree + base_r[:, None]
                else:
                    es = np.ones((Rp, Ep), dtype=wdt)
                    es[grp[g2], epos] = fval[g2]
                    work = agree + base_r[:, None] * norms_w[None, :] * es
            if obs.all():
                wprod *= work
            else:
                wprod[obs] *= work[obs]
        # inverse-CDF categorical per record (one uniform per record);
        # f64 cumulative sum keeps the CDF monotone
        if big:
            np.cumsum(wprod, axis=1, out=cum)
        else:
            cum = np.cumsum(wprod, axis=1, dtype=np.float64)
        target = u_rec[r0:r1] * cum[:, -1]
        sel = (cum < target[:, None]).sum(axis=1)
        new_rec_ent[r0:r1] = e0 + np.minimum(sel, Ep - 1)
        # underflow guard: a record disagreeing with every entity across
        # several rare-valued attributes can drive all f32 products to 0.0,
        # which would silently select entity 0 — recompute those rows in
        # log-space f64 (the true weights are strictly positive: every term
        # has the theta*phi(x) floor)
        bad = np.flatnonzero(~np.isfinite(cum[:, -1]) | (cum[:, -1] <= 0.0))
        for i in bad:
            lw = _dense_row_logweights(state, fm, theta_ra, r0 + i, e0, e1)
            w = np.exp(lw - lw.max())
            c = np.cumsum(w)
            tgt = u_rec[r0 + i] * c[-1]
            new_rec_ent[r0 + i] = e0 + min(int((c < tgt).sum()), Ep - 1)
    return new_rec_ent


def _dense_row_logweights(state, fm, theta_ra, r, e0, e1):
    """f64 log-weights of one record against its partition's entities
    (the PCG-II product of GibbsUpdates.scala:370-393, in log space)."""
    attrs = fm.attrs
    ev = state.ent_values
    rv = state.rec_values
    Ep = e1 - e0
    lw = np.zeros(Ep, dtype=np.float64)
    for a, ia in enumerate(attrs):
        x = int(rv[r, a])
        if x < 0:
            continue
        y = ev[e0:e1, a]
        th = float(theta_ra[r, a])
        base = th * float(ia.index.probs[x])
        if ia.is_constant:
            w = base + (y == x) * (1.0 - th)
        else:
            si = ia.index.sim_index
            es = np.ones(Ep, dtype=np.float64)
            row = slice(int(si.row_ptr[x]), int(si.row_ptr[x + 1]))
            cols, vals = si.col[row], si.expsim[row]
            # scatter this record-value's sparse sim row onto matching entities
            eorder = np.argsort(y, kind="stable")
            ys = y[eorder]
            lo = np.searchsorted(ys, cols)
            cnt = np.searchsorted(ys, cols, side="right") - lo
            grp, goff = _ragged_expand(cnt)
            epos = eorder[lo[grp] + (np.arange(goff[-1]) - goff[grp])]
            es[epos] = vals[grp]
            w = base * ia.index.sim_norms[y] * es + (y == x) * (1.0 - th)
        lw += np.log(w)
    return lw


def sweep_fast(state, cache, partitioner, num_partitions, rank=0, timers=None,
               collapsed=True, collapsed_ids=False):
    """One PCG-I (``collapsed=True``) or plain-Gibbs sweep over the rank's
    (partition-sorted) state, in place. The variants share the indexed link
    and distortion updates; the value update differs — Gibbs copies the
    (unique) value of any linked non-distorted record and samples the
    perturbation WITHOUT the collapsed ``(1/theta - 1)`` self term
    (GibbsUpdates.scala:605-646 vs :576-599).

    Same contract as ``cpu_engine.sweep``: advances current_seed by
    num_partitions and increments the iteration counter. ``timers`` (a dict)
    accumulates per-phase wall-clock ms when given (DBLINK_PHASE_TIMERS=1).
    """
    import time as _time

    def _mark(name, _t=[None]):
        if timers is None:
            return
        now = _time.perf_counter()
        if _t[0] is not None and name is not None:
            timers[name] = timers.get(name, 0.0) + (now - _t[0]) * 1000.0
        _t[0] = now

    _mark(None)
    fm = get_fast_model(cache)
    attrs, A = fm.attrs, fm.A
    seed = int(state.current_seed)
    it = int(state.iteration) + 1
    E = state.num_entities
    R = state.num_records
    ev = state.ent_values
    rv = state.rec_values
    rdist = state.rec_dist.astype(bool)
    obs = rv >= 0
    theta = state.dist_probs.probs  # [A, F]
    theta_ra = theta[np.arange(A)[None, :], state.rec_file[:, None]]  # [R, A]

    # ---- phase 1: link update (indexed, Gumbel-max) -------------------------
    if E and int(state.ent_part.max()) >= num_partitions:
        raise RuntimeError(
            "state has partition ids beyond the partitioner's range "
            "(resuming with an unfitted partitioner?)"
        )
    if collapsed_ids:
        state.rec_ent = _link_dense_collapsed(
            state, fm, theta_ra, num_partitions, seed, it, rank
        )
        _mark("link")
        return _value_and_rest(state, fm, partitioner, num_partitions, rank,
                               seed, it, theta_ra, obs, rdist, collapsed,
                               _mark)
    native = _link_indexed_native(state, fm, num_partitions, seed, it)
    if native is not None:
        state.rec_ent = native
        _mark("link")
        return _value_and_rest(state, fm, partitioner, num_partitions, rank,
                               seed, it, theta_ra, obs, rdist, collapsed,
                               _mark)
    ent_ptr = np.searchsorted(state.ent_part, np.arange(num_partitions + 1))
    keys = (
        (state.ent_part[:, None].astype(np.int64) * A + np.arange(A)[None, :])
        * fm.Vmax
        + ev
    ).reshape(-1)
    order = np.argsort(keys)
    postings = (order // A).astype(np.int64)  # row-major [E, A] flatten
    rec_part = state.ent_part[state.rec_ent]
    qkeys = (
        (rec_part[:, None].astype(np.int64) * A + np.arange(A)[None, :]) * fm.Vmax
        + np.maximum(rv, 0)
    )
    nk = num_partitions * A * fm.Vmax
    if nk <= (1 << 26):
        # dense key prefix: candidate ranges by direct lookup (the GPU
        # engine's counting-sort index, in numpy)
        ptr = np.zeros(nk + 1, dtype=np.int64)
        np.cumsum(np.bincount(keys, minlength=nk), out=ptr[1:])
        qflat = qkeys.reshape(-1)
        lo = ptr[qflat].reshape(R, A)
        hi = ptr[qflat + 1].reshape(R, A)
    else:
        skeys = keys[order]
        lo = np.searchsorted(skeys, qkeys.reshape(-1)).reshape(R, A)
        hi = np.searchsorted(skeys, qkeys.reshape(-1), side="right").reshape(R, A)
    nd = obs & ~rdist
    sizes = np.where(nd, hi - lo, np.int64(1) << 60)
    base_a = np.argmin(sizes, axis=1)
    rows = np.arange(R)
    has_nd = nd.any(axis=1)
    base_n = np.where(has_nd, sizes[rows, base_a], ent_ptr[rec_part + 1] - ent_ptr[rec_part])
    base_lo = np.where(has_nd, lo[rows, base_a], ent_ptr[rec_part])

    if (base_n == 0).any():
        raise RuntimeError("empty candidate set: state invariant violated")
    rec_of, off = _ragged_expand(base_n)
    pos = np.arange(off[-1]) - off[rec_of]
    cand = np.where(
        has_nd[rec_of],
        postings[np.minimum(base_lo[rec_of] + pos, len(postings) - 1)],
        base_lo[rec_of] + pos,
    )
    n_cand = len(cand)
    evc = ev.astype(np.int32)[cand]  # [n, A] candidate entity values
    rvc = rv.astype(np.int32)[rec_of]
    # non-distorted observed attributes (other than the base posting list)
    # must agree exactly; the per-record mask is [R, A], gathered once
    mask_ra = nd & (
        (np.arange(A)[None, :] != base_a[:, None]) | ~has_nd[:, None]
    )
    ok = ~np.any(mask_ra[rec_of] & (evc != rvc), axis=1)
    logw = np.zeros(n_cand)
    odm = obs & rdist
    for a in range(A):
        if attrs[a].is_constant:
            continue
        od_m = odm[rec_of, a]
        if od_m.any():
            y = evc[od_m, a]
            x = rvc[od_m, a]
            ld = fm.sim_logdense[a]
            if ld is not None:
                logw[od_m] += ld[x, y]
            else:
                qk = x.astype(np.int64) * fm.num_values[a] + y
                fk = fm.sim_keys[a]
                pos = np.minimum(np.searchsorted(fk, qk), len(fk) - 1)
                hit = fk[pos] == qk
                logw[od_m] += fm.log_norms[a][y] + np.where(
                    hit, fm.sim_logvals[a][pos], 0.0)
    # Gumbel draws keyed by flat position: the candidate list order is a
    # deterministic function of the state, so position-keyed iid uniforms
    # give the same conditional distribution as entity-keyed ones
    u = _philox_dense(seed, it, _PH_LINK, n_cand, rank)
    score = np.where(ok, logw - np.log(-np.log(u)), -np.inf)
    # segmented argmax via two reduceats (max value, then first position
    # attaining it; Gumbel ties have measure zero)
    segmax = np.maximum.reduceat(score, off[:-1])
    if not np.isfinite(segmax).all():
        raise RuntimeError("empty candidate set: state invariant violated")
    at_max = np.where(score == segmax[rec_of], np.arange(n_cand), n_cand)
    best_flat = np.minimum.reduceat(at_max, off[:-1])
    state.rec_ent = cand[best_flat].astype(np.int64)
    _mark("link")

    _value_and_rest(state, fm, partitioner, num_partitions, rank, seed, it,
                    theta_ra, obs, rdist, collapsed, _mark)


def _value_and_rest(state, fm, partitioner, num_partitions, rank, seed, it,
                    theta_ra, obs, rdist, collapsed, _mark):
    """Phases 2-3 (value + distortion) and bookkeeping, shared by the
    indexed (PCG-I / Gibbs) and dense (PCG-II) link variants."""
    attrs, A = fm.attrs, fm.A
    E = state.num_entities
    R = state.num_records
    ev = state.ent_values
    rv = state.rec_values

    # ---- phase 2: collapsed entity-value update -----------------------------
    # records in entity order, computed once (native + numpy paths and the
    # oracle fallback all address records through it)
    rorder_all = _stable_argsort(state.rec_ent, state.ent_values.shape[0])
    native = _value_native(state, fm, collapsed, seed, it, rank, rorder_all)
    if native is not None:
        new_ev, fallback = native
    else:
        # ---- phase 2: collapsed entity-value update -----------------------------
        kobs = np.zeros((E, A), dtype=np.int64)
        np.add.at(kobs, state.rec_ent, obs.astype(np.int64))
        ea_ids = (np.arange(E)[:, None] * A + np.arange(A)[None, :])
        u_mix, u_a1, u_a2, u_sel = (
            u.reshape(E, A)
            for u in _philox_uniform4(seed, it, _PH_VAL, ea_ids.reshape(-1), 0, rank)
        )

        new_ev = ev.copy()
        fallback = []
        if not collapsed:
            # Gibbs: a non-distorted observed copy pins the value (all such
            # copies agree with the entity by chain invariant)
            ndm = obs & ~rdist
            nd_count = np.zeros((E, A), dtype=np.int64)
            np.add.at(nd_count, state.rec_ent, ndm.astype(np.int64))
            copy_val = np.zeros((E, A), dtype=rv.dtype)
            for a in range(A):
                m = ndm[:, a]
                copy_val[state.rec_ent[m], a] = rv[m, a]
        for a in range(A):
            ia = attrs[a]
            k = kobs[:, a].copy()
            if not collapsed:
                pinned = nd_count[:, a] > 0
                new_ev[pinned, a] = copy_val[pinned, a]
                k[pinned] = -1  # handled; excluded from every draw branch below
                if ia.is_constant:
                    # constant attr, observed but all distorted: plain phi draw
                    k[k >= 1] = 0
            k0 = k == 0
            if k0.any():  # no observed copies: draw from phi
                t = fm.phi_tables[a]
                new_ev[k0, a] = _alias_draw_vec(u_a1[k0, a], u_a2[k0, a], t.prob, t.alias)
            kcap = fm.kmax
            k1 = k == 1
            if k1.any():
                # single observed copy (the common case): one sim row, no merge
                e1 = np.flatnonzero(k1)
                rr_s = rorder_all[obs[rorder_all, a]]
                r1 = rr_s[np.searchsorted(state.rec_ent[rr_s], e1)]
                x1 = rv[r1, a]
                th1 = theta_ra[r1, a]
                if ia.is_constant:
                    # collapsed closed form: total = 1/theta - 1 => P(base) = theta
                    # (the non-collapsed constant case never reaches here)
                    take_base = u_mix[e1, a] < th1
                    t = fm.phi_tables[a]
                    tb = e1[take_base]
                    new_ev[tb, a] = _alias_draw_vec(u_a1[tb, a], u_a2[tb, a],
                                                    t.prob, t.alias)
                    new_ev[e1[~take_base], a] = x1[~take_base]
                else:
                    si = ia.index.sim_index
                    rlen = si.row_ptr[x1 + 1] - si.row_ptr[x1]
                    grp, goff = _ragged_expand(rlen)
                    flat_i = si.row_ptr[x1[grp]] + (np.arange(goff[-1]) - goff[grp])
                    fcol = si.col[flat_i]
                    w = si.expsim[flat_i].copy()
                    if collapsed:
                        pxn = ia.index.probs[x1] * ia.index.sim_norms[x1]
                        selfm = fcol == x1[grp]
                        gsf = grp[selfm]
                        w[selfm] += (1.0 / th1[gsf] - 1.0) / pxn[gsf]
                    wgt = (ia.index.probs[fcol] * ia.index.sim_norms[fcol]
                           / fm.pow_totals[a][1]) * (w - 1.0)
                    c = np.cumsum(wgt)
                    basec = np.where(goff[:-1] > 0, c[goff[:-1] - 1], 0.0)
                    tot = c[goff[1:] - 1] - basec
                    take_base = u_mix[e1, a] < 1.0 / (1.0 + tot)
                    t1 = ia.index.sim_norm_dist(1)
                    tb = e1[take_base]
                    new_ev[tb, a] = _alias_draw_vec(u_a1[tb, a], u_a2[tb, a],
                                                    t1.prob, t1.alias)
                    pert = ~take_base
                    if pert.any():
                        target = basec[pert] + u_sel[e1[pert], a] * tot[pert]
                        j = np.searchsorted(c, target, side="right")
                        j = np.minimum(j, goff[1:][pert] - 1)
                        new_ev[e1[pert], a] = fcol[j]
            km = (k >= 2) & (k <= kcap)
            for e in np.flatnonzero(k > kcap):  # beyond the cached powers: oracle path
                fallback.append((int(e), a))
            if not km.any():
                continue
            e_idx = np.flatnonzero(km)
            kk = k[e_idx]
            # ragged (pair -> its observed records), entity-sorted
            rr_s = rorder_all[obs[rorder_all, a]]
            gstart = np.searchsorted(state.rec_ent[rr_s], e_idx)
            pgrp, poff = _ragged_expand(kk)
            ridx = rr_s[gstart[pgrp] + (np.arange(poff[-1]) - poff[pgrp])]
            x = rv[ridx, a]
            th = theta_ra[ridx, a]
            # each record contributes its sim row (constants: the singleton {x})
            # with the closed-form self-term added at v == x
            if ia.is_constant:
                fcol = x
                px = ia.index.probs[x]
                w = 1.0 + (1.0 / th - 1.0) / px  # collapsed only (see gate above)
                pair_of = pgrp
            else:
                si = ia.index.sim_index
                rlen = si.row_ptr[x + 1] - si.row_ptr[x]
                ggrp, goff2 = _ragged_expand(rlen)
                flat_i = si.row_ptr[x[ggrp]] + (np.arange(goff2[-1]) - goff2[ggrp])
                fcol = si.col[flat_i]
                w = si.expsim[flat_i].copy()
                if collapsed:
                    pxn = ia.index.probs[x] * ia.index.sim_norms[x]
                    selfm = fcol == x[ggrp]
                    gs_ = ggrp[selfm]
                    w[selfm] += (1.0 / th[gs_] - 1.0) / pxn[gs_]
                pair_of = pgrp[ggrp]
            # union-combine: product of the records' factors per (pair, value)
            so = np.lexsort((fcol, pair_of))
            pc, cc, wc = pair_of[so], fcol[so], w[so]
            runs = np.flatnonzero(np.r_[True, (pc[1:] != pc[:-1]) | (cc[1:] != cc[:-1])])
            vw = np.multiply.reduceat(wc, runs)
            ucol = cc[runs]
            upair = pc[runs]
            # base distribution: p_k(v) = phi(v) * norm(v)^k / Z_k (phi for consts)
            if ia.is_constant:
                basep = ia.index.probs[ucol]
            else:
                kw = kk[upair].astype(np.float64)
                basep = (ia.index.probs[ucol] * ia.index.sim_norms[ucol] ** kw
                         / fm.pow_totals[a][kk[upair]])
            weight = basep * (vw - 1.0)
            c = np.cumsum(weight)
            pstart = np.flatnonzero(np.r_[True, upair[1:] != upair[:-1]])
            pend = np.r_[pstart[1:], len(upair)]
            basec = np.where(pstart > 0, c[pstart - 1], 0.0)
            tot = c[pend - 1] - basec
            take_base = u_mix[e_idx, a] < 1.0 / (1.0 + tot)
            if ia.is_constant:
                t = fm.phi_tables[a]
                tb = e_idx[take_base]
                new_ev[tb, a] = _alias_draw_vec(u_a1[tb, a], u_a2[tb, a], t.prob, t.alias)
            else:
                for kv in np.unique(kk[take_base]):
                    sel = take_base & (kk == kv)
                    tb = e_idx[sel]
                    t = ia.index.sim_norm_dist(int(kv))
                    new_ev[tb, a] = _alias_draw_vec(u_a1[tb, a], u_a2[tb, a],
                                                    t.prob, t.alias)
            pert = ~take_base
            if pert.any():
                target = basec[pert] + u_sel[e_idx[pert], a] * tot[pert]
                j = np.searchsorted(c, target, side="right")
                j = np.minimum(j, pend[pert] - 1)
                new_ev[e_idx[pert], a] = ucol[j]


    if fallback:
        from . import cpu_engine as ce

        class _P:  # lightweight partition view over the whole rank
            pass

        part = _P()
        part.rec_values = rv
        part.rec_file = state.rec_file
        part.rec_dist = state.rec_dist
        starts = np.searchsorted(state.rec_ent[rorder_all], np.arange(E + 1))
        # one sweep-keyed stream, consumed in deterministic (e, a) order
        rng = np.random.Generator(
            np.random.Philox(key=((seed & ((1 << 63) - 1)) << 64) | (it << 32) | rank)
        )
        for e, a in sorted(fallback):
            linked = rorder_all[starts[e]:starts[e + 1]]
            if collapsed:
                new_ev[e, a] = ce._update_entity_value_collapsed(
                    rng, a, attrs[a], part, linked, state.dist_probs
                )
            else:
                new_ev[e, a] = ce._update_entity_value(
                    rng, a, attrs[a], part, linked
                )
    state.ent_values = new_ev
    _mark("value")

    # ---- phase 3: distortion update -----------------------------------------
    z8 = _distortion_native(state, fm, seed, it, rank)
    if z8 is not None:
        state.rec_dist = z8
    else:
        y_link = state.ent_values[state.rec_ent]  # [R, A]
        u_d = _philox_dense(seed, it, _PH_DIST, R * A, rank).reshape(R, A)
        self_mass = np.stack(
            [ia.index.self_mass[np.maximum(rv[:, a], 0)] for a, ia in enumerate(attrs)], 1
        )
        pr1 = theta_ra * self_mass
        p_agree = pr1 / (pr1 + (1.0 - theta_ra))
        z = np.where(
            ~obs,
            u_d < theta_ra,
            np.where(rv == y_link, u_d < p_agree, True),
        )
        state.rec_dist = z.astype(np.uint8)
    _mark("distortion")

    # ---- partition reassignment + bookkeeping -------------------------------
    state.ent_part = partitioner.get_partition_ids(state.ent_values).astype(np.int32)
    state.current_seed += num_partitions
    state.iteration += 1
    _mark("kd")
