"""GPU engine: the partitioned Gibbs sweep as CDNA4 HIP kernels.

Design (MI355X-first, not a port of the Spark flow):

- all chain state lives in HBM3E as flat SoA tensors sorted by partition id;
  per-partition segments are addressed by offset tables, so ONE kernel launch
  covers every partition this rank owns (256 CUs want >>256 workgroups)
- the inverted index is rebuilt per sweep as a dense counting sort over
  (partition, slot, value) keys with LDS-aggregated atomics; candidate
  ranges are read straight off the key prefix
- link / value / distortion updates run as the fused kernels of
  ``ops/csrc/kernels.hip`` with Philox counter RNG (order-independent draws)
- summaries reduce on-device to a small packed tensor -> one RCCL all-reduce
- cluster migration is an RCCL all-to-all-v over xGMI, then a device-side
  re-sort by partition id

The host syncs once per iteration (summary + theta update, as in the
reference's driver loop) and pulls full state back only when a sample is
written or the state is saved.
"""

from __future__ import annotations

import logging
import os

import numpy as np
import torch

from ..parallel import comm
from .cpu_engine import CpuEngine, SamplerFlags
from .state import ChainState, SummaryVars

log = logging.getLogger("dblink_amd.gpu")


def _require_native():
    from .. import ops

    return ops.native()


class GpuModel:
    """Device-resident read-only model tensors (the broadcast RecordsCache)."""

    def __init__(self, cache, device, max_cluster_size):
        self.cache = cache
        self.device = device
        attrs = cache.indexed_attributes
        A = len(attrs)
        if A > 16:
            raise ValueError(
                f"{A} matching attributes exceed the GPU engine's limit of 16 "
                "(kernels.hip MAX_ATTRS); use dblink.engine = cpu"
            )
        self.A = A
        self.F = cache.num_files
        self.Kc = int(max_cluster_size)

        voff = np.zeros(A + 1, dtype=np.int64)
        for a, ia in enumerate(attrs):
            voff[a + 1] = voff[a] + ia.index.num_values
        self.Vtot = int(voff[A])
        self.Vmax = int(max(ia.index.num_values for ia in attrs))

        phi = np.concatenate([ia.index.probs for ia in attrs]).astype(np.float64)
        norm = np.concatenate([ia.index.sim_norms for ia in attrs]).astype(np.float64)

        # CSR similarity index with per-attribute row offsets
        row_ptr = np.zeros(self.Vtot + 1, dtype=np.int64)
        cols, sims = [], []
        nnz = 0
        for a, ia in enumerate(attrs):
            V = ia.index.num_values
            if ia.is_constant:
                row_ptr[voff[a] + 1 : voff[a] + V + 1] = nnz
            else:
                si = ia.index.sim_index
                row_ptr[voff[a] + 1 : voff[a] + V + 1] = nnz + si.row_ptr[1:]
                cols.append(si.col)
                sims.append(np.log(si.expsim))  # store sim = log(expsim)
                nnz += int(si.row_ptr[-1])
        csr_col = np.concatenate(cols) if cols else np.empty(0, dtype=np.int32)
        csr_sim = np.concatenate(sims) if sims else np.empty(0, dtype=np.float64)

        # alias tables
        phi_prob = np.concatenate([ia.index.distribution.prob for ia in attrs])
        phi_alias = np.concatenate([ia.index.distribution.alias for ia in attrs])

        pow_prob, pow_alias = [], []
        pow_off = np.full(A, -1, dtype=np.int64)
        log_pow_total = np.zeros((A, self.Kc + 1), dtype=np.float64)
        off = 0
        for a, ia in enumerate(attrs):
            if ia.is_constant:
                continue
            V = ia.index.num_values
            pow_off[a] = off
            for k in range(1, self.Kc + 1):
                t = ia.index.sim_norm_dist(k)
                pow_prob.append(t.prob)
                pow_alias.append(t.alias)
                log_pow_total[a, k] = np.log(ia.index.sim_norm_total(k))
            off += self.Kc * V

        # Static k=1 perturbation tables: for a single linked record with
        # value x, the perturbation weights are base(v)*(f(v)-1) where the
        # self-term correction adds exactly (1/theta - 1)/Z1 of mass. So the
        # whole distribution reduces to the PRECOMPUTED row quantities
        #   raw_w1[j]   = phi(col)*norm(col)*(expsim-1)        (f64)
        #   row prefix  = exclusive cumsum of raw_w1 within the row
        #   rawsum[x]   = row total
        # and a k=1 draw is one binary search (value_update_k1_kernel).
        excl_list, rawsum = [], np.zeros(self.Vtot, dtype=np.float64)
        for a, ia in enumerate(attrs):
            if ia.is_constant:
                continue
            si = ia.index.sim_index
            pr = ia.index.probs
            nm = ia.index.sim_norms
            raw = pr[si.col] * nm[si.col] * (si.expsim - 1.0)
            c = np.cumsum(raw)
            counts = np.diff(si.row_ptr)
            start = si.row_ptr[:-1]
            base_c = np.where(start > 0, c[np.maximum(start - 1, 0)], 0.0)
            excl = c - raw - np.repeat(base_c, counts)
            ends = si.row_ptr[1:] - 1
            sums = np.where(counts > 0, c[np.maximum(ends, 0)], 0.0) - base_c
            excl_list.append(excl)
            rawsum[voff[a] : voff[a + 1]] = sums
        csr_excl = np.concatenate(excl_list) if excl_list else np.empty(0)
        z1 = np.array(
            [1.0 if ia.is_constant else ia.index.sim_norm_total(1) for ia in attrs]
        )

        # High-similarity sub-index for the hierarchical link sampler: only
        # entries with sim >= tau are enumerated exactly; lower-sim entities
        # ride in the A* complement under the proposal shift |od| * tau.
        # Exact for ANY tau (A* needs only an upper bound) — tau trades the
        # exact-scan size against A* iterations.
        self.heavy_tau = float(os.environ.get("DBLINK_HEAVY_TAU", "1.0"))
        # filter on the f32 values the kernel compares, so membership in the
        # big index and the kernel's pool-rejection test agree exactly
        big_mask = csr_sim.astype(np.float32) >= np.float32(self.heavy_tau)
        big_counts = np.zeros(self.Vtot, dtype=np.int64)
        rows_of = np.repeat(np.arange(self.Vtot), np.diff(row_ptr))
        np.add.at(big_counts, rows_of[big_mask], 1)
        row_ptr_big = np.zeros(self.Vtot + 1, dtype=np.int64)
        row_ptr_big[1:] = np.cumsum(big_counts)
        col_big = csr_col[big_mask]
        sim_big = csr_sim[big_mask]

        self_expsim = np.concatenate(
            [
                np.full(
                    ia.index.num_values,
                    1.0 if ia.is_constant else float(np.exp(ia.spec.similarity_fn.max_similarity)),
                )
                for ia in attrs
            ]
        )

        def dev(arr, dtype):
            return torch.from_numpy(np.ascontiguousarray(arr)).to(dtype).to(device)

        self.voff = dev(voff, torch.int64)
        self.phi = dev(phi, torch.float32)
        self.log_phi = dev(np.log(phi), torch.float32)
        self.norm_lin = dev(norm, torch.float32)
        self.log_norm = dev(np.log(norm), torch.float32)
        self.csr_row_ptr = dev(row_ptr, torch.int64)
        self.csr_col = dev(csr_col, torch.int32)
        self.csr_sim = dev(csr_sim, torch.float32)
        self.csr_row_ptr_big = dev(row_ptr_big, torch.int64)
        self.csr_col_big = dev(col_big, torch.int32)
        self.csr_sim_big = dev(sim_big, torch.float32)
        self.phi_prob = dev(phi_prob, torch.float32)
        self.phi_alias = dev(phi_alias, torch.int32)
        self.pow_prob = dev(np.concatenate(pow_prob) if pow_prob else np.empty(0), torch.float32)
        self.pow_alias = dev(
            np.concatenate(pow_alias) if pow_alias else np.empty(0, np.int64), torch.int32
        )
        self.pow_off = dev(pow_off, torch.int64)
        self.log_pow_total = dev(log_pow_total.reshape(-1), torch.float32)
        self.attr_const = dev(
            np.array([1 if ia.is_constant else 0 for ia in attrs], dtype=np.uint8), torch.uint8
        )
        self.self_expsim = dev(self_expsim, torch.float32)
        self.csr_excl = dev(csr_excl, torch.float64)
        self.csr_rawsum = dev(rawsum, torch.float64)
        self.z1 = dev(z1, torch.float64)
        self.theta = torch.zeros((A, self.F), dtype=torch.float32, device=device)
        self._build_value_ktables(device)

    def _build_value_ktables(self, device):
        """Per-(value, k) perturbation tables for k in [2, kmax]: the raw
        weights phi(c) norm(c)^k (e^{k s} - 1) of each sim-row entry, as
        exclusive row prefixes + row totals (f64). Lets the dominant k >= 2
        value-update class (all linked records sharing one value; measured
        d~1.1-1.6 mean distinct values) draw by ONE binary search instead of
        an O(k row) wave merge. Memory-capped by DBLINK_KTAB_MAX_BYTES."""
        import torch

        self.ktab_excl = self.ktab_rawsum = None
        self.ktab_max = 0
        nnz = int(self.csr_col.numel())
        if nnz == 0 or self.Kc < 2:
            return
        cap = int(os.environ.get("DBLINK_KTAB_MAX_BYTES", str(48 << 30)))
        levels = min(self.Kc - 1, max(0, cap // (16 * (nnz + self.Vtot))))
        if levels < 1:
            return
        kmax = 1 + levels
        row_counts = (self.csr_row_ptr[1:] - self.csr_row_ptr[:-1])
        row_of = torch.repeat_interleave(
            torch.arange(self.Vtot, device=device, dtype=torch.int64), row_counts)
        a_of = torch.searchsorted(self.voff, row_of, right=True) - 1
        gcol = self.voff[a_of] + self.csr_col.to(torch.int64)
        lp = self.log_phi.double()
        ln = self.log_norm.double()
        sim = self.csr_sim.double()
        base_log = lp[gcol] + ln[gcol]  # reused; k-scaled below
        ln_g = ln[gcol]
        start = self.csr_row_ptr[:-1]
        ends = self.csr_row_ptr[1:] - 1
        excl = torch.empty((levels, nnz), dtype=torch.float64, device=device)
        rawsum = torch.zeros((levels, self.Vtot), dtype=torch.float64, device=device)
        for k in range(2, kmax + 1):
            raw = torch.exp(base_log + (k - 1) * ln_g) * torch.expm1(k * sim)
            c = torch.cumsum(raw, 0)
            base_c = torch.where(start > 0, c[torch.clamp(start - 1, min=0)],
                                 torch.zeros((), dtype=torch.float64, device=device))
            excl[k - 2] = c - raw - torch.repeat_interleave(base_c, row_counts)
            rs = torch.where(row_counts > 0, c[torch.clamp(ends, min=0)],
                             torch.zeros((), dtype=torch.float64, device=device)) - base_c
            rawsum[k - 2] = rs
        self.ktab_excl = excl
        self.ktab_rawsum = rawsum
        self.ktab_max = kmax

        # 2-distinct-value tables: raw_{k,m}(x-row entry c) =
        # phi(c) norm(c)^k (e^{m s} - 1) for 2 <= k <= k2max, 1 <= m < k
        # (level L = (k-2)(k-1)/2 + m - 1). The d=2 class then draws from
        # two table CDFs plus an exact residual over the row intersection.
        # MEASURED NEGATIVE (DBLINK_KTAB2=1 to enable): the residual walk
        # still pays the per-entry cross-row binary searches that dominate
        # the merge path, so it saves nothing (10M stationary value phase
        # 55 -> 60 ms) — kept as a validated experiment (kd2 distribution
        # test) and for schemas where intersections are tiny.
        self.k2tab_excl = self.k2tab_rawsum = None
        self.k2tab_max = 0
        if os.environ.get("DBLINK_KTAB2", "0") != "1":
            return  # experiment off by default: skip the multi-GB build too
        cap2 = int(os.environ.get("DBLINK_KTAB2_MAX_BYTES", str(64 << 30)))
        k2 = 2
        while k2 < min(kmax, 5) and \
                ((k2 + 1) * k2 // 2) * 16 * (nnz + self.Vtot) <= cap2:
            k2 += 1
        lv2 = k2 * (k2 - 1) // 2
        if lv2 * 16 * (nnz + self.Vtot) > cap2 or lv2 < 1:
            return
        excl2 = torch.empty((lv2, nnz), dtype=torch.float64, device=device)
        rawsum2 = torch.zeros((lv2, self.Vtot), dtype=torch.float64, device=device)
        for k in range(2, k2 + 1):
            for m in range(1, k):
                L = (k - 2) * (k - 1) // 2 + m - 1
                raw = torch.exp(base_log + (k - 1) * ln_g) * torch.expm1(m * sim)
                c = torch.cumsum(raw, 0)
                base_c = torch.where(start > 0, c[torch.clamp(start - 1, min=0)],
                                     torch.zeros((), dtype=torch.float64,
                                                 device=device))
                excl2[L] = c - raw - torch.repeat_interleave(base_c, row_counts)
                rawsum2[L] = torch.where(
                    row_counts > 0, c[torch.clamp(ends, min=0)],
                    torch.zeros((), dtype=torch.float64, device=device)) - base_c
        self.k2tab_excl = excl2
        self.k2tab_rawsum = rawsum2
        self.k2tab_max = k2


class GpuStateTensors:
    """Device mirror of ChainState's mutable arrays."""

    FIELDS = ("ent_values", "ent_part", "rec_values", "rec_file", "rec_ent", "rec_dist", "rec_gid")

    def __init__(self, state: ChainState, device):
        for f in self.FIELDS:
            arr = getattr(state, f)
            setattr(self, f, torch.from_numpy(np.ascontiguousarray(arr)).to(device))
        self.rec_part = self.ent_part[self.rec_ent].contiguous()
        self.device = device

    @classmethod
    def like(cls, other):
        """Second buffer set for the double-buffered single-rank sweep.
        Records never move on a single rank, so the four fields that are
        only written element-wise (or not at all) are SHARED, not cloned —
        the per-sweep re-sort permutes entities and remaps rec_ent only."""
        o = cls.__new__(cls)
        for f in ("ent_values", "ent_part", "rec_ent", "rec_part"):
            setattr(o, f, torch.empty_like(getattr(other, f)))
        for f in ("rec_values", "rec_file", "rec_dist", "rec_gid"):
            setattr(o, f, getattr(other, f))
        o.device = other.device
        return o

    def to_host(self, state: ChainState):
        for f in self.FIELDS:
            setattr(state, f, getattr(self, f).cpu().numpy())
        # single-rank GPU states keep records in identity order; the CPU
        # engine re-sorts lazily if it ever picks this state up
        state.cpu_sorted = False

    @property
    def E(self):
        return int(self.ent_values.shape[0])

    @property
    def R(self):
        return int(self.rec_values.shape[0])


class GpuEngine(CpuEngine):
    """Drop-in engine with the sweep on a ROCm device.

    Inherits host-side pieces (theta update, summary packing/reduction,
    linkage structure) from CpuEngine; the Markov transition itself runs as
    HIP kernels.

    Single-rank sweeps are captured into a hipGraph after a short warm-up:
    the whole iteration (inverted-index sort, link/value/distortion kernels,
    KD descent, partition re-sort, summary reduction) replays as one graph,
    eliminating per-op launch and dispatch overhead. Kernels read
    (seed, iteration) from a device control buffer so the captured graph
    stays valid as the chain advances. Multi-rank runs stay eager (the
    all-to-all exchange has data-dependent message sizes).
    """

    def __init__(self, cache, partitioner, world_size=1, rank=0, device=None):
        super().__init__(cache, partitioner, world_size=world_size, rank=rank)
        if device is None:
            device = torch.device("cuda", 0)
        self.device = device
        self.C = _require_native()
        self.model = GpuModel(cache, device, cache_kc(cache, partitioner))
        zt = torch.empty(0, dtype=torch.float64)
        if self.model.ktab_max >= 2 and os.environ.get("DBLINK_KTAB", "1") != "0":
            self.C.set_value_ktables(
                self.model.ktab_excl, self.model.ktab_rawsum,
                self.model.self_expsim, self.model.ktab_max,
                int(self.model.csr_col.numel()))
        else:
            self.C.set_value_ktables(zt, zt, torch.empty(0, dtype=torch.float32),
                                     0, 0)
        if self.model.k2tab_max >= 2 and os.environ.get("DBLINK_KTAB2", "0") == "1":
            self.C.set_value_k2tables(self.model.k2tab_excl,
                                      self.model.k2tab_rawsum,
                                      self.model.k2tab_max)
        else:
            self.C.set_value_k2tables(zt, zt, 0)
        self.flat_tree = None
        self._gs = None
        self._ent_id_base = rank << 40
        self._err = torch.zeros(1, dtype=torch.int32, device=device)
        # per-phase HIP-event timers (SURVEY.md §5.1 observability): eager mode
        # only; accumulate totals, report via phase_times()
        self.phase_timers = os.environ.get("DBLINK_PHASE_TIMERS", "") == "1"
        self._graphs_enabled = (
            os.environ.get("DBLINK_GRAPHS", "1") != "0" and world_size == 1
        )
        self._loglik_buf = torch.zeros(256, dtype=torch.float64, device=device)
        A, F = self.model.A, self.model.F
        self._packed = torch.zeros(2 + A * F + A + 1, dtype=torch.float64, device=device)
        self._ctrl = torch.zeros(2, dtype=torch.int64, device=device)
        self._ctrl_pin = torch.zeros(2, dtype=torch.int64, pin_memory=True)
        self._theta_pin = torch.zeros((A, F), dtype=torch.float32, pin_memory=True)
        self._empty_i64 = torch.empty(0, dtype=torch.int64, device=device)
        self._counts = torch.zeros(1 + A * F + A + 1, dtype=torch.int64, device=device)
        # counting-sort inverted-index buffers (dense per-key histogram /
        # prefix / cursor over (partition, slot, value)); sized at first sweep
        self._idx_counts = None
        self._idx_ptr = None
        self._idx_cursor = None
        self._idx_nk = 0
        # constant-attribute pairs (composite postings shrink the link
        # iteration base when only constants are non-distorted). Off by
        # default: measured slightly net-negative on the RLdata schema (the
        # larger posting sort outweighs the smaller bases); enable for
        # constant-heavy schemas with DBLINK_CONST_PAIRS=1.
        pairs = []
        consts = (
            [a for a, ia in enumerate(cache.indexed_attributes) if ia.is_constant]
            if os.environ.get("DBLINK_CONST_PAIRS", "") == "1" else []
        )
        for i in range(len(consts)):
            for j in range(i + 1, len(consts)):
                a1, a2 = consts[i], consts[j]
                v1 = cache.indexed_attributes[a1].index.num_values
                v2 = cache.indexed_attributes[a2].index.num_values
                if v1 * v2 <= 65536:
                    pairs.append((a1, a2, v2, v1 * v2))
        self._pair_a1 = torch.tensor([p_[0] for p_ in pairs], dtype=torch.int32, device=device)
        self._pair_a2 = torch.tensor([p_[1] for p_ in pairs], dtype=torch.int32, device=device)
        self._pair_v2 = torch.tensor([p_[2] for p_ in pairs], dtype=torch.int32, device=device)
        self._pair_vmax = max((p_[3] for p_ in pairs), default=0)
        self._num_pairs = len(pairs)
        self._graphs = {}
        self._graph_key = None
        self._graph_warm = 0
        self._flip = 0
        self._gs_alt = None
        self._phase_events = []  # (name, start_event, end_event)
        self._phase_totals = {}
        self._hist_ready = False
        self._summary_host = None
        # hierarchical A* link sampler for huge candidate sets (stationary
        # high-distortion regime); DBLINK_HEAVY=0 reverts to pure scans
        self._heavy_thresh = (
            int(os.environ.get("DBLINK_HEAVY_THRESH", "2048"))
            if os.environ.get("DBLINK_HEAVY", "1") != "0" else 0
        )
        # the hierarchical sampler pays when candidate scans dominate (big
        # partitions); hipGraphs pay when kernel launches dominate (small,
        # sub-ms sweeps). Stream-workspace sorts (rocprim/torch) fault under
        # hipGraph replay at scale, so the two are mutually exclusive: heavy
        # routing engages above this record count and turns graphs off.
        self._heavy_min_records = int(os.environ.get("DBLINK_HEAVY_MIN", "50000"))
        self._heavy_stats = torch.zeros(4, dtype=torch.int64, device=device)
        self._value_stats = None
        if os.environ.get("DBLINK_VALUE_STATS", "") == "1":
            self._value_stats = torch.zeros(8, dtype=torch.int64, device=device)
            self.C.set_value_stats(self._value_stats)
        # overlapped migration (migrants-only async all-to-all with the
        # posting build and summary readback hidden under it) is the default
        # multi-rank path; DBLINK_OVERLAP=0 selects the eager reference path
        self._overlap = os.environ.get("DBLINK_OVERLAP", "1") != "0"

    def _ensure_idx_buffers(self, zero=False):
        """Size (and optionally zero) the dense counting-sort buffers for the
        inverted index; shared by the sweep body and the overlapped-migration
        histogram prebuild."""
        T = self.model.A + self._num_pairs
        vmax = max(self.model.Vmax, self._pair_vmax)
        nk = self.num_partitions * T * vmax
        if self._idx_nk != nk:
            self._idx_nk = nk
            self._idx_counts = torch.zeros(nk, dtype=torch.int32, device=self.device)
            self._idx_ptr = torch.zeros(nk + 1, dtype=torch.int64, device=self.device)
            self._idx_cursor = torch.empty(nk, dtype=torch.int32, device=self.device)
        elif zero:
            self._idx_counts.zero_()
        return nk

    def _heavy_active(self, R):
        return self._heavy_thresh > 0 and R >= self._heavy_min_records

    def _sort_bufs(self, n, E):
        """Persistent radix-sort buffers (keys double-buffer, the static
        entity-id pattern, output postings, rocprim workspace)."""
        if getattr(self, "_sb_key", None) != (n, E):
            tb = int(self.C.radix_sort_pairs_temp_bytes(n))
            dev = self.device
            self._sb = (
                torch.empty(n, dtype=torch.int64, device=dev),
                torch.empty(n, dtype=torch.int64, device=dev),
                (torch.arange(n, device=dev, dtype=torch.int64) % E).to(torch.int32),
                torch.empty(n, dtype=torch.int32, device=dev),
                torch.empty(tb, dtype=torch.uint8, device=dev),
            )
            self._sb_key = (n, E)
        return self._sb

    def _hist_add(self, ent_part, ent_values, reset=False):
        """Accumulate (partition, slot, value) posting counts for a subset of
        entities — the overlapped-migration prebuild of the next sweep's
        inverted index (atomic adds, so stay-home and arrival contributions
        compose in any order)."""
        self._ensure_idx_buffers(zero=reset)
        vmax = max(self.model.Vmax, self._pair_vmax)
        self.C.postings_hist(ent_part, ent_values, self._pair_a1,
                             self._pair_a2, self._pair_v2, vmax,
                             self._idx_counts)
        if not reset:
            self._hist_ready = True

    # ---- state residency -----------------------------------------------------

    def _gpu_state(self, state: ChainState) -> GpuStateTensors:
        if self._gs is None:
            self._gs = GpuStateTensors(state, self.device)
            self._upload_tree()
        return self._gs

    def _upload_tree(self):
        flat = self.partitioner.as_flat()
        self.flat_tree = {
            k: torch.from_numpy(np.ascontiguousarray(v.astype(np.int32))).to(self.device)
            for k, v in flat.items()
        }
        if self.flat_tree["rset"].numel() == 0:
            self.flat_tree["rset"] = torch.zeros(1, dtype=torch.int32, device=self.device)

    def sync_state(self, state: ChainState):
        if self._gs is not None:
            self._gs.to_host(state)

    # ---- the transition ------------------------------------------------------

    def step(self, state: ChainState, flags: SamplerFlags):
        gs = self._gpu_state(state)

        # theta from previous summary (host, Philox stream keyed by iteration)
        self._update_dist_probs(state)
        self._theta_pin.copy_(torch.from_numpy(state.dist_probs.probs).float())
        self._ctrl_pin[0] = state.current_seed
        self._ctrl_pin[1] = state.iteration + 1

        key = (flags.collapsed_entity_ids, flags.collapsed_entity_values, flags.sequential)
        if self.world_size <= 1:
            # double-buffered state: the end-of-sweep partition re-sort
            # gathers directly into the other buffer set; graphs are captured
            # per direction (A->B and B->A) and replayed alternately
            if self._gs_alt is None:
                self._gs_alt = GpuStateTensors.like(gs)
            out = self._gs_alt
            # graph replay saves ~5 us x ~100 launches; at multi-million-record
            # scales a sweep is tens of ms and the heavy-path sort is not
            # graph-safe (rocprim/torch stream-workspace sorts fault under
            # replay — the r01 ">4M records" fault), so capture only where
            # launch overhead actually shows
            use_graphs = (
                self._graphs_enabled
                and gs.R <= int(os.environ.get("DBLINK_GRAPH_MAX_RECORDS", "4000000"))
                and not self._heavy_active(gs.R)
            )
            if use_graphs:
                gkey = (key, self._flip)
                if self._graph_key not in (None, key):
                    self._graphs = {}
                    self._graph_warm = 0
                self._graph_key = key
                g = self._graphs.get(gkey)
                if g is not None:
                    g.replay()
                elif self._graph_warm < 2:
                    self._sweep_body(gs, flags, gs_out=out)
                    self._graph_warm += 1
                else:
                    g = torch.cuda.CUDAGraph()
                    torch.cuda.synchronize()
                    with torch.cuda.graph(g):
                        self._sweep_body(gs, flags, gs_out=out)
                    self._graphs[gkey] = g
                    g.replay()
            else:
                self._sweep_body(gs, flags, graph_safe=False, gs_out=out)
            self._gs, self._gs_alt = out, gs
            self._flip ^= 1
        else:
            self._sweep_body(gs, flags, graph_safe=False)
            if comm.is_distributed():
                from ..parallel.migration import (
                    migrate_and_sort_tensors,
                    migrate_overlapped,
                )

                # summary stats are partition-agnostic: pack them from the
                # pre-migration state and overlap the small all-reduce (on
                # its own communicator) with the migration all-to-all
                self._pack_summary(gs, loglik_done=True)
                work = comm.all_reduce_sum_async(self._packed)
                if self._overlap:
                    hist_add = None
                    if (not flags.sequential and not flags.collapsed_entity_ids
                            and self._ensure_idx_buffers() <= (1 << 28)):
                        hist_add = self._hist_add

                    def during_flight():
                        # payloads in the air: land the summary reduction and
                        # stage the host readback that feeds the next theta
                        # draw (the one required host sync of the loop)
                        if work is not None:
                            work.wait()
                        self._summary_host = self._packed.cpu().numpy().copy()

                    migrate_overlapped(gs, self.world_size, self.rank,
                                       during_flight=during_flight,
                                       hist_add=hist_add)
                else:
                    migrate_and_sort_tensors(gs, self.world_size)
                    if work is not None:
                        work.wait()
                self._summary_reduced = True

        state.current_seed += self.num_partitions
        state.iteration += 1
        state.summary = self._read_summary(state)
        return state

    def _mark(self, name, graph_safe):
        """Record a named HIP event at a sweep-phase boundary (eager mode)."""
        if not self.phase_timers or graph_safe:
            return
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        self._phase_events.append((name, ev))

    def phase_times(self):
        """Drain recorded boundary events -> cumulative {phase: ms}."""
        if self._phase_events:
            torch.cuda.synchronize()
            prev = None
            for name, ev in self._phase_events:
                if name == "start":
                    prev = ev
                    continue
                if prev is not None:
                    self._phase_totals[name] = (
                        self._phase_totals.get(name, 0.0) + prev.elapsed_time(ev)
                    )
                prev = ev
            self._phase_events = []
        return dict(self._phase_totals)

    def _sweep_body(self, gs: GpuStateTensors, flags: SamplerFlags, graph_safe=True,
                    gs_out=None):
        """One full device-side iteration. Capture-safe for world_size == 1:
        control values read from the device ctrl buffer, no host
        synchronisation; with ``gs_out`` the final re-sort gathers into the
        other buffer set (double buffering - see _sort_into)."""
        m = self.model
        A, E, R = m.A, gs.E, gs.R
        dev = self.device

        self._mark("start", graph_safe)
        m.theta.copy_(self._theta_pin, non_blocking=True)
        self._ctrl.copy_(self._ctrl_pin, non_blocking=True)
        ctrl = self._ctrl
        seed, it = 0, 0  # kernels read the ctrl buffer
        # posting histogram possibly pre-built during the previous sweep's
        # migration overlap; consume the flag unconditionally (a sampler
        # switch must not leave stale counts armed)
        hist_pre = self._hist_ready
        self._hist_ready = False

        ent_ptr = torch.searchsorted(
            gs.ent_part.to(torch.int64).contiguous(),
            torch.arange(self.num_partitions + 1, device=dev, dtype=torch.int64),
        )

        # --- inverted index (counting sort over (partition, slot, value)) ----
        # Posting order within a key is arbitrary: the link kernels only
        # enumerate ranges (membership checks are entity-value compares) and
        # Gumbel draws are keyed by entity id, so no stable sort is needed.
        # Ranges come straight off the dense key prefix — no searchsorted.
        dense_idx = False
        if not flags.sequential and not flags.collapsed_entity_ids:
            T = A + self._num_pairs
            vmax = max(m.Vmax, self._pair_vmax)
            nk = self.num_partitions * T * vmax
            if nk <= (1 << 28):  # dense counters (1 GiB cap; always true in practice)
                dense_idx = True
                if not hist_pre:
                    self._ensure_idx_buffers(zero=True)
                    self.C.postings_hist(gs.ent_part, gs.ent_values, self._pair_a1,
                                         self._pair_a2, self._pair_v2, vmax,
                                         self._idx_counts)
                torch.cumsum(self._idx_counts, 0, dtype=torch.int64,
                             out=self._idx_ptr[1:])
                if self._heavy_active(R) or \
                        os.environ.get("DBLINK_FORCE_SORT", "") == "1":
                    # heavy sampler draws postings by segment INDEX, so the
                    # order within each key must be deterministic: sort
                    # (key * E + entity) — same boundaries as the prefix,
                    # entities ascending within each segment. rocprim with a
                    # persistent workspace (torch.sort's allocator-managed
                    # workspace faults under hipGraph replay).
                    ekeys, keys_out, vals_in, postings, temp = self._sort_bufs(
                        T * E, E)
                    self.C.build_ekeys_stable(gs.ent_part, gs.ent_values,
                                              self._pair_a1, self._pair_a2,
                                              self._pair_v2, vmax, ekeys)
                    end_bit = min(64, int(nk * E).bit_length())
                    self.C.radix_sort_pairs_i64_i32(ekeys, keys_out, vals_in,
                                                    postings, end_bit, temp)
                else:
                    self._idx_cursor.copy_(self._idx_ptr[:-1])
                    postings = torch.empty(T * E, dtype=torch.int32, device=dev)
                    self.C.postings_scatter(gs.ent_part, gs.ent_values,
                                            self._pair_a1, self._pair_a2,
                                            self._pair_v2, vmax,
                                            self._idx_cursor, postings)
                cand_lo = torch.empty((R, T), dtype=torch.int64, device=dev)
                cand_hi = torch.empty((R, T), dtype=torch.int64, device=dev)
                self.C.cand_ranges(gs.rec_part, gs.rec_values, self._pair_a1,
                                   self._pair_a2, self._pair_v2, self._idx_ptr,
                                   vmax, cand_lo, cand_hi)
            else:  # degenerate key space: radix sort + batched searchsorted
                keys = torch.empty(T * E, dtype=torch.int64, device=dev)
                qkeys = torch.empty(R * T, dtype=torch.int64, device=dev)
                self.C.build_keys(gs.ent_part, gs.ent_values, gs.rec_part,
                                  gs.rec_values, self._pair_a1, self._pair_a2,
                                  self._pair_v2, vmax, keys, qkeys)
                sorted_keys, perm = torch.sort(keys)
                postings = (perm % E).to(torch.int32)
                cand_lo = torch.searchsorted(sorted_keys, qkeys, right=False).view(R, T).contiguous()
                cand_hi = torch.searchsorted(sorted_keys, qkeys, right=True).view(R, T).contiguous()

        self._mark("index", graph_safe)

        # --- phase 1: link update --------------------------------------------
        rec_ent_new = torch.empty_like(gs.rec_ent)
        if flags.sequential or flags.collapsed_entity_ids:
            self.C.link_update_dense(
                gs.rec_values, gs.rec_dist, gs.rec_gid, gs.rec_part, gs.rec_file,
                gs.ent_values, ent_ptr, m.theta, m.phi, m.norm_lin, m.voff,
                m.csr_row_ptr, m.csr_col, m.csr_sim, m.attr_const,
                1 if flags.collapsed_entity_ids else 0, seed, it, rec_ent_new, ctrl,
            )
        else:
            # three-way routing, all device-side (no host sync, graph-safe):
            # mode 0 wave scan, mode 1 thread scan (short candidate lists),
            # mode 2 hierarchical A* sampler (huge partition-slice candidate
            # sets — the stationary high-distortion regime)
            heavy_th = self._heavy_thresh if (
                dense_idx and self._heavy_active(R)) else 0
            if R >= 50_000 or heavy_th > 0:
                mode_mask = torch.empty(R, dtype=torch.uint8, device=dev)
                self.C.classify_modes(gs.rec_values, gs.rec_dist, gs.rec_part,
                                      ent_ptr, cand_lo, cand_hi, self._num_pairs,
                                      16, heavy_th, mode_mask)
            else:
                # small problems keep every record on the wave path: splitting
                # 10k records across three kernels leaves the chip idle
                # (~150 waves total) and costs more than it saves
                mode_mask = torch.empty(0, dtype=torch.uint8, device=dev)
            self.C.link_update(
                gs.rec_values, gs.rec_dist, gs.rec_gid, gs.rec_part,
                cand_lo, cand_hi, postings, gs.ent_values,
                ent_ptr, m.log_norm, m.voff, m.csr_row_ptr, m.csr_col, m.csr_sim,
                m.attr_const, seed, it, rec_ent_new, gs.rec_ent, self._err,
                mode_mask, ctrl, self._pair_a1, self._pair_a2,
            )
            if heavy_th > 0:
                self.C.link_update_heavy(
                    mode_mask, gs.rec_values, gs.rec_dist, gs.rec_gid,
                    gs.rec_part, gs.ent_values, ent_ptr, m.log_norm, m.voff,
                    m.csr_row_ptr, m.csr_col, m.csr_sim, m.attr_const,
                    m.csr_row_ptr_big, m.csr_col_big, m.csr_sim_big,
                    m.heavy_tau, postings, self._idx_ptr, vmax,
                    self._num_pairs, seed, it, ctrl, rec_ent_new, gs.rec_ent,
                    self._err, self._heavy_stats,
                )
        gs.rec_ent.copy_(rec_ent_new)
        self._mark("link", graph_safe)

        # --- entity -> records CSR -------------------------------------------
        sorted_re, order = torch.sort(gs.rec_ent, stable=True)
        ent_rec_ptr = torch.searchsorted(
            sorted_re, torch.arange(E + 1, device=dev, dtype=torch.int64)
        )
        ent_rec_idx = order

        # --- phase 2: value update (in place) --------------------------------
        obs = gs.rec_values >= 0
        kobs = torch.zeros(E * A, dtype=torch.int32, device=dev)
        pair_idx = (gs.rec_ent.view(R, 1) * A
                    + torch.arange(A, device=dev, dtype=torch.int64).view(1, A))
        kobs.scatter_add_(0, pair_idx.reshape(-1), obs.reshape(-1).to(torch.int32))
        if getattr(self, "value_allwave", False):  # A/B debugging aid
            wave_all = torch.arange(E * A, device=dev, dtype=torch.int64)
            self.C.value_update(
                gs.rec_values, gs.rec_dist, gs.rec_file, ent_rec_ptr, ent_rec_idx,
                gs.ent_values, m.theta, m.phi, m.log_phi, m.norm_lin, m.log_norm,
                m.voff, m.csr_row_ptr, m.csr_col, m.csr_sim, m.phi_prob, m.phi_alias,
                m.pow_prob, m.pow_alias, m.pow_off, m.log_pow_total, m.attr_const,
                m.Kc, 1 if flags.collapsed_entity_values else 0,
                1 if flags.sequential else 0, seed, it, self._ent_id_base, self._err,
                wave_all, self._empty_i64, self._empty_i64,
                m.csr_excl, m.csr_rawsum, m.z1, ctrl,
                torch.empty(0, dtype=torch.int32, device=dev),
            )
            self._mark("value", graph_safe)
            self._loglik_buf.zero_()
            self.C.distortion_update(
                gs.rec_values, gs.rec_dist, gs.rec_file, gs.rec_gid, gs.rec_ent,
                gs.ent_values, m.theta, m.phi, m.norm_lin, m.self_expsim, m.voff,
                m.attr_const, seed, it, ctrl, m.log_phi, m.log_norm, m.csr_row_ptr,
                m.csr_col, m.csr_sim, self._loglik_buf,
            )
            self._mark("distortion", graph_safe)
            ent_part_new = torch.empty_like(gs.ent_part)
            self.C.kd_descent(
                gs.ent_values, self.flat_tree["kind"], self.flat_tree["attr"],
                self.flat_tree["a"], self.flat_tree["b"], self.flat_tree["rset"],
                ent_part_new,
            )
            gs.ent_part.copy_(ent_part_new)
            self._mark("kd", graph_safe)
            self._finish_local(gs, gs_out, ent_rec_ptr, graph_safe)
            return
        # kernels self-select on kobs: no host-side pair lists, no sync
        self.C.value_update(
            gs.rec_values, gs.rec_dist, gs.rec_file, ent_rec_ptr, ent_rec_idx,
            gs.ent_values, m.theta, m.phi, m.log_phi, m.norm_lin, m.log_norm,
            m.voff, m.csr_row_ptr, m.csr_col, m.csr_sim, m.phi_prob, m.phi_alias,
            m.pow_prob, m.pow_alias, m.pow_off, m.log_pow_total, m.attr_const,
            m.Kc, 1 if flags.collapsed_entity_values else 0,
            1 if flags.sequential else 0, seed, it, self._ent_id_base, self._err,
            self._empty_i64, self._empty_i64, self._empty_i64,
            m.csr_excl, m.csr_rawsum, m.z1, ctrl, kobs,
        )

        self._mark("value", graph_safe)
        # --- phase 3: distortion update (log-likelihood fused in) ------------
        self._loglik_buf.zero_()
        self.C.distortion_update(
            gs.rec_values, gs.rec_dist, gs.rec_file, gs.rec_gid, gs.rec_ent,
            gs.ent_values, m.theta, m.phi, m.norm_lin, m.self_expsim, m.voff,
            m.attr_const, seed, it, ctrl, m.log_phi, m.log_norm, m.csr_row_ptr,
            m.csr_col, m.csr_sim, self._loglik_buf,
        )

        self._mark("distortion", graph_safe)
        # --- partition reassignment ------------------------------------------
        ent_part_new = torch.empty_like(gs.ent_part)
        self.C.kd_descent(
            gs.ent_values, self.flat_tree["kind"], self.flat_tree["attr"],
            self.flat_tree["a"], self.flat_tree["b"], self.flat_tree["rset"],
            ent_part_new,
        )
        gs.ent_part.copy_(ent_part_new)
        self._mark("kd", graph_safe)

        self._finish_local(gs, gs_out, ent_rec_ptr, graph_safe)

    def _finish_local(self, gs, gs_out, ent_rec_ptr, graph_safe):
        """Single-rank sweep tail: re-sort by partition id and pack the
        summary. isolate COUNT is permutation-invariant, so the pre-sort CSR
        is still valid for the summary counts."""
        if self.world_size > 1:
            return
        self._sort_into(gs, gs_out)
        self._mark("sort", graph_safe)
        self._pack_summary(gs_out, ent_rec_ptr=ent_rec_ptr, loglik_done=True)
        self._mark("summary", graph_safe)

    def _sort_into(self, gs: GpuStateTensors, out: GpuStateTensors):
        """Re-sort ENTITIES by partition id, gathering straight into the
        other buffer set (two alternating hipGraphs capture A->B and B->A).
        Records keep stable identity order on a single rank: kernels address
        them through rec_ent / rec_part, so only those remap per sweep."""
        E, A = gs.E, self.model.A
        dev = self.device
        order = torch.argsort(gs.ent_part.to(torch.int64), stable=True)
        inv = torch.empty_like(order)
        inv[order] = torch.arange(order.numel(), device=dev)
        torch.gather(gs.ent_part, 0, order, out=out.ent_part)
        torch.gather(gs.ent_values, 0, order.view(E, 1).expand(E, A),
                     out=out.ent_values)
        # records stay put: remap their entity ids through the permutation
        torch.gather(inv, 0, gs.rec_ent, out=out.rec_ent)
        torch.gather(out.ent_part, 0, out.rec_ent, out=out.rec_part)

    # ---- summary -------------------------------------------------------------

    def _pack_summary(self, gs: GpuStateTensors, ent_rec_ptr=None, loglik_done=False):
        m = self.model
        E = gs.E
        self._counts.zero_()
        if not loglik_done:  # standalone summary (no sweep ran this iteration)
            self._loglik_buf.zero_()
            self.C.summary_loglik(
                gs.ent_values, gs.rec_values, gs.rec_dist, gs.rec_ent, m.log_phi,
                m.log_norm, m.voff, m.csr_row_ptr, m.csr_col, m.csr_sim, m.attr_const,
                self._loglik_buf,
            )
        if ent_rec_ptr is None:
            sorted_re, _ = torch.sort(gs.rec_ent)
            ent_rec_ptr = torch.searchsorted(
                sorted_re, torch.arange(E + 1, device=self.device, dtype=torch.int64)
            )
        self.C.summary_counts(
            gs.rec_dist, gs.rec_file, ent_rec_ptr, E, self._counts,
            self._loglik_buf, self._packed,
        )

    def _read_summary(self, state: ChainState) -> SummaryVars:
        from .cpu_engine import add_prior_terms

        m = self.model
        A, F = m.A, m.F
        packed = self._packed
        if self.world_size > 1 and comm.is_distributed() and not getattr(
            self, "_summary_reduced", False
        ):
            packed = packed.clone()
            comm.all_reduce_sum_(packed)
        self._summary_reduced = False
        if self._summary_host is not None:
            # readback already staged during the migration overlap window
            host, self._summary_host = self._summary_host, None
        else:
            host = packed.cpu().numpy()
        err = int(self._err.cpu())
        if err:
            raise RuntimeError(
                f"{err} empty candidate sets in link update (invariant violated)"
            )
        out = SummaryVars(
            num_isolates=int(round(host[1])),
            log_likelihood=float(host[0]),
            agg_distortions=host[2 : 2 + A * F].reshape(A, F).astype(np.int64),
            rec_distortions=host[2 + A * F :].astype(np.int64),
        )
        add_prior_terms(out, self.cache, state.dist_probs)
        return out

    def initial_summary(self, state: ChainState):
        gs = self._gpu_state(state)
        self._pack_summary(gs)
        state.summary = self._read_summary(state)

    def linkage_arrays(self, state: ChainState):
        """Device-side linkage grouping (sort + boundaries on the GPU; only
        offsets and gids come back to the host for the Parquet writer)."""
        gs = self._gs
        if gs is None:
            return super().linkage_arrays(state)
        E = gs.E
        pid = gs.ent_part.to(torch.int64)[gs.rec_ent]
        key = pid * (E + 1) + gs.rec_ent
        order = torch.argsort(key)
        sk = key[order]
        gids = gs.rec_gid[order]
        ne = sk.numel()
        if ne == 0:
            z = np.zeros(1, dtype=np.int64)
            return np.empty(0, np.int32), z, z, np.empty(0, np.int64)
        cb_mask = torch.ones_like(sk, dtype=torch.bool)
        cb_mask[1:] = sk[1:] != sk[:-1]
        cb = torch.nonzero(cb_mask).squeeze(1)
        cluster_pid = (sk[cb] // (E + 1)).to(torch.int64)
        # every owned partition gets a row (empty list when all its entities
        # are isolated), matching the reference's getLinkageStructure
        owned = torch.unique(gs.ent_part.to(torch.int64))
        counts = torch.zeros(owned.numel(), dtype=torch.int64, device=self.device)
        counts.scatter_add_(0, torch.searchsorted(owned, cluster_pid),
                            torch.ones_like(cluster_pid))
        pid_offsets = torch.zeros(owned.numel() + 1, dtype=torch.int64,
                                  device=self.device)
        torch.cumsum(counts, 0, out=pid_offsets[1:])
        cluster_offsets = torch.cat(
            [cb, torch.tensor([ne], device=self.device)]
        ).cpu().numpy()
        return (
            owned.to(torch.int32).cpu().numpy(),
            pid_offsets.cpu().numpy().astype(np.int64),
            cluster_offsets.astype(np.int64),
            gids.cpu().numpy(),
        )

    def linkage_structure(self, state: ChainState, rec_id_of=None):
        self.sync_state(state)
        return super().linkage_structure(state, rec_id_of)


def cache_kc(cache, partitioner):
    """How many power distributions to pre-cache per attribute (the
    reference precaches 1..expectedMaxClusterSize, RecordsCache.scala:112)."""
    for ia in cache.indexed_attributes:
        if not ia.is_constant:
            return max(ia.index._max_cached_power, 1)
    return 1
