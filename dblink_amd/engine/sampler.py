"""MCMC driver loop: burn-in, thinning, buffered sample writing, diagnostics,
checkpointing and final state persistence (parity: ``Sampler.scala:51-124``).

Engine-agnostic: the per-iteration transition is supplied by an engine object
(CpuEngine or GpuEngine) with:

- ``step(state, flags)``    one Markov transition (theta update, sweep,
                            migration, summary) — ``State.nextState``
- ``linkage_structure(state)``  {pid -> [[record-id str, ...], ...]} for this
                            rank's partitions — ``State.getLinkageStructure``
"""

from __future__ import annotations

import logging
import time

from ..parallel import comm
from .cpu_engine import SamplerFlags
from .writers import DiagnosticsWriter, LinkageChainWriter

log = logging.getLogger("dblink_amd.sampler")


def sample(
    engine,
    state,
    sample_size,
    output_path,
    burnin_interval=0,
    thinning_interval=1,
    checkpoint_interval=20,
    write_buffer_size=10,
    flags: SamplerFlags = None,
    rank=0,
    write_output=True,
):
    if sample_size <= 0:
        raise ValueError("`sampleSize` must be positive.")
    if burnin_interval < 0:
        raise ValueError("`burninInterval` must be non-negative.")
    if thinning_interval <= 0:
        raise ValueError("`thinningInterval` must be positive.")
    flags = flags or SamplerFlags()

    sample_ctr = 0
    initial_iteration = state.iteration
    continue_chain = initial_iteration != 0

    linkage_writer = diagnostics_writer = None
    if write_output:
        linkage_writer = LinkageChainWriter(
            output_path, rank=rank, buffer_size=write_buffer_size, append=continue_chain
        )
        if rank == 0:
            diagnostics_writer = DiagnosticsWriter(
                output_path, engine.cache, continue_chain=continue_chain
            )

    def sync():
        if hasattr(engine, "sync_state"):
            engine.sync_state(state)

    def record_sample():
        nonlocal sample_ctr
        if write_output:
            pid_list, pid_offsets, cluster_offsets, gids = engine.linkage_arrays(state)
            dictionary = getattr(engine, "rec_ids_array", None)
            if dictionary is not None:
                # dictionary-encoded: indices are the gids, strings stored once
                linkage_writer.append_arrays(
                    state.iteration, pid_list, pid_offsets, cluster_offsets,
                    gids, id_dictionary=dictionary,
                )
            else:
                rec_id_of = getattr(engine, "rec_id_of", None)
                if rec_id_of is not None:
                    record_ids = [rec_id_of(int(g)) for g in gids]
                else:
                    record_ids = gids.astype(str)
                linkage_writer.append_arrays(
                    state.iteration, pid_list, pid_offsets, cluster_offsets, record_ids
                )
            if diagnostics_writer is not None:
                diagnostics_writer.write_row(state)

    if not continue_chain and burnin_interval == 0:
        record_sample()

    if burnin_interval > 0:
        log.info("Running burn-in for %d iterations.", burnin_interval)
    t0 = time.time()
    while sample_ctr < sample_size:
        engine.step(state, flags)
        completed = state.iteration - initial_iteration
        if completed - 1 == burnin_interval and burnin_interval > 0:
            log.info("Burn-in complete.")
        if completed >= burnin_interval and (completed - burnin_interval) % thinning_interval == 0:
            record_sample()
            sample_ctr += 1
        if checkpoint_interval and completed % checkpoint_interval == 0:
            sync()
            state.save(output_path, rank=rank, world_size=engine.world_size,
                       extra={"partitioner": engine.partitioner})
    dt = time.time() - t0
    iters = state.iteration - initial_iteration
    log.info(
        "Sampling complete: %d iterations in %.2fs (%.3f it/s). Writing final state.",
        iters, dt, iters / dt if dt > 0 else float("nan"),
    )
    if write_output:
        linkage_writer.close()
        if diagnostics_writer is not None:
            diagnostics_writer.close()
        comm.barrier()
        sync()
        state.save(output_path, rank=rank, world_size=engine.world_size,
                   extra={"partitioner": engine.partitioner})
    return state
