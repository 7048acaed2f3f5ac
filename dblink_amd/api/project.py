"""Project lifecycle + step execution (parity: ``Project.scala``,
``ProjectStep.scala``, ``ProjectSteps.scala``, ``Run.scala``).

Accepts the same HOCON surface as the reference (docs/configuration.md):

    dblink.data.{path, recordIdentifier, fileIdentifier, entityIdentifier,
                 nullValue, matchingAttributes[]}
    dblink.{outputPath, checkpointPath, randomSeed, populationSize,
            expectedMaxClusterSize, partitioner, steps[]}

plus one extension: ``dblink.engine`` in {"auto", "cpu", "gpu"} (default
auto: GPU when a ROCm device is visible).
"""

from __future__ import annotations

import logging
import os

import numpy as np

from ..analysis import chain as chain_q
from ..analysis import metrics as metrics_m
from ..engine import sampler as sampler_m
from ..engine.cpu_engine import CpuEngine, SamplerFlags
from ..engine.init import deterministic_init
from ..engine.state import ChainState
from ..models.records import Attribute, BetaShapeParameters, RecordsCache, load_csv
from ..models.similarity import similarity_fn_from_config
from ..parallel import comm
from ..parallel.partitioning import partitioner_from_config
from ..utils import hocon

log = logging.getLogger("dblink_amd.project")

SUPPORTED_SAMPLERS = {"PCG-I", "PCG-II", "Gibbs", "Gibbs-Sequential"}
SUPPORTED_METRICS = {"pairwise", "cluster"}
SUPPORTED_QUANTITIES = {"cluster-size-distribution", "partition-sizes", "shared-most-probable-clusters"}


class Project:
    def __init__(self, config: hocon.Config, rank=0, world_size=1, device=None):
        self.config = config
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.data_path = config.get_string("dblink.data.path")
        self.output_path = config.get_string("dblink.outputPath")
        self.checkpoint_path = config.get_or("dblink.checkpointPath", None)
        self.rec_id_attribute = config.get_string("dblink.data.recordIdentifier")
        self.file_id_attribute = (
            config.get_string("dblink.data.fileIdentifier")
            if config.has_path("dblink.data.fileIdentifier")
            else None
        )
        self.ent_id_attribute = (
            config.get_string("dblink.data.entityIdentifier")
            if config.has_path("dblink.data.entityIdentifier")
            else None
        )
        self.null_value = config.get_or("dblink.data.nullValue", "NA")
        self.random_seed = config.get_long("dblink.randomSeed")
        self.population_size = (
            config.get_int("dblink.populationSize")
            if config.has_path("dblink.populationSize")
            else None
        )
        self.expected_max_cluster_size = config.get_or("dblink.expectedMaxClusterSize", 10)
        self.matching_attributes = self._parse_matching_attributes(
            config.get_config_list("dblink.data.matchingAttributes")
        )
        self.attribute_names = [a.name for a in self.matching_attributes]
        self.partitioner = partitioner_from_config(
            config.get_config("dblink.partitioner"), self.attribute_names
        )
        self.engine_kind = config.get_or("dblink.engine", "auto")

        # lazily populated
        self._table = None
        self._ent_ids = None
        self._cache = None
        self._encoded = None
        self._engine = None

    # ---- parsing -------------------------------------------------------------

    @staticmethod
    def _parse_matching_attributes(cfg_list):
        if not cfg_list:
            raise ValueError(
                "dblink.data.matchingAttributes must contain at least one attribute"
            )
        out = []
        for c in cfg_list:
            sim = similarity_fn_from_config(c.get_config("similarityFunction"))
            prior = BetaShapeParameters(
                c.get_double("distortionPrior.alpha"), c.get_double("distortionPrior.beta")
            )
            out.append(Attribute(c.get_string("name"), sim, prior))
        return out

    # ---- data ----------------------------------------------------------------

    @property
    def table(self):
        if self._table is None:
            self._table, self._ent_ids = load_csv(
                self.data_path,
                self.rec_id_attribute,
                self.file_id_attribute,
                self.attribute_names,
                null_value=self.null_value,
                ent_id_col=self.ent_id_attribute,
            )
        return self._table

    @property
    def cache(self) -> RecordsCache:
        if self._cache is None:
            self._cache = RecordsCache.build(
                self.table, self.matching_attributes, self.expected_max_cluster_size
            )
        return self._cache

    def encoded_records(self):
        if self._encoded is None:
            values, files = self.cache.transform_records(self.table)
            self._encoded = (values, files)
        return self._encoded

    def rec_id_of(self, gid: int) -> str:
        return self.table.rec_ids[gid]

    # ---- engine --------------------------------------------------------------

    def engine(self):
        if self._engine is None:
            kind = self.engine_kind
            if kind == "auto":
                import torch

                kind = "gpu" if torch.cuda.is_available() else "cpu"
            if kind == "gpu":
                from ..engine.gpu_engine import GpuEngine

                self._engine = GpuEngine(
                    self.cache, self.partitioner, world_size=self.world_size,
                    rank=self.rank, device=self.device,
                )
            else:
                self._engine = CpuEngine(
                    self.cache, self.partitioner, world_size=self.world_size, rank=self.rank
                )
            self._engine.rec_id_of = self.rec_id_of
            self._engine.rec_ids_array = self.table.rec_ids
        return self._engine

    # ---- state ---------------------------------------------------------------

    def saved_state(self):
        if ChainState.exists(self.output_path):
            log.info("Resuming from saved state at %s", self.output_path)
            state = ChainState.load(self.output_path, rank=self.rank,
                                    world_size=self.world_size)
            # restore the FITTED partition function saved with the chain —
            # a freshly parsed partitioner is unfit (single leaf) and would
            # silently collapse the partition structure on resume
            saved = getattr(state, "saved_partitioner", None)
            if saved is not None:
                self.partitioner = saved
                self._engine = None
            return state
        return None

    def generate_initial_state(self):
        log.info("Generating new initial state")
        values, files = self.encoded_records()
        R = values.shape[0]
        # shard records contiguously over ranks
        bounds = np.linspace(0, R, self.world_size + 1).astype(np.int64)
        lo, hi = int(bounds[self.rank]), int(bounds[self.rank + 1])
        state = deterministic_init(
            values[lo:hi],
            files[lo:hi],
            np.arange(lo, hi, dtype=np.int64),
            self.cache,
            self.partitioner,
            self.random_seed,
            population_size=self.population_size,
            rank=self.rank,
            world_size=self.world_size,
        )
        engine = self.engine()
        engine.initial_summary(state)
        return state

    def true_clusters(self):
        """Ground-truth clustering from the entity-id column (Project.scala:156-166)."""
        if self.ent_id_attribute is None:
            return None
        _ = self.table
        if self._ent_ids is None:
            return None
        membership = dict(zip(self.table.rec_ids, self._ent_ids))
        return metrics_m.membership_to_clusters(membership)

    def mk_string(self):
        lines = []
        lines.append("Data settings")
        lines.append("-------------")
        lines.append(f"  * Using data files located at '{self.data_path}'")
        lines.append(f"  * The record identifier attribute is '{self.rec_id_attribute}'")
        lines.append(
            f"  * The file identifier attribute is '{self.file_id_attribute}'"
            if self.file_id_attribute
            else "  * There is no file identifier"
        )
        lines.append(
            f"  * The entity identifier attribute is '{self.ent_id_attribute}'"
            if self.ent_id_attribute
            else "  * There is no entity identifier"
        )
        names = ", ".join(f"'{n}'" for n in self.attribute_names)
        lines.append(f"  * The matching attributes are {names}")
        lines.append("")
        lines.append("Hyperparameter settings")
        lines.append("-----------------------")
        for i, a in enumerate(self.matching_attributes):
            lines.append(
                f"  * '{a.name}' (id={i}) with {a.similarity_fn.mk_string()} and {a.distortion_prior.mk_string()}"
            )
        lines.append(f"  * Size of latent population is {self.population_size}")
        lines.append("")
        lines.append("Partition function settings")
        lines.append("---------------------------")
        lines.append("  * " + self.partitioner.mk_string())
        lines.append("")
        lines.append("Project settings")
        lines.append("----------------")
        lines.append(f"  * Using randomSeed={self.random_seed}")
        lines.append(f"  * Using expectedMaxClusterSize={self.expected_max_cluster_size}")
        lines.append(f"  * Saving Markov chain and complete final state to '{self.output_path}'")
        lines.append(f"  * Saving checkpoints to '{self.checkpoint_path}'")
        return "\n".join(lines) + "\n"


# ---- steps -------------------------------------------------------------------


class SampleStep:
    def __init__(self, project, sample_size, burnin_interval=0, thinning_interval=1,
                 resume=True, sampler="PCG-I", checkpoint_interval=20):
        # matching Sampler.scala's require(): a non-positive thinning interval
        # is a config error, not something to silently coerce
        if sample_size <= 0:
            raise ValueError("`sampleSize` must be positive.")
        if burnin_interval < 0:
            raise ValueError("`burninInterval` must be non-negative.")
        if thinning_interval <= 0:
            raise ValueError("`thinningInterval` must be positive.")
        assert sampler in SUPPORTED_SAMPLERS, f"sampler must be one of {SUPPORTED_SAMPLERS}"
        self.p = project
        self.sample_size = sample_size
        self.burnin_interval = burnin_interval
        self.thinning_interval = thinning_interval
        self.resume = resume
        self.sampler = sampler
        self.checkpoint_interval = checkpoint_interval

    def scale_warning(self, num_records=None):
        """PCG-II / Gibbs-Sequential score every record against every entity
        of its partition (GibbsUpdates.scala:363-395, 434-466) — quadratic
        by design. Returns a warning string above a size threshold."""
        if self.sampler not in ("PCG-II", "Gibbs-Sequential"):
            return None
        try:
            n = num_records if num_records is not None else self.p.table.num_records
        except Exception:
            return None
        parts = max(1, getattr(self.p.partitioner, "num_partitions", 1))
        per = n // parts
        if per > 20000:
            return (
                f"sampler {self.sampler} is quadratic in partition size "
                f"(~{per} records/entities per partition here): sweeps will "
                "be orders of magnitude slower than PCG-I; increase the "
                "partitioner's numLevels or use PCG-I"
            )
        return None

    def execute(self):
        log.info(self.mk_string())
        warn = self.scale_warning()
        if warn:
            log.warning(warn)
        state = (self.p.saved_state() if self.resume else None) or self.p.generate_initial_state()
        flags = SamplerFlags.for_sampler(self.sampler)
        sampler_m.sample(
            self.p.engine(),
            state,
            self.sample_size,
            self.p.output_path,
            burnin_interval=self.burnin_interval,
            thinning_interval=self.thinning_interval,
            checkpoint_interval=self.checkpoint_interval,
            flags=flags,
            rank=self.p.rank,
        )

    def mk_string(self):
        mode = "saved state" if self.resume else "new initial state"
        return (
            f"SampleStep: Evolving the chain from {mode} with sampleSize={self.sample_size}, "
            f"burninInterval={self.burnin_interval}, thinningInterval={self.thinning_interval} "
            f"and sampler={self.sampler}"
        )


class EvaluateStep:
    def __init__(self, project, lower_iteration_cutoff=0, metrics=("pairwise",), use_existing_smpc=False):
        assert project.ent_id_attribute, "Ground truth entity ids are required for evaluation"
        assert all(m in SUPPORTED_METRICS for m in metrics)
        self.p = project
        self.cutoff = lower_iteration_cutoff
        self.metrics = list(metrics)
        self.use_existing_smpc = use_existing_smpc

    def execute(self):
        log.info(self.mk_string())
        if self.p.rank != 0:
            comm.barrier()
            return
        true_clusters = self.p.true_clusters()
        smpc_path = os.path.join(self.p.output_path, "shared-most-probable-clusters.csv")
        smpc = None
        if self.use_existing_smpc and os.path.exists(smpc_path):
            smpc = chain_q.read_clusters_csv(smpc_path)
        else:
            table = chain_q.load_chain(self.p.output_path, self.cutoff)
            if table is not None:
                smpc = (
                    chain_q.shared_most_probable_clusters_fast(table)
                    if table.num_rows > 1000
                    else chain_q.shared_most_probable_clusters(table)
                )
                chain_q.save_clusters_csv(smpc, smpc_path)
            else:
                log.error("No linkage chain")
        if smpc is not None:
            results = []
            for m in self.metrics:
                if m == "pairwise":
                    results.append(metrics_m.PairwiseMetrics.compute(smpc, true_clusters).mk_string())
                elif m == "cluster":
                    results.append(metrics_m.ClusteringMetrics.compute(smpc, true_clusters).mk_string())
            with open(os.path.join(self.p.output_path, "evaluation-results.txt"), "w") as f:
                f.write("\n".join(results) + "\n")
        comm.barrier()

    def mk_string(self):
        ms = ", ".join(f"'{m}'" for m in self.metrics)
        return (
            f"EvaluateStep: Evaluating sMPC clusters (computed from the chain for iterations >= "
            f"{self.cutoff}) using {{{ms}}} metrics"
        )


class SummarizeStep:
    def __init__(self, project, lower_iteration_cutoff=0, quantities=()):
        assert quantities and all(q in SUPPORTED_QUANTITIES for q in quantities)
        self.p = project
        self.cutoff = lower_iteration_cutoff
        self.quantities = list(quantities)

    def execute(self):
        log.info(self.mk_string())
        if self.p.rank != 0:
            comm.barrier()
            return
        table = chain_q.load_chain(self.p.output_path, self.cutoff)
        if table is None:
            log.error("No linkage chain")
            comm.barrier()
            return
        for q in self.quantities:
            if q == "cluster-size-distribution":
                chain_q.save_cluster_size_distribution(
                    chain_q.cluster_size_distribution(table), self.p.output_path
                )
            elif q == "partition-sizes":
                chain_q.save_partition_sizes(chain_q.partition_sizes(table), self.p.output_path)
            elif q == "shared-most-probable-clusters":
                smpc = (
                    chain_q.shared_most_probable_clusters_fast(table)
                    if table.num_rows > 1000
                    else chain_q.shared_most_probable_clusters(table)
                )
                chain_q.save_clusters_csv(
                    smpc, os.path.join(self.p.output_path, "shared-most-probable-clusters.csv")
                )
        comm.barrier()

    def mk_string(self):
        qs = ", ".join(f"'{q}'" for q in self.quantities)
        return (
            f"SummarizeStep: Calculating summary quantities {{{qs}}} along the chain for "
            f"iterations >= {self.cutoff}"
        )


class CopyFilesStep:
    def __init__(self, project, file_names, destination_path, overwrite=False, delete_source=False):
        self.p = project
        self.file_names = list(file_names)
        self.destination_path = destination_path
        self.overwrite = overwrite
        self.delete_source = delete_source

    def execute(self):
        import shutil

        log.info(self.mk_string())
        if self.p.rank != 0:
            comm.barrier()
            return
        os.makedirs(self.destination_path, exist_ok=True)
        for name in self.file_names:
            src = os.path.join(self.p.output_path, name)
            if not os.path.exists(src):
                continue
            dst = os.path.join(self.destination_path, os.path.basename(name))
            if os.path.exists(dst) and not self.overwrite:
                continue
            if os.path.isdir(src):
                if os.path.exists(dst):
                    shutil.rmtree(dst)
                shutil.copytree(src, dst)
            else:
                shutil.copy2(src, dst)
            if self.delete_source:
                if os.path.isdir(src):
                    shutil.rmtree(src)
                else:
                    os.remove(src)
        comm.barrier()

    def mk_string(self):
        fs = ", ".join(self.file_names)
        return f"CopyFilesStep: Copying {{{fs}}} to destination {self.destination_path}"


def parse_steps(config: hocon.Config, project: Project):
    """``ProjectSteps.scala:53-83``."""
    steps = []
    for step in config.get_config_list("dblink.steps"):
        name = step.get_string("name")
        if name == "sample":
            steps.append(
                SampleStep(
                    project,
                    sample_size=step.get_int("parameters.sampleSize"),
                    burnin_interval=step.get_or("parameters.burninInterval", 0),
                    thinning_interval=step.get_or("parameters.thinningInterval", 1),
                    resume=step.get_or("parameters.resume", True),
                    sampler=step.get_or("parameters.sampler", "PCG-I"),
                    checkpoint_interval=step.get_or("parameters.checkpointInterval", 20),
                )
            )
        elif name == "evaluate":
            steps.append(
                EvaluateStep(
                    project,
                    lower_iteration_cutoff=step.get_or("parameters.lowerIterationCutoff", 0),
                    metrics=step.get_string_list("parameters.metrics"),
                    use_existing_smpc=step.get_or("parameters.useExistingSMPC", False),
                )
            )
        elif name == "summarize":
            steps.append(
                SummarizeStep(
                    project,
                    lower_iteration_cutoff=step.get_or("parameters.lowerIterationCutoff", 0),
                    quantities=step.get_string_list("parameters.quantities"),
                )
            )
        elif name == "copy-files":
            steps.append(
                CopyFilesStep(
                    project,
                    file_names=step.get_string_list("parameters.fileNames"),
                    destination_path=step.get_string("parameters.destinationPath"),
                    overwrite=step.get_or("parameters.overwrite", False),
                    delete_source=step.get_or("parameters.deleteSource", False),
                )
            )
        else:
            raise ValueError(f"unsupported step: {name}")
    return steps


def native_limit_problems(project, table=None):
    """Native-path (HIP kernel) limits, checked against the actual data.

    Returns (errors, warnings). Errors are limits the GPU engine cannot run
    past (kernels.hip MAX_ATTRS; migration distortion bitmask); warnings are
    documented degradations (CPU fallback of the V x V similarity sweep above
    64-byte values; byte-level edit distance above 255 distinct characters).
    """
    errors, warnings = [], []
    A = len(project.matching_attributes)
    if A > 16:
        errors.append(
            f"{A} matching attributes exceed the native-path limit of 16 "
            "(kernels.hip MAX_ATTRS / migration distortion bitmask); "
            "reduce the attribute count or run with dblink.engine = cpu"
        )
    if table is not None:
        for ai, attr in enumerate(project.matching_attributes):
            if attr.is_constant:
                continue
            vals = [v for v in set(table.columns[ai].tolist()) if v is not None]
            charset = {ch for v in vals for ch in v}
            maxlen = max((len(v) for v in vals), default=0)
            if len(charset) > 255:
                warnings.append(
                    f"attribute {attr.name!r} has {len(charset)} distinct "
                    "characters (> 255): the native similarity pass computes "
                    "BYTE-level (UTF-8) edit distance there, which can differ "
                    "from the reference's character-level distance "
                    "(SimilarityFn.scala:92-98) for multi-byte characters"
                )
            elif maxlen > 64:
                warnings.append(
                    f"attribute {attr.name!r} has values up to {maxlen} "
                    "characters (> 64): the V x V similarity sweep runs on "
                    "the CPU (OpenMP) pass instead of the GPU kernel — "
                    "identical results, slower index build"
                )
    return errors, warnings


def check_config(path):
    """Validate a project config without running: parse the HOCON, construct
    the Project (attribute/partitioner/step validation) and load the data,
    reporting problems instead of raising. Returns a process exit code."""
    problems = []
    warnings = []
    try:
        cfg = hocon.parse_file(path)
        project = Project(cfg, rank=0, world_size=1)
        steps = parse_steps(cfg, project)
    except Exception as exc:  # configuration-level failure
        print(f"INVALID: {exc}")
        return 1
    n = 0
    table = None
    try:
        table = project.table
        n = table.num_records
        if n == 0:
            problems.append("data loaded but contains zero records")
    except Exception as exc:
        problems.append(f"data loading failed: {exc}")
    errs, warns = native_limit_problems(project, table)
    problems.extend(errs)
    warnings.extend(warns)
    for s in steps:
        w = s.scale_warning(n) if isinstance(s, SampleStep) else None
        if w:
            warnings.append(w)
    for msg in warnings:
        print(f"WARNING: {msg}")
    if problems:
        for msg in problems:
            print(f"INVALID: {msg}")
        return 1
    print(f"OK: {len(project.matching_attributes)} matching attributes, "
          f"{n} records, {len(steps)} steps "
          f"({', '.join(type(s).__name__ for s in steps)})")
    return 0


def run_config(config_path, rank=None, world_size=None, device=None):
    """CLI entry body (``Run.scala:27-50``)."""
    cfg = hocon.parse_file(config_path)
    if rank is None:
        rank, world_size, device = comm.init_from_env()
    project = Project(cfg, rank=rank, world_size=world_size, device=device)
    os.makedirs(project.output_path, exist_ok=True)
    if rank == 0:
        # rolling run log next to the outputs (parity: the reference's
        # log4j RollingFileAppender writing ./dblink.log)
        import logging.handlers

        root = logging.getLogger()
        if not any(isinstance(h, logging.handlers.RotatingFileHandler)
                   for h in root.handlers):
            fh = logging.handlers.RotatingFileHandler(
                os.path.join(project.output_path, "dblink.log"),
                maxBytes=10_000_000, backupCount=3)
            fh.setFormatter(logging.Formatter(
                "%(asctime)s %(levelname)s %(name)s: %(message)s"))
            root.addHandler(fh)
    if rank == 0:
        with open(os.path.join(project.output_path, "run.txt"), "w") as f:
            f.write(project.mk_string())
    steps = parse_steps(cfg, project)
    if rank == 0:
        lines = ["Scheduled steps", "---------------"] + ["  * " + s.mk_string() for s in steps]
        log.info("\n".join(lines))
        for msg in native_limit_problems(project, project.table)[1]:
            log.warning(msg)
    for step in steps:
        step.execute()
    return project
