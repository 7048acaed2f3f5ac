"""End-to-end CPU pipeline test on a small synthetic RLdata-shaped dataset.

Mirrors the reference's RLdata500 example run as the correctness oracle
(README.md:37-50): sample with PCG-I, summarize, evaluate against ground
truth; pairwise F1 and ARI must land in a high band (the sampler must
actually find the duplicate pairs)."""

import os

import numpy as np
import pytest

from dblink_amd.api.project import Project, parse_steps
from dblink_amd.utils import hocon
from dblink_amd.utils.synthdata import write_csv

CONF_TEMPLATE = """
dblink : {{
    lowDistortion : {{alpha : 0.5, beta : 50.0}}
    constSimFn : {{ name : "ConstantSimilarityFn" }}
    levSimFn : {{
        name : "LevenshteinSimilarityFn",
        parameters : {{ threshold : 7.0, maxSimilarity : 10.0 }}
    }}
    data : {{
        path : "{data}"
        recordIdentifier : "rec_id",
        entityIdentifier : "ent_id"
        nullValue : "NA"
        matchingAttributes : [
            {{name : "by", similarityFunction : ${{dblink.constSimFn}}, distortionPrior : ${{dblink.lowDistortion}}}},
            {{name : "bm", similarityFunction : ${{dblink.constSimFn}}, distortionPrior : ${{dblink.lowDistortion}}}},
            {{name : "bd", similarityFunction : ${{dblink.constSimFn}}, distortionPrior : ${{dblink.lowDistortion}}}},
            {{name : "fname_c1", similarityFunction : ${{dblink.levSimFn}}, distortionPrior : ${{dblink.lowDistortion}}}},
            {{name : "lname_c1", similarityFunction : ${{dblink.levSimFn}}, distortionPrior : ${{dblink.lowDistortion}}}}
        ]
    }}
    randomSeed : 319158
    expectedMaxClusterSize : 10
    engine : "cpu"
    partitioner : {{
        name : "KDTreePartitioner",
        parameters : {{ numLevels : {levels}, matchingAttributes : [{part_attrs}] }}
    }}
    outputPath : "{out}/"
    checkpointPath : "{out}/ckpt/"
    steps : [
        {{name : "sample", parameters : {{
            sampleSize : {samples}, burninInterval : {burnin}, thinningInterval : {thin},
            resume : false, sampler : "{sampler}", checkpointInterval : 0
        }}}},
        {{name : "summarize", parameters : {{
            lowerIterationCutoff : 0,
            quantities : ["cluster-size-distribution", "partition-sizes"]
        }}}},
        {{name : "evaluate", parameters : {{
            lowerIterationCutoff : {cutoff},
            metrics : ["pairwise", "cluster"],
            useExistingSMPC : false
        }}}}
    ]
}}
"""


def run_project(tmp_path, n_records=300, sampler="PCG-I", levels=0, samples=40,
                burnin=20, thin=2, cutoff=30, seed=42):
    data = str(tmp_path / "data.csv")
    write_csv(data, n_records, dup_fraction=0.1, seed=seed)
    out = str(tmp_path / "results")
    part_attrs = '"fname_c1"' if levels > 0 else ""
    conf = CONF_TEMPLATE.format(
        data=data, out=out, samples=samples, burnin=burnin, thin=thin,
        cutoff=cutoff, sampler=sampler, levels=levels, part_attrs=part_attrs,
    )
    conf_path = tmp_path / "test.conf"
    conf_path.write_text(conf)
    cfg = hocon.parse_file(str(conf_path))
    project = Project(cfg, rank=0, world_size=1)
    os.makedirs(project.output_path, exist_ok=True)
    with open(os.path.join(project.output_path, "run.txt"), "w") as f:
        f.write(project.mk_string())
    for step in parse_steps(cfg, project):
        step.execute()
    return project, out


@pytest.mark.slow
def test_pcg1_end_to_end(tmp_path):
    project, out = run_project(
        tmp_path, n_records=300, samples=100, burnin=100, thin=4, cutoff=150
    )

    # Output contract files exist
    assert os.path.exists(os.path.join(out, "run.txt"))
    assert os.path.exists(os.path.join(out, "diagnostics.csv"))
    assert os.path.isdir(os.path.join(out, "linkage-chain.parquet"))
    assert os.path.exists(os.path.join(out, "cluster-size-distribution.csv"))
    assert os.path.exists(os.path.join(out, "partition-sizes.csv"))
    assert os.path.exists(os.path.join(out, "shared-most-probable-clusters.csv"))
    assert os.path.exists(os.path.join(out, "evaluation-results.txt"))
    assert os.path.exists(os.path.join(out, "driver-state"))

    # Diagnostics schema
    with open(os.path.join(out, "diagnostics.csv")) as f:
        header = f.readline().strip().split(",")
        assert header[:5] == ["iteration", "systemTime-ms", "numObservedEntities",
                              "logLikelihood", "popSize"]
        assert "aggDist-fname_c1" in header
        assert "recDistortion-0" in header and "recDistortion-5" in header
        rows = [line.strip().split(",") for line in f if line.strip()]
    assert len(rows) >= 100
    loglik = [float(r[3]) for r in rows]
    assert all(np.isfinite(loglik))
    # chain mixes: the last quarter should be stationary-ish (bounded spread),
    # not diverging (the deterministic init sits near a likelihood peak, so we
    # do NOT require an increase — same as the reference's behavior)
    tail = np.array(loglik[-25:])
    assert np.std(tail) < 0.05 * abs(np.mean(tail))

    # Accuracy oracle
    with open(os.path.join(out, "evaluation-results.txt")) as f:
        txt = f.read()
    metrics = {}
    for line in txt.splitlines():
        for key in ("Precision", "Recall", "F1-score", "Adj. Rand index"):
            if key in line:
                metrics[key] = float(line.split(":")[1])
    # Threshold calibrated against chain-seed variance: over 16 chains
    # (8 seeds x {vectorized, per-record} sweeps) F1 on this config is
    # 0.54-0.67 (mean 0.58) - the sMPC point estimate from 100 samples of a
    # 300-record chain is noisy. 0.5 rejects broken samplers (a random or
    # exact-match-only clustering scores < 0.35 here) without flaking.
    assert metrics["F1-score"] > 0.5, txt
    assert metrics["Adj. Rand index"] > 0.5, txt


@pytest.mark.slow
def test_partitioned_run_and_resume(tmp_path):
    """numLevels=1 (2 partitions) and chain resume (append) behavior."""
    project, out = run_project(tmp_path, n_records=200, levels=1, samples=20,
                               burnin=0, thin=1, cutoff=5)
    import pyarrow.parquet  # noqa: F401

    from dblink_amd.analysis.chain import load_chain

    table = load_chain(out)
    iters1 = max(table["iteration"].to_pylist())
    pids = set(table["partitionId"].to_pylist())
    assert pids == {0, 1}

    # resume: run sample step again with resume=true
    cfg = hocon.parse_file(str(tmp_path / "test.conf"))
    project2 = Project(cfg, rank=0, world_size=1)
    from dblink_amd.api.project import SampleStep

    SampleStep(project2, sample_size=5, resume=True, sampler="PCG-I",
               checkpoint_interval=0).execute()
    table2 = load_chain(out)
    assert max(table2["iteration"].to_pylist()) > iters1


@pytest.mark.parametrize("sampler", ["PCG-II", "Gibbs", "Gibbs-Sequential"])
def test_sampler_variants_run(tmp_path, sampler):
    """All four sampler variants run and produce finite log-likelihoods."""
    project, out = run_project(
        tmp_path, n_records=80, sampler=sampler, samples=5, burnin=0, thin=1, cutoff=0
    )
    with open(os.path.join(out, "diagnostics.csv")) as f:
        f.readline()
        loglik = [float(line.split(",")[3]) for line in f if line.strip()]
    assert len(loglik) >= 5 and all(np.isfinite(loglik))


def test_two_file_project(tmp_path):
    """Two source files via the fileIdentifier column: per-(attribute, file)
    distortion probabilities (DistortionProbs.scala:27-44 semantics)."""
    from dblink_amd.api.project import Project, SampleStep
    from dblink_amd.utils import hocon
    from dblink_amd.utils.synthdata import generate

    n = 120
    cols, header = generate(n, dup_fraction=0.1, seed=6, num_files=2)
    data = tmp_path / "two.csv"
    with open(data, "w") as f:
        f.write(",".join(header) + "\n")
        for i in range(n):
            f.write(",".join(str(cols[h][i]) for h in header) + "\n")
    conf = CONF_TEMPLATE.format(
        data=str(data), out=str(tmp_path / "res"), samples=5, burnin=0, thin=1,
        cutoff=0, sampler="PCG-I", levels=0, part_attrs="",
    ).replace('recordIdentifier : "rec_id",',
              'recordIdentifier : "rec_id",\n        fileIdentifier : "file_id",')
    cfg = hocon.parse_string(conf)
    project = Project(cfg, rank=0, world_size=1)
    assert project.cache.num_files == 2
    import os
    os.makedirs(project.output_path, exist_ok=True)
    SampleStep(project, sample_size=5, resume=False, checkpoint_interval=0).execute()
    # theta matrix has a column per file and both were updated
    state = project.saved_state()
    assert state.dist_probs.probs.shape == (5, 2)
    assert np.all(state.dist_probs.probs > 0) and np.all(state.dist_probs.probs < 1)
    assert state.summary.agg_distortions.shape == (5, 2)


def test_missing_values_and_population_size(tmp_path):
    """Missing values (NA) flow through init, link, value and distortion
    updates; an explicit populationSize > numRecords creates isolates."""
    from dblink_amd.api.project import Project, SampleStep
    from dblink_amd.utils import hocon
    from dblink_amd.utils.synthdata import write_csv

    data = str(tmp_path / "data.csv")
    write_csv(data, 150, dup_fraction=0.1, seed=3, missing_fraction=0.15)
    out = str(tmp_path / "res")
    conf = CONF_TEMPLATE.format(
        data=data, out=out, samples=6, burnin=0, thin=1, cutoff=0,
        sampler="PCG-I", levels=0, part_attrs="",
    ).replace("randomSeed : 319158", "randomSeed : 319158\n    populationSize : 200")
    cfg = hocon.parse_string(conf)
    project = Project(cfg, rank=0, world_size=1)
    # missing values encoded as -1
    values, _ = project.encoded_records()
    assert (values < 0).any()
    assert project.cache.missing_counts  # per (file, attr) counts recorded
    import os
    os.makedirs(project.output_path, exist_ok=True)
    SampleStep(project, sample_size=6, resume=False, checkpoint_interval=0).execute()
    state = project.saved_state()
    assert state.population_size == 200
    assert state.num_entities == 200  # 150 records + 50 forced isolates
    assert state.summary.num_isolates >= 50
    assert np.isfinite(state.summary.log_likelihood)


def test_copy_files_step(tmp_path):
    """copy-files step parity (ProjectStep.scala:157-179): copies named
    outputs (files and directories) to a destination, honoring overwrite."""
    from dblink_amd.api.project import CopyFilesStep, Project
    from dblink_amd.utils import hocon
    from dblink_amd.utils.synthdata import write_csv

    data = str(tmp_path / "d.csv")
    write_csv(data, 50, seed=1)
    out = str(tmp_path / "res")
    conf = CONF_TEMPLATE.format(data=data, out=out, samples=2, burnin=0, thin=1,
                                cutoff=0, sampler="PCG-I", levels=0, part_attrs="")
    cfg = hocon.parse_string(conf)
    project = Project(cfg)
    os.makedirs(out, exist_ok=True)
    (tmp_path / "res" / "diagnostics.csv").write_text("iteration\n0\n")
    os.makedirs(tmp_path / "res" / "linkage-chain.parquet", exist_ok=True)
    (tmp_path / "res" / "linkage-chain.parquet" / "p.parquet").write_text("x")
    dest = str(tmp_path / "backup")
    CopyFilesStep(project, ["diagnostics.csv", "linkage-chain.parquet",
                            "missing.txt"], dest).execute()
    assert os.path.exists(os.path.join(dest, "diagnostics.csv"))
    assert os.path.exists(os.path.join(dest, "linkage-chain.parquet", "p.parquet"))
    assert not os.path.exists(os.path.join(dest, "missing.txt"))
    # no-overwrite: modify source, copy again without overwrite -> unchanged
    (tmp_path / "res" / "diagnostics.csv").write_text("changed")
    CopyFilesStep(project, ["diagnostics.csv"], dest, overwrite=False).execute()
    assert open(os.path.join(dest, "diagnostics.csv")).read() == "iteration\n0\n"
    CopyFilesStep(project, ["diagnostics.csv"], dest, overwrite=True).execute()
    assert open(os.path.join(dest, "diagnostics.csv")).read() == "changed"


def test_run_manifest_contents(tmp_path):
    """run.txt carries the project manifest (Run.scala:38-42 /
    Project.scala:58-96): data path, attributes with similarity functions
    and priors, partitioner, output paths."""
    project, out = run_project(tmp_path, n_records=60, samples=3, burnin=0,
                               thin=1, cutoff=0)
    txt = open(os.path.join(out, "run.txt")).read()
    for needle in ("fname_c1", "LevenshteinSimilarityFn", "ConstantSimilarityFn",
                   "KDTreePartitioner", "beta", "results"):
        assert needle.lower() in txt.lower(), (needle, txt)


def test_summarize_smpc_quantity(tmp_path):
    """The summarize step's shared-most-probable-clusters quantity writes the
    sMPC CSV without an evaluate step (ProjectSteps supportedSummaryQuantities)."""
    data = str(tmp_path / "d.csv")
    write_csv(data, 120, dup_fraction=0.1, seed=3)
    out = str(tmp_path / "res")
    conf = CONF_TEMPLATE.format(
        data=data, out=out, samples=10, burnin=0, thin=1, cutoff=0,
        sampler="PCG-I", levels=0, part_attrs="",
    ).replace(
        'quantities : ["cluster-size-distribution", "partition-sizes"]',
        'quantities : ["shared-most-probable-clusters"]',
    )
    conf_path = tmp_path / "p.conf"
    conf_path.write_text(conf)
    cfg = hocon.parse_file(str(conf_path))
    project = Project(cfg, rank=0, world_size=1)
    os.makedirs(project.output_path, exist_ok=True)
    steps = parse_steps(cfg, project)
    for step in steps[:2]:  # sample + summarize only
        step.execute()
    smpc = os.path.join(out, "shared-most-probable-clusters.csv")
    assert os.path.exists(smpc)
    from dblink_amd.analysis.chain import read_clusters_csv

    clusters = read_clusters_csv(smpc)
    covered = set().union(*clusters) if clusters else set()
    assert len(covered) == 120  # sMPC covers every record exactly once
    assert sum(len(c) for c in clusters) == 120


REFERENCE_RLDATA500 = "/root/reference/examples/RLdata500.csv"


@pytest.mark.slow
@pytest.mark.skipif(not os.path.exists(REFERENCE_RLDATA500),
                    reason="reference RLdata500 not available")
def test_real_rldata500_with_reference_config(tmp_path):
    """THE parity benchmark: the reference's shipped RLdata500 dataset with
    its RLdata500.conf VERBATIM (only file paths redirected). The d-blink
    methodology reports ~0.9 pairwise F1 on this dataset; a correct
    implementation must land there."""
    src = open("/root/reference/examples/RLdata500.conf").read()
    out = str(tmp_path / "rl500")
    conf = (src
            .replace("./examples/RLdata500.csv", REFERENCE_RLDATA500)
            .replace("./examples/RLdata500_results/", out + "/"))
    conf_path = tmp_path / "rl500.conf"
    conf_path.write_text(conf)
    cfg = hocon.parse_file(str(conf_path))
    project = Project(cfg, rank=0, world_size=1)
    os.makedirs(project.output_path, exist_ok=True)
    for step in parse_steps(cfg, project):
        step.execute()
    txt = open(os.path.join(out, "evaluation-results.txt")).read()
    f1 = float([l for l in txt.splitlines() if "F1-score" in l][0].split(":")[1])
    ari = float([l for l in txt.splitlines() if "Rand" in l][0].split(":")[1])
    # measured 0.907/0.907 (seed fixed by the config); generous floor
    assert f1 > 0.8, txt
    assert ari > 0.8, txt


REFERENCE_RLDATA10000 = "/root/reference/examples/RLdata10000.csv"


@pytest.mark.slow
# runs in the default suite since the native CPU sweeps brought it from
# ~7 min to ~25 s; DBLINK_SLOW_TESTS=0 opts out on very slow machines
@pytest.mark.skipif(os.environ.get("DBLINK_SLOW_TESTS") == "0",
                    reason="opted out with DBLINK_SLOW_TESTS=0")
@pytest.mark.skipif(not os.path.exists(REFERENCE_RLDATA10000),
                    reason="reference RLdata10000 not available")
def test_real_rldata10000_published_quality(tmp_path):
    """The reference's shipped RLdata10000 dataset with its RLdata10000.conf,
    distortion prior tightened to Beta(10, 10000) and a 4,000-iteration
    burn-in (the shipped demo prior holds a recall-favouring F1 0.76
    operating point even at 21k iterations — FP analysis in BENCH.md).
    The d-blink methodology reports ~0.94 pairwise F1 for this dataset;
    measured 0.944 here."""
    src = open("/root/reference/examples/RLdata10000.conf").read()
    out = str(tmp_path / "rl10k")
    conf = (src
            .replace("./examples/RLdata10000.csv", REFERENCE_RLDATA10000)
            .replace("./examples/RLdata10000_results/", out + "/")
            .replace("/tmp/spark_checkpoint/", str(tmp_path / "ckpt") + "/")
            .replace("lowDistortion : {alpha : 10.0, beta : 1000.0}",
                     "lowDistortion : {alpha : 10.0, beta : 10000.0}")
            .replace("burninInterval : 0", "burninInterval : 4000")
            .replace("lowerIterationCutoff : 100",
                     "lowerIterationCutoff : 4000"))
    conf_path = tmp_path / "rl10k.conf"
    conf_path.write_text(conf)
    cfg = hocon.parse_file(str(conf_path))
    project = Project(cfg, rank=0, world_size=1)
    os.makedirs(project.output_path, exist_ok=True)
    for step in parse_steps(cfg, project):
        step.execute()
    txt = open(os.path.join(out, "evaluation-results.txt")).read()
    f1 = float([l for l in txt.splitlines() if "F1-score" in l][0].split(":")[1])
    assert f1 > 0.9, txt


def test_check_config_warns_on_quadratic_sampler(tmp_path, capsys):
    """--check surfaces the PCG-II quadratic-partition warning and rejects
    configs beyond native-path limits."""
    from dblink_amd.api.project import check_config
    from dblink_amd.utils.synthdata import write_csv

    data = str(tmp_path / "d.csv")
    write_csv(data, 300, dup_fraction=0.1, seed=4)
    conf = tmp_path / "p.conf"
    conf.write_text("""
    dblink : {
      pr : {alpha : 0.5, beta : 50.0}
      data : { path : "%s", recordIdentifier : "rec_id", nullValue : "NA",
               matchingAttributes : [
        {name : "by", similarityFunction : {name : "ConstantSimilarityFn"}, distortionPrior : ${dblink.pr}},
        {name : "fname_c1", similarityFunction : {name : "LevenshteinSimilarityFn", parameters : {threshold : 7.0, maxSimilarity : 10.0}}, distortionPrior : ${dblink.pr}} ] }
      randomSeed : 7
      engine : "cpu"
      partitioner : {name : "KDTreePartitioner",
                     parameters : {numLevels : 0, matchingAttributes : []}}
      outputPath : "%s/out/"
      checkpointPath : "%s/ckpt/"
      steps : [{name : "sample", parameters : {sampleSize : 1, sampler : "PCG-II"}}]
    }
    """ % (data, tmp_path, tmp_path))
    rc = check_config(str(conf))
    out = capsys.readouterr().out
    assert rc == 0
    # 300 records in 1 partition is fine; now patch the threshold down to
    # force the warning text through the same path
    import dblink_amd.api.project as prj

    class FakeStep(prj.SampleStep):
        pass

    # direct unit check of the warning rule
    cfg = prj.hocon.parse_file(str(conf))
    project = prj.Project(cfg, rank=0, world_size=1)
    step = prj.parse_steps(cfg, project)[0]
    assert step.scale_warning(num_records=10_000_000) is not None
    assert step.scale_warning(num_records=300) is None


def test_strings8_schema_cpu_chain():
    """The 8-Levenshtein-attribute schema (BASELINE config #4's shape)
    through the CPU engine: exercises multi-attribute od-masks, the native
    link/value kernels at A=8, and summary bookkeeping."""
    import os
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from bench import build_cache_and_records

    from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
    from dblink_amd.engine.init import deterministic_init
    from dblink_amd.parallel.partitioning import KDTreePartitioner

    cache, rv, rf = build_cache_and_records(400, seed=13, schema="strings8")
    assert rv.shape[1] == 8
    partitioner = KDTreePartitioner(1, [0])
    state = deterministic_init(rv, rf, np.arange(400, dtype=np.int64),
                               cache, partitioner, seed=13)
    engine = CpuEngine(cache, partitioner)
    engine.initial_summary(state)
    flags = SamplerFlags.for_sampler("PCG-I")
    lls = []
    for _ in range(30):
        engine.step(state, flags)
        lls.append(state.summary.log_likelihood)
    assert all(np.isfinite(lls))
    assert state.summary.agg_distortions.shape == (8, 1)
    # conservation + value sanity
    assert state.rec_ent.min() >= 0 and state.rec_ent.max() < state.num_entities
    for a in range(8):
        V = cache.indexed_attributes[a].index.num_values
        assert state.ent_values[:, a].min() >= 0
        assert state.ent_values[:, a].max() < V
