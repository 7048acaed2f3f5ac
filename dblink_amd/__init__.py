"""dblink_amd — MI355X-native distributed Bayesian entity resolution.

A from-scratch rebuild of the capabilities of cleanzr/dblink (Spark/Scala)
for a single 8-GPU AMD MI355X node: partitioned Gibbs sampling over latent
entities with PyTorch-ROCm orchestration, hand-written CDNA4 (gfx950) HIP
kernels for the hot paths, and RCCL over xGMI for cross-partition entity
migration and summary reductions.
"""

__version__ = "0.1.0"


def run_config(path):
    """Execute a HOCON project config (the CLI's entry point, importable)."""
    from .api.project import run_config as _run

    return _run(path)
