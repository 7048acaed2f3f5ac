"""Convergence diagnostics over the chain's diagnostics CSV.

The reference derives its throughput from the ``systemTime-ms`` column of
``diagnostics.csv``; this module reproduces that measurement
and adds effective-sample-size (ESS) estimates — the BASELINE.json secondary
metric (ESS/sec) — using the initial monotone positive-sequence estimator
(Geyer 1992), the standard estimator for reversible MCMC.

CLI: ``python -m dblink_amd.analysis.diagnostics <outputPath>``
"""

from __future__ import annotations

import csv
import os

import numpy as np


def read_diagnostics(output_path):
    """Parse diagnostics.csv -> {column -> float ndarray}."""
    path = os.path.join(output_path, "diagnostics.csv")
    with open(path, newline="") as f:
        reader = csv.DictReader(f)
        rows = list(reader)
    out = {}
    for k in rows[0]:
        out[k] = np.array([float(r[k]) for r in rows])
    return out


def ess(x: np.ndarray) -> float:
    """Effective sample size via the initial monotone positive sequence
    estimator over pairwise autocovariance sums (Geyer 1992)."""
    x = np.asarray(x, dtype=np.float64)
    n = x.size
    if n < 4:
        return float(n)
    x = x - x.mean()
    var = np.dot(x, x) / n
    if var == 0:
        return float(n)
    # autocovariances via FFT
    m = 1 << (2 * n - 1).bit_length()
    f = np.fft.rfft(x, m)
    acov = np.fft.irfft(f * np.conj(f), m)[:n].real / n
    rho = acov / var
    # pair sums Gamma_k = rho_{2k} + rho_{2k+1}; truncate at first negative,
    # then enforce monotone decrease
    gammas = []
    k = 0
    while 2 * k + 1 < n:
        g = rho[2 * k] + rho[2 * k + 1]
        if g <= 0:
            break
        gammas.append(g)
        k += 1
    for i in range(1, len(gammas)):
        gammas[i] = min(gammas[i], gammas[i - 1])
    tau = -1.0 + 2.0 * sum(gammas)
    tau = max(tau, 1.0 / n)
    return float(min(n, n / tau))


def summarize(output_path):
    """Throughput + ESS summary in the reference's measurement terms."""
    d = read_diagnostics(output_path)
    t = d["systemTime-ms"]
    iters = d["iteration"]
    out = {}
    if len(t) >= 2 and t[-1] > t[0]:
        # iterations/sec from the diagnostics timestamps, exactly as one
        # would measure the reference (DiagnosticsWriter.scala:64)
        out["iterations_per_sec"] = float(
            (iters[-1] - iters[0]) / ((t[-1] - t[0]) / 1000.0)
        )
    wall = (t[-1] - t[0]) / 1000.0 if len(t) >= 2 else float("nan")
    for col in ("logLikelihood", "numObservedEntities"):
        if col in d:
            e = ess(d[col])
            out[f"ess_{col}"] = e
            if wall and wall > 0:
                out[f"ess_{col}_per_sec"] = e / wall
    return out


def main(argv=None):
    import argparse
    import json

    ap = argparse.ArgumentParser(description="chain diagnostics summary")
    ap.add_argument("output_path")
    args = ap.parse_args(argv)
    print(json.dumps(summarize(args.output_path), indent=2))


if __name__ == "__main__":
    main()
