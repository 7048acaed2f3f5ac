"""Distortion probabilities theta_{attribute, file} and their Gibbs update.

Parity: ``DistortionProbs.scala:27-44`` (prior-mean init) and
``GibbsUpdates.scala:305-320`` (Beta posterior draw on the host).

Stored as a dense float64 matrix [num_attributes, num_files].
"""

from __future__ import annotations

import numpy as np


class DistortionProbs:
    def __init__(self, probs: np.ndarray):
        self.probs = np.asarray(probs, dtype=np.float64)  # [A, F]

    @classmethod
    def from_prior_mean(cls, priors, num_files: int):
        """theta_{a,f} = alpha_a / (alpha_a + beta_a) (DistortionProbs.scala:33-43)."""
        probs = np.empty((len(priors), num_files), dtype=np.float64)
        for a, p in enumerate(priors):
            probs[a, :] = p.alpha / (p.alpha + p.beta)
        return cls(probs)

    def __call__(self, attr_id: int, file_id: int) -> float:
        return float(self.probs[attr_id, file_id])


def update_dist_probs(agg_distortions, priors, file_sizes_dense, rng) -> DistortionProbs:
    """theta_{a,f} ~ Beta(alpha_a + numDist, beta_a + numRecords_f - numDist)
    (GibbsUpdates.scala:305-320).

    ``agg_distortions`` is an int64 [A, F] matrix of distortion counts,
    ``file_sizes_dense`` an int64 [F] vector.
    """
    A, F = agg_distortions.shape
    probs = np.empty((A, F), dtype=np.float64)
    for a, p in enumerate(priors):
        eff_dist = agg_distortions[a].astype(np.float64) + p.alpha
        eff_non = file_sizes_dense.astype(np.float64) - agg_distortions[a] + p.beta
        probs[a, :] = rng.beta(eff_dist, eff_non)
    return DistortionProbs(probs)
