"""Evaluation metrics: pairwise precision/recall/F1 and adjusted Rand index.

Parity: ``analysis/PairwiseMetrics.scala``, ``analysis/ClusteringMetrics.scala``,
``analysis/ClusteringContingencyTable.scala``, ``analysis/BinaryConfusionMatrix.scala``,
``analysis/BinaryClassificationMetrics.scala``.
"""

from __future__ import annotations

from collections import Counter, defaultdict
from itertools import combinations


def to_pairwise_links(clusters):
    """Set of canonical (a, b) string pairs, a < b (analysis/package.scala:70-75)."""
    links = set()
    for cluster in clusters:
        for a, b in combinations(sorted(cluster), 2):
            links.add((a, b))
    return links


def to_membership(clusters):
    out = {}
    for i, cluster in enumerate(clusters):
        for rid in cluster:
            out[rid] = i
    return out


def membership_to_clusters(membership):
    """{record -> label} -> list of clusters (analysis/package.scala:52-63)."""
    agg = defaultdict(set)
    for rid, label in membership.items():
        agg[label].add(rid)
    return list(agg.values())


class BinaryConfusionMatrix:
    def __init__(self, tp, fp, fn):
        self.TP, self.FP, self.FN = tp, fp, fn

    @property
    def P(self):
        return self.TP + self.FN

    @property
    def PP(self):
        return self.TP + self.FP


def links_confusion_matrix(predicted_links, true_links):
    tp = len(predicted_links & true_links)
    fp = len(predicted_links - true_links)
    fn = len(true_links - predicted_links)
    return BinaryConfusionMatrix(tp, fp, fn)


def precision(cm):
    return cm.TP / cm.PP if cm.PP else float("nan")


def recall(cm):
    return cm.TP / cm.P if cm.P else float("nan")


def f_measure(cm, beta=1.0):
    b2 = beta * beta
    pr, re = precision(cm), recall(cm)
    denom = b2 * pr + re
    return (1 + b2) * pr * re / denom if denom else float("nan")


class PairwiseMetrics:
    def __init__(self, precision_, recall_, f1):
        self.precision = precision_
        self.recall = recall_
        self.f1score = f1

    @classmethod
    def compute(cls, predicted_clusters, true_clusters):
        cm = links_confusion_matrix(
            to_pairwise_links(predicted_clusters), to_pairwise_links(true_clusters)
        )
        return cls(precision(cm), recall(cm), f_measure(cm, 1.0))

    def mk_string(self):
        return (
            "=====================================\n"
            "          Pairwise metrics           \n"
            "-------------------------------------\n"
            f" Precision:       {self.precision}\n"
            f" Recall:          {self.recall}\n"
            f" F1-score:        {self.f1score}\n"
            "=====================================\n"
        )


def _comb2(x):
    return x * (x - 1) // 2 if x >= 2 else 0


def adjusted_rand_index(predicted_clusters, true_clusters):
    """Sparse contingency-table ARI (ClusteringMetrics.scala:44-74)."""
    pred_m = to_membership(predicted_clusters)
    true_m = to_membership(true_clusters)
    if set(pred_m) != set(true_m):
        raise ValueError("Clusterings do not partition the same set of elements.")
    n = len(true_m)
    table = Counter()
    for rid, pu in pred_m.items():
        table[(pu, true_m[rid])] += 1
    pred_sums = Counter()
    true_sums = Counter()
    total_comb = 0
    for (pu, tu), c in table.items():
        pred_sums[pu] += c
        true_sums[tu] += c
        total_comb += _comb2(c)
    pred_comb = sum(_comb2(c) for c in pred_sums.values())
    true_comb = sum(_comb2(c) for c in true_sums.values())
    expected = pred_comb * true_comb / _comb2(n)
    max_index = (pred_comb + true_comb) / 2.0
    return (total_comb - expected) / (max_index - expected)


class ClusteringMetrics:
    def __init__(self, ari):
        self.adj_rand_index = ari

    @classmethod
    def compute(cls, predicted_clusters, true_clusters):
        return cls(adjusted_rand_index(predicted_clusters, true_clusters))

    def mk_string(self):
        return (
            "=====================================\n"
            "          Cluster metrics            \n"
            "-------------------------------------\n"
            f" Adj. Rand index: {self.adj_rand_index}\n"
            "=====================================\n"
        )
