"""Data model: attribute specs, record loading, dictionary encoding, and the
records cache (per-attribute indexes + file statistics).

Capability parity with the reference:
- ``Attribute`` / ``IndexedAttribute``: ``package.scala:128-158``
- ``RecordsCache``: ``RecordsCache.scala:34-135`` (value counts, file sizes,
  missing counts, per-attribute ``AttributeIndex`` with precached powers,
  string -> value-id record transform; missing values encode as -1)
- CSV loading with header + nullValue handling: ``Project.scala:173-180``

The MI355X build is columnar: records are numpy object columns, counted and
dictionary-encoded with pandas factorize (C-speed hashing) so 10M-record
datasets encode in seconds, then held as dense int32 matrices.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

from .attribute_index import AttributeIndex
from .similarity import SimilarityFn


@dataclass(frozen=True)
class BetaShapeParameters:
    alpha: float
    beta: float

    def __post_init__(self):
        if self.alpha <= 0 or self.beta <= 0:
            raise ValueError("shape parameters must be positive")

    def mk_string(self):
        return f"BetaShapeParameters(alpha={self.alpha}, beta={self.beta})"


@dataclass(frozen=True)
class Attribute:
    name: str
    similarity_fn: SimilarityFn
    distortion_prior: BetaShapeParameters

    @property
    def is_constant(self):
        return self.similarity_fn.is_constant


class RecordsTable:
    """Raw records, columnar: string ids, file ids, and one object column per
    matching attribute (None = missing)."""

    def __init__(self, rec_ids, file_ids, columns):
        self.rec_ids = np.asarray(rec_ids, dtype=object)
        self.file_ids = np.asarray(file_ids, dtype=object)
        self.columns = [np.asarray(c, dtype=object) for c in columns]

    @classmethod
    def from_rows(cls, rec_ids, file_ids, rows):
        """rows: list of per-record value lists (None = missing)."""
        if rows:
            cols = [np.array([r[a] for r in rows], dtype=object) for a in range(len(rows[0]))]
        else:
            cols = []
        return cls(rec_ids, file_ids, cols)

    @property
    def num_records(self):
        return len(self.rec_ids)

    @property
    def num_attributes(self):
        return len(self.columns)


def load_csv(path, rec_id_col, file_id_col, attribute_names, null_value="NA", ent_id_col=None):
    """Load one or more CSV files into a RecordsTable.

    ``path`` may be a single file or a comma-separated list of files / globs.
    When ``file_id_col`` is None, the file id of every record is "0"
    (``State.scala:359-375``). Malformed rows (wrong column count) are dropped,
    matching Spark's DROPMALFORMED mode (``Project.scala:177``).

    Returns (RecordsTable, ent_ids or None).
    """
    import csv
    import glob

    paths = []
    for p in str(path).split(","):
        p = p.strip()
        expanded = sorted(glob.glob(p)) if any(ch in p for ch in "*?[") else [p]
        paths.extend(expanded)

    rec_ids, file_ids, rows, ent_ids = [], [], [], []
    for p in paths:
        with open(p, "r", encoding="utf-8", newline="") as f:
            reader = csv.reader(f)
            header = next(reader)
            col = {name: i for i, name in enumerate(header)}
            try:
                rid_i = col[rec_id_col]
            except KeyError:
                raise ValueError(f"record identifier column {rec_id_col!r} not in {p}")
            fid_i = col[file_id_col] if file_id_col is not None else None
            eid_i = col[ent_id_col] if ent_id_col is not None and ent_id_col in col else None
            attr_is = [col[a] for a in attribute_names]
            ncols = len(header)
            for row in reader:
                if len(row) != ncols:
                    continue  # DROPMALFORMED
                rec_ids.append(row[rid_i])
                file_ids.append(row[fid_i] if fid_i is not None else "0")
                rows.append(
                    [None if row[i] == null_value or row[i] == "" else row[i] for i in attr_is]
                )
                if eid_i is not None:
                    ent_ids.append(row[eid_i])
    table = RecordsTable.from_rows(rec_ids, file_ids, rows)
    return table, (ent_ids if ent_ids else None)


class RecordsCache:
    """Per-attribute domain indexes plus file-level statistics."""

    def __init__(self, indexed_attributes, file_sizes, missing_counts=None):
        self.indexed_attributes = list(indexed_attributes)
        self.file_sizes = dict(file_sizes)  # {file_id: count}
        self.missing_counts = missing_counts or {}
        self.file_ids = sorted(self.file_sizes)  # stable order; id -> dense int
        self.file_id_to_int = {f: i for i, f in enumerate(self.file_ids)}

    @property
    def num_records(self):
        return sum(self.file_sizes.values())

    @property
    def num_attributes(self):
        return len(self.indexed_attributes)

    @property
    def num_files(self):
        return len(self.file_ids)

    def distortion_prior(self):
        return [a.spec.distortion_prior for a in self.indexed_attributes]

    @classmethod
    def build(cls, table: RecordsTable, attributes, max_cluster_size: int, pair_sweep=None):
        """Columnar counting pass, then per-attribute index build
        (``RecordsCache.scala:68-118``)."""
        import pandas as pd

        file_ser = pd.Series(table.file_ids)
        file_sizes = file_ser.value_counts().to_dict()

        missing_counts = {}
        indexed = []
        for a, spec in enumerate(attributes):
            ser = pd.Series(table.columns[a])
            miss = ser.isna()
            if miss.any():
                for fid, cnt in file_ser[miss].value_counts().items():
                    missing_counts[(fid, a)] = int(cnt)
            counts = ser.value_counts()
            index = AttributeIndex(
                {k: float(v) for k, v in counts.items()},
                spec.similarity_fn,
                precache_powers=max_cluster_size,
                pair_sweep=pair_sweep,
            )
            indexed.append(IndexedAttribute(spec, index))
        return cls(indexed, file_sizes, missing_counts)

    def transform_records(self, table: RecordsTable):
        """Encode string values to dense int32 value ids; missing -> -1
        (``RecordsCache.scala:120-134``). Also returns dense int32 file ids."""
        import pandas as pd

        R, A = table.num_records, self.num_attributes
        out = np.full((R, A), -1, dtype=np.int32)
        for a in range(A):
            codes, uniques = pd.factorize(pd.Series(table.columns[a]))
            sid = self.indexed_attributes[a].index._string_to_id
            mapping = np.fromiter(
                (sid.get(u, -1) for u in uniques), dtype=np.int32, count=len(uniques)
            )
            mapping = np.concatenate([mapping, [-1]])  # codes == -1 (missing)
            out[:, a] = mapping[codes]
        file_codes, file_uniques = pd.factorize(pd.Series(table.file_ids))
        fmap = np.fromiter(
            (self.file_id_to_int[u] for u in file_uniques), dtype=np.int32,
            count=len(file_uniques),
        )
        files = fmap[file_codes]
        return out, files.astype(np.int32)


@dataclass
class IndexedAttribute:
    spec: Attribute
    index: AttributeIndex

    @property
    def name(self):
        return self.spec.name

    @property
    def is_constant(self):
        return self.spec.is_constant

    @property
    def distortion_prior(self):
        return self.spec.distortion_prior
