"""Synthetic RLdata-shaped dataset generator (vectorized).

The reference ships RLdata500 / RLdata10000 (German name + birth-date fields,
10% duplicates, distorted duplicates; ``README.md:25-34``). There is no
network access here and data files are not copied from the reference, so
benchmarks and end-to-end tests generate datasets of the same SHAPE:

- columns: fname_c1, lname_c1 (strings, Levenshtein-matched),
  by, bm, bd (birth year/month/day, constant-sim categorical)
- ``dup_fraction`` of the records are duplicates of an earlier record with a
  small number of attribute distortions (typo edits for names, redraws for
  dates)
- ground-truth entity ids are included (``ent_id``) for evaluation

Deterministic given ``seed``. Vectorized so 10M-record datasets build in
seconds (the per-duplicate typo loop touches only the ~10% duplicates).
"""

from __future__ import annotations

import numpy as np

# Syllable pools for generated names (not copied from RLdata).
_FIRST_SYLL = ["an", "be", "ca", "da", "el", "fi", "ga", "han", "in", "jo", "ka", "lu",
               "ma", "ni", "ol", "pe", "re", "sa", "to", "ul", "vi", "wa", "chris", "ste"]
_LAST_SYLL = ["bach", "berg", "mann", "stein", "feld", "hof", "meier", "schmid", "mueller",
              "weber", "wagner", "becker", "koch", "richter", "wolf", "schroe", "neu", "lang"]


def _make_pool(rng, syllables, n, min_parts=3):
    """n distinct names; the part count grows until the combinatorial space
    comfortably covers n (fixes the stall when n > |syllables|^3)."""
    syl = np.array(syllables, dtype=object)
    s = len(syllables)
    max_parts = min_parts
    cap = s ** max_parts
    while cap < 4 * n:
        max_parts += 1
        cap *= s
    pool = set()
    while len(pool) < n:
        batch = max(8192, 2 * (n - len(pool)))
        ks = rng.integers(min_parts, max_parts + 1, size=batch)
        idx = rng.integers(0, s, size=(batch, max_parts))
        for b in range(batch):
            pool.add("".join(syl[idx[b, : ks[b]]]).upper())
            if len(pool) >= n:
                break
    return np.array(sorted(pool), dtype=object)


def _typo(rng, s):
    """One random edit: substitute / delete / insert / transpose."""
    letters = "ABCDEFGHIJKLMNOPQRSTUVWXYZ"
    op = rng.integers(0, 4)
    i = int(rng.integers(0, len(s))) if len(s) else 0
    if op == 0 and len(s) > 0:  # substitute
        return s[:i] + letters[rng.integers(0, 26)] + s[i + 1 :]
    if op == 1 and len(s) > 1:  # delete
        return s[:i] + s[i + 1 :]
    if op == 2:  # insert
        return s[:i] + letters[rng.integers(0, 26)] + s[i:]
    if op == 3 and len(s) > 1 and i < len(s) - 1:  # transpose
        return s[:i] + s[i + 1] + s[i] + s[i + 2 :]
    return s


def generate(
    num_records: int,
    dup_fraction: float = 0.1,
    seed: int = 0,
    num_first: int | None = None,
    num_last: int | None = None,
    missing_fraction: float = 0.0,
    num_files: int = 1,
    extra_string_attrs: int = 0,
):
    """Return (columns dict of numpy object/str arrays, header list)."""
    rng = np.random.default_rng(seed)
    n_ent = int(round(num_records * (1.0 - dup_fraction)))
    n_dup = num_records - n_ent

    # Name-vocabulary size: n^0.8 at RLdata scales, saturating per Heaps'
    # law (~sqrt growth) beyond 10k entities. Real name vocabularies
    # saturate — RLdata10000 has 539 distinct first names for 10,000
    # records — and an unboundedly growing syllable-name pool makes the
    # domain's similarity structure unrealistically dense at 1M+ records
    # (phi-weighted similar-mass 2.8% with 68-entry sim rows at V = 63k,
    # vs 1.4% / 5-entry rows for real RLdata names; the capped pool
    # measures 1.9%). Datasets of <= ~10k records are unchanged.
    def vocab(n):
        base = 10_000 ** 0.8
        return max(30, int(n ** 0.8 if n <= 10_000 else base * (n / 10_000) ** 0.5))

    if num_first is None:
        num_first = vocab(n_ent)
    if num_last is None:
        num_last = vocab(n_ent)

    fpool = _make_pool(rng, _FIRST_SYLL, num_first)
    lpool = _make_pool(rng, _LAST_SYLL, num_last)
    # Zipf-ish frequency skew, like real name distributions
    fw = 1.0 / np.arange(1, len(fpool) + 1) ** 0.9
    lw = 1.0 / np.arange(1, len(lpool) + 1) ** 0.9
    fw /= fw.sum()
    lw /= lw.sum()

    ent_fname = rng.choice(fpool, size=n_ent, p=fw)
    ent_lname = rng.choice(lpool, size=n_ent, p=lw)
    ent_by = rng.integers(1900, 2000, size=n_ent)
    ent_bm = rng.integers(1, 13, size=n_ent)
    ent_bd = rng.integers(1, 29, size=n_ent)
    extra = [rng.choice(fpool, size=n_ent, p=fw) for _ in range(extra_string_attrs)]

    rows_ent = np.concatenate([np.arange(n_ent), rng.integers(0, n_ent, size=n_dup)])
    order = rng.permutation(num_records)
    ent_of_row = rows_ent[order]
    is_dup = order >= n_ent

    n_attr = 5 + extra_string_attrs
    fn = ent_fname[ent_of_row].copy()
    ln = ent_lname[ent_of_row].copy()
    by = ent_by[ent_of_row].copy()
    bm = ent_bm[ent_of_row].copy()
    bd = ent_bd[ent_of_row].copy()
    xv = [e[ent_of_row].copy() for e in extra]

    # distort 1-2 attributes of each duplicate (loop touches only duplicates)
    dup_rows = np.flatnonzero(is_dup)
    n_dist = rng.integers(1, 3, size=dup_rows.size)
    for row, nd in zip(dup_rows, n_dist):
        targets = rng.choice(n_attr, size=nd, replace=False)
        for t in targets:
            if t == 0:
                fn[row] = _typo(rng, fn[row])
            elif t == 1:
                ln[row] = _typo(rng, ln[row])
            elif t == 2:
                by[row] = rng.integers(1900, 2000)
            elif t == 3:
                bm[row] = rng.integers(1, 13)
            elif t == 4:
                bd[row] = rng.integers(1, 29)
            else:
                xv[t - 5][row] = _typo(rng, xv[t - 5][row])

    cols = {
        "rec_id": np.arange(1, num_records + 1).astype(str),
        "file_id": (np.arange(num_records) % num_files).astype(str),
        "ent_id": (ent_of_row + 1).astype(str),
        "fname_c1": fn.astype(object),
        "lname_c1": ln.astype(object),
        "by": by.astype(str).astype(object),
        "bm": bm.astype(str).astype(object),
        "bd": bd.astype(str).astype(object),
    }
    for j in range(extra_string_attrs):
        cols[f"xattr{j}"] = xv[j].astype(object)

    if missing_fraction > 0:
        attr_cols = ["fname_c1", "lname_c1", "by", "bm", "bd"] + [
            f"xattr{j}" for j in range(extra_string_attrs)
        ]
        for k in attr_cols:
            mask = rng.random(num_records) < missing_fraction
            c = cols[k]
            c[mask] = "NA"

    header = ["rec_id", "file_id", "ent_id", "fname_c1", "lname_c1", "by", "bm", "bd"] + [
        f"xattr{j}" for j in range(extra_string_attrs)
    ]
    return cols, header


def write_csv(path, num_records, **kw):
    cols, header = generate(num_records, **kw)
    with open(path, "w", encoding="utf-8") as f:
        f.write(",".join(header) + "\n")
        for i in range(num_records):
            f.write(",".join(str(cols[h][i]) for h in header) + "\n")
    return path
