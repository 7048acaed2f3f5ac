"""Synthetic RLdata-shaped dataset generator.

The reference ships RLdata500 / RLdata10000 (German name + birth-date fields,
10% duplicates, distorted duplicates; ``README.md:25-34``). There is no
network access here and data files are not copied from the reference, so
benchmarks and end-to-end tests generate datasets of the same SHAPE:

- columns: fname_c1, lname_c1 (strings, Levenshtein-matched),
  by, bm, bd (birth year/month/day, constant-sim categorical)
- `dup_fraction` of the records are duplicates of an earlier record with a
  small number of attribute distortions (typo edits for names, redraws for
  dates)
- ground-truth entity ids are included (`ent_id`) for evaluation

Deterministic given `seed`.
"""

from __future__ import annotations

import numpy as np

# Frequency-skewed synthetic name pools (generated, not copied from RLdata).
_FIRST_SYLL = ["an", "be", "ca", "da", "el", "fi", "ga", "han", "in", "jo", "ka", "lu",
               "ma", "ni", "ol", "pe", "re", "sa", "to", "ul", "vi", "wa", "chris", "ste"]
_LAST_SYLL = ["bach", "berg", "mann", "stein", "feld", "hof", "meier", "schmid", "mueller",
              "weber", "wagner", "becker", "koch", "richter", "wolf", "schroe", "neu", "lang"]


def _make_pool(rng, syllables, n, min_parts=2, max_parts=3):
    pool = set()
    while len(pool) < n:
        k = rng.integers(min_parts, max_parts + 1)
        name = "".join(rng.choice(syllables) for _ in range(k))
        pool.add(name.upper())
    return sorted(pool)


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
    """Return (columns dict, header list) for an RLdata-shaped dataset.

    Entities get distinct ids; each duplicate record distorts 1-2 attributes
    of its entity's true values.
    """
    rng = np.random.default_rng(seed)
    n_ent = int(round(num_records * (1.0 - dup_fraction)))
    n_dup = num_records - n_ent

    if num_first is None:
        num_first = max(30, int(n_ent ** 0.8))
    if num_last is None:
        num_last = max(30, int(n_ent ** 0.8))

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
    extra = [
        rng.choice(fpool, size=n_ent, p=fw) for _ in range(extra_string_attrs)
    ]

    rows_ent = list(range(n_ent)) + list(rng.integers(0, n_ent, size=n_dup))
    order = rng.permutation(num_records)

    cols = {k: [] for k in ["rec_id", "file_id", "ent_id", "fname_c1", "lname_c1", "by", "bm", "bd"]}
    for j in range(extra_string_attrs):
        cols[f"xattr{j}"] = []

    for out_i, idx in enumerate(order):
        e = rows_ent[idx]
        is_dup = idx >= n_ent
        fn, ln = str(ent_fname[e]), str(ent_lname[e])
        by, bm, bd = int(ent_by[e]), int(ent_bm[e]), int(ent_bd[e])
        xvals = [str(extra[j][e]) for j in range(extra_string_attrs)]
        if is_dup:
            # distort 1-2 attributes
            n_dist = int(rng.integers(1, 3))
            targets = rng.choice(5 + extra_string_attrs, size=n_dist, replace=False)
            for t in targets:
                if t == 0:
                    fn = _typo(rng, fn)
                elif t == 1:
                    ln = _typo(rng, ln)
                elif t == 2:
                    by = int(rng.integers(1900, 2000))
                elif t == 3:
                    bm = int(rng.integers(1, 13))
                elif t == 4:
                    bd = int(rng.integers(1, 29))
                else:
                    xvals[t - 5] = _typo(rng, xvals[t - 5])
        vals = {"fname_c1": fn, "lname_c1": ln, "by": str(by), "bm": str(bm), "bd": str(bd)}
        for j in range(extra_string_attrs):
            vals[f"xattr{j}"] = xvals[j]
        if missing_fraction > 0:
            for k in list(vals):
                if rng.random() < missing_fraction:
                    vals[k] = "NA"
        cols["rec_id"].append(str(out_i + 1))
        cols["file_id"].append(str(out_i % num_files))
        cols["ent_id"].append(str(e + 1))
        for k, v in vals.items():
            cols[k].append(v)

    header = ["rec_id", "file_id", "ent_id", "fname_c1", "lname_c1", "by", "bm", "bd"] + [
        f"xattr{j}" for j in range(extra_string_attrs)
    ]
    return cols, header


def write_csv(path, num_records, **kw):
    cols, header = generate(num_records, **kw)
    with open(path, "w", encoding="utf-8") as f:
        f.write(",".join(header) + "\n")
        for i in range(num_records):
            f.write(",".join(cols[h][i] for h in header) + "\n")
    return path
