"""CSV loader edge cases (parity: Project.scala:173-180 Spark read options —
header, DROPMALFORMED, nullValue — and State.scala:359-375 file ids)."""

import numpy as np
import pytest

from dblink_amd.models.records import load_csv


def _write(path, text):
    path.write_text(text)
    return str(path)


def test_malformed_rows_dropped(tmp_path):
    p = _write(tmp_path / "a.csv",
               "rec_id,name,year\n"
               "1,ANNA,1950\n"
               "2,BOB\n"             # wrong column count -> dropped
               "3,CLAIRE,1960,extra\n"  # wrong column count -> dropped
               "4,DAVE,1970\n")
    table, _ = load_csv(p, "rec_id", None, ["name", "year"])
    assert table.num_records == 2
    assert list(table.rec_ids) == ["1", "4"]


def test_null_value_and_empty_become_missing(tmp_path):
    p = _write(tmp_path / "a.csv",
               "rec_id,name,year\n"
               "1,NA,1950\n"
               "2,,1955\n"
               "3,EVE,NA\n")
    table, _ = load_csv(p, "rec_id", None, ["name", "year"], null_value="NA")
    name, year = table.columns
    assert name[0] is None and name[1] is None and name[2] == "EVE"
    assert year[2] is None


def test_multiple_files_and_file_ids(tmp_path):
    a = _write(tmp_path / "a.csv", "rec_id,src,name\n1,x,ANNA\n")
    b = _write(tmp_path / "b.csv", "rec_id,src,name\n2,y,BOB\n")
    table, _ = load_csv(f"{a}, {b}", "rec_id", "src", ["name"])
    assert table.num_records == 2
    assert list(table.file_ids) == ["x", "y"]
    # glob form
    table2, _ = load_csv(str(tmp_path / "*.csv"), "rec_id", None, ["name"])
    assert table2.num_records == 2
    assert set(table2.file_ids) == {"0"}


def test_missing_identifier_column_raises(tmp_path):
    p = _write(tmp_path / "a.csv", "id,name\n1,ANNA\n")
    with pytest.raises(ValueError, match="record identifier"):
        load_csv(p, "rec_id", None, ["name"])


def test_entity_identifier_optional(tmp_path):
    p = _write(tmp_path / "a.csv", "rec_id,name,ent\n1,ANNA,7\n2,ANNE,7\n")
    _, ents = load_csv(p, "rec_id", None, ["name"], ent_id_col="ent")
    assert ents == ["7", "7"]
    _, none_ents = load_csv(p, "rec_id", None, ["name"], ent_id_col="absent")
    assert none_ents is None


def test_all_missing_record_runs_through_both_cpu_paths():
    """A record with every attribute missing links uniformly and must not
    break either CPU sweep implementation."""
    import os
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from bench import build_cache_and_records
    from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
    from dblink_amd.engine.init import deterministic_init
    from dblink_amd.parallel.partitioning import KDTreePartitioner

    n = 60
    cache, rv, rf = build_cache_and_records(n, seed=4)
    rv = rv.copy()
    rv[3, :] = -1  # all attributes missing
    for fast in ("1", "0"):
        os.environ["DBLINK_CPU_FAST"] = fast
        try:
            partitioner = KDTreePartitioner(0, [])
            state = deterministic_init(rv, rf, np.arange(n, dtype=np.int64),
                                       cache, partitioner, seed=4)
            engine = CpuEngine(cache, partitioner)
            engine.initial_summary(state)
            targets = set()
            for _ in range(30):
                engine.step(state, SamplerFlags.for_sampler("PCG-I"))
                row = int(np.flatnonzero(state.rec_gid == 3)[0])  # rows permute
                e = int(state.rec_ent[row])
                # identify the target entity by its linked-record gid set
                targets.add(frozenset(state.rec_gid[state.rec_ent == e].tolist()))
            assert np.isfinite(state.summary.log_likelihood)
            assert len(targets) > 3  # uniform link draw actually moves
        finally:
            os.environ.pop("DBLINK_CPU_FAST", None)
