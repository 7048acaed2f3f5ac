"""HOCON parser tests, including the exact shape of the reference's example
configs (examples/RLdata500.conf structure)."""

import os
import textwrap

import pytest

from dblink_amd.utils import hocon

EXAMPLE = textwrap.dedent(
    """
    dblink : {
        // comment
        lowDistortion : {alpha : 0.5, beta : 50.0}
        constSimFn : {
            name : "ConstantSimilarityFn",
        }
        levSimFn : {
            name : "LevenshteinSimilarityFn",
            parameters : {
                threshold : 7.0
                maxSimilarity : 10.0
            }
        }
        data : {
            path : "./examples/data.csv"
            recordIdentifier : "rec_id",
            entityIdentifier : "ent_id" // optional
            nullValue : "NA"
            matchingAttributes : [
                {name : "by", similarityFunction : ${dblink.constSimFn}, distortionPrior : ${dblink.lowDistortion}},
                {name : "fname_c1", similarityFunction : ${dblink.levSimFn}, distortionPrior : ${dblink.lowDistortion}}
            ]
        }
        randomSeed : 319158
        expectedMaxClusterSize : 10
        partitioner : {
            name : "KDTreePartitioner",
            parameters : {
                numLevels : 0,
                matchingAttributes : []
            }
        }
        outputPath : "./out/"
        checkpointPath : "/tmp/ckpt/"
        steps : [
            {name : "sample", parameters : {sampleSize : 100, burninInterval : 0,
                thinningInterval : 10, resume : false, sampler : "PCG-I"}},
            {name : "evaluate", parameters : {lowerIterationCutoff : 100,
                metrics : ["pairwise", "cluster"], useExistingSMPC : false}}
        ]
    }
    """
)


def test_parse_example():
    cfg = hocon.parse_string(EXAMPLE)
    assert cfg.get_string("dblink.data.path") == "./examples/data.csv"
    assert cfg.get_long("dblink.randomSeed") == 319158
    assert cfg.get_int("dblink.expectedMaxClusterSize") == 10
    attrs = cfg.get_config_list("dblink.data.matchingAttributes")
    assert len(attrs) == 2
    # substitution resolution
    assert attrs[0].get_string("similarityFunction.name") == "ConstantSimilarityFn"
    assert attrs[1].get_double("similarityFunction.parameters.threshold") == 7.0
    assert attrs[0].get_double("distortionPrior.alpha") == 0.5
    steps = cfg.get_config_list("dblink.steps")
    assert steps[0].get_string("name") == "sample"
    assert steps[0].get_int("parameters.sampleSize") == 100
    assert steps[0].get_or("parameters.resume", True) is False
    assert steps[1].get_string_list("parameters.metrics") == ["pairwise", "cluster"]


def test_missing_and_defaults():
    cfg = hocon.parse_string("a { b : 1 }")
    assert cfg.get_int("a.b") == 1
    assert not cfg.has_path("a.c")
    assert cfg.get_or("a.c", 42) == 42
    with pytest.raises(hocon.ConfigMissingError):
        cfg.get("a.c")


def test_newline_separated_and_equals():
    cfg = hocon.parse_string("a = 1\nb : two\nc { d = true }\n")
    assert cfg.get_int("a") == 1
    assert cfg.get_string("b") == "two"
    assert cfg.get_bool("c.d") is True


def test_duplicate_key_merge():
    cfg = hocon.parse_string("a { x : 1 }\na { y : 2 }\n")
    assert cfg.get_int("a.x") == 1
    assert cfg.get_int("a.y") == 2


def test_dotted_keys():
    cfg = hocon.parse_string("a.b.c : 3")
    assert cfg.get_int("a.b.c") == 3


def test_unquoted_strings_and_numbers():
    cfg = hocon.parse_string("x : hello\ny : 2.5\nz : -3\nw : null")
    assert cfg.get_string("x") == "hello"
    assert cfg.get_double("y") == 2.5
    assert cfg.get_int("z") == -3
    assert cfg.get("w") is None


def test_project_rejects_empty_matching_attributes(tmp_path):
    from dblink_amd.api.project import Project
    from dblink_amd.utils import hocon as h

    cfg = h.parse_string(
        """
        dblink : {
          data : { path : "x.csv", recordIdentifier : "id",
                   matchingAttributes : [] }
          randomSeed : 1
          expectedMaxClusterSize : 4
          partitioner : {name : "KDTreePartitioner",
                         parameters : {numLevels : 0, matchingAttributes : []}}
          outputPath : "o/"
          checkpointPath : "c/"
          steps : []
        }
        """
    )
    import pytest

    with pytest.raises(ValueError, match="matchingAttributes"):
        Project(cfg, rank=0, world_size=1)


def test_hocon_parser_robustness_randomized():
    """Property test: the HOCON subset parser round-trips simple typed values
    and never crashes un-pythonically on random key/value combinations."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from dblink_amd.utils import hocon as h

    key = st.text(alphabet="abcXYZ", min_size=1, max_size=8)

    @settings(max_examples=40, deadline=None)
    @given(
        k1=key, k2=key,
        i=st.integers(min_value=-10**9, max_value=10**9),
        f=st.floats(min_value=-1e6, max_value=1e6, allow_nan=False),
        s=st.text(alphabet="abc XYZ_-.", max_size=20),
        b=st.booleans(),
    )
    def check(k1, k2, i, f, s, b):
        text = (
            "root : {\n"
            f"  {k1} : {{ i : {i}, f : {f!r}, s : {json_str(s)}, b : {str(b).lower()} }}\n"
            f"  {k2}2 : ${{root.{k1}.i}}\n"
            "}\n"
        )
        cfg = h.parse_string(text)
        assert cfg.get_int(f"root.{k1}.i") == i
        assert cfg.get_double(f"root.{k1}.f") == pytest.approx(f)
        assert cfg.get_string(f"root.{k1}.s") == s
        assert cfg.get_bool(f"root.{k1}.b") is b
        assert cfg.get_int(f"root.{k2}2") == i

    def json_str(x):
        import json

        return json.dumps(x)

    check()


def test_cli_check_mode(tmp_path):
    """`python -m dblink_amd --check conf` validates without running."""
    import subprocess
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import os as _os

    from dblink_amd.utils.synthdata import write_csv

    data = str(tmp_path / "d.csv")
    write_csv(data, 50, dup_fraction=0.1, seed=1)
    conf = tmp_path / "p.conf"
    conf.write_text(
        """
        dblink : {
          data : { path : \"""" + data + """\", recordIdentifier : "rec_id",
                   matchingAttributes : [
                     {name : "fname_c1",
                      similarityFunction : {name : "LevenshteinSimilarityFn",
                        parameters : {threshold : 7.0, maxSimilarity : 10.0}},
                      distortionPrior : {alpha : 0.5, beta : 50.0}} ] }
          randomSeed : 1
          partitioner : {name : "KDTreePartitioner",
                         parameters : {numLevels : 0, matchingAttributes : []}}
          outputPath : \"""" + str(tmp_path / "o") + """/\"
          checkpointPath : \"""" + str(tmp_path / "c") + """/\"
          steps : [{name : "summarize", parameters :
                    {lowerIterationCutoff : 0, quantities : ["partition-sizes"]}}]
        }
        """
    )
    from dblink_amd.api.cli import main as cli_main

    assert cli_main(["--check", str(conf)]) == 0
    conf2 = tmp_path / "bad.conf"
    conf2.write_text(conf.read_text().replace(data, str(tmp_path / "nope.csv")))
    assert cli_main(["--check", str(conf2)]) == 1
