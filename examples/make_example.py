#!/usr/bin/env python3
"""Generate a synthetic RLdata-shaped example project: CSV + HOCON config.

The reference ships RLdata500/RLdata10000; this environment has no network
access, so an equivalent-shape dataset (with ground-truth `ent_id` for the
evaluate step) is generated instead.

    python examples/make_example.py --records 1000 --out /tmp/dblink_demo
    python -m dblink_amd /tmp/dblink_demo/project.conf
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dblink_amd.utils.synthdata import write_csv

CONF = """dblink : {{
    lowDistortion : {{alpha : {alpha}, beta : {beta}}}

    constSimFn : {{ name : "ConstantSimilarityFn" }}
    levSimFn : {{
        name : "LevenshteinSimilarityFn",
        parameters : {{ threshold : 7.0, maxSimilarity : 10.0 }}
    }}

    data : {{
        path : "{out}/records.csv"
        recordIdentifier : "rec_id",{file_id_line}
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

    partitioner : {{
        name : "KDTreePartitioner",
        parameters : {{ numLevels : {levels}, matchingAttributes : [{part_attrs}] }}
    }}

    outputPath : "{out}/results/"
    checkpointPath : "{out}/ckpt/"

    steps : [
        {{name : "sample", parameters : {{
            sampleSize : {samples}, burninInterval : {burnin},
            thinningInterval : {thin}, resume : false, sampler : "PCG-I"
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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--records", type=int, default=1000)
    ap.add_argument("--out", default="/tmp/dblink_demo")
    ap.add_argument("--levels", type=int, default=0)
    ap.add_argument("--samples", type=int, default=100)
    ap.add_argument("--burnin", type=int, default=100)
    ap.add_argument("--thin", type=int, default=10)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--files", type=int, default=1,
                    help="spread records over N source files (record linkage "
                         "with per-file distortion probabilities)")
    args = ap.parse_args()

    os.makedirs(args.out, exist_ok=True)
    write_csv(os.path.join(args.out, "records.csv"), args.records,
              dup_fraction=0.1, seed=args.seed, num_files=args.files)
    part_attrs = '"fname_c1"' if args.levels > 0 else ""
    file_id_line = '\n        fileIdentifier : "file_id",' if args.files > 1 else ""
    # distortion prior concentration scales with data size, mirroring the
    # reference's example configs (RLdata500.conf Beta(0.5, 50) at n=500,
    # RLdata10000.conf Beta(10, 1000)): a weak prior at large n lets the
    # chain drift to a high-distortion mode
    alpha, beta = (10.0, 1000.0) if args.records >= 5000 else (0.5, 50.0)
    conf = CONF.format(out=args.out, levels=args.levels, part_attrs=part_attrs,
                       samples=args.samples, burnin=args.burnin, thin=args.thin,
                       file_id_line=file_id_line, alpha=alpha, beta=beta,
                       cutoff=args.burnin + (args.samples * args.thin) // 2)
    with open(os.path.join(args.out, "project.conf"), "w") as f:
        f.write(conf)
    print(f"wrote {args.out}/records.csv and {args.out}/project.conf")


if __name__ == "__main__":
    main()
