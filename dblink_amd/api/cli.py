"""Command-line entry: ``python -m dblink_amd <config.conf>``.

Single-process by default; multi-GPU via
``torchrun --standalone --nproc-per-node N -m dblink_amd <config.conf>``
(one rank per MI355X GPU over RCCL).
"""

from __future__ import annotations

import argparse
import logging
import sys


def main(argv=None):
    parser = argparse.ArgumentParser(
        prog="dblink_amd", description="MI355X-native distributed Bayesian entity resolution"
    )
    parser.add_argument("config", help="path to HOCON configuration file")
    parser.add_argument("-v", "--verbose", action="store_true")
    parser.add_argument("--check", action="store_true",
                        help="validate the configuration (and that the data "
                             "files and columns exist) without running")
    from .. import __version__

    parser.add_argument("--version", action="version",
                        version=f"dblink_amd {__version__}")
    args = parser.parse_args(argv)

    logging.basicConfig(
        level=logging.DEBUG if args.verbose else logging.INFO,
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )
    if args.check:
        from .project import check_config

        return check_config(args.config)
    from .project import run_config

    run_config(args.config)
    return 0


if __name__ == "__main__":
    sys.exit(main())
