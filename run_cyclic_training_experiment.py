#!/usr/bin/env python3
"""Cyclic-training experiment driver (reference:
run_cyclic_training_experiment.py — same skeleton as run_experiment.py
with CyclicPruningHarness; the reference's cyclic path crashes as shipped
(SURVEY §2.6.3), this one works).
"""

from __future__ import annotations

import sys

from run_experiment import run
from turboprune_amd.config import compose
from turboprune_amd.config.compose import parse_cli
from turboprune_amd.harness import CyclicPruningHarness


def main(argv=None):
    config_name, overrides = parse_cli(argv if argv is not None
                                       else sys.argv[1:])
    cfg = compose(config_name, overrides)
    return run(cfg, CyclicPruningHarness)


if __name__ == "__main__":
    main()
