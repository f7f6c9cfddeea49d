#!/usr/bin/env python3
"""template.py — torchrun-compatible entry point (reference template.py parity).

Run (same launch shape as the reference README.md:353):

    torchrun --nproc_per_node=4 --master-addr 127.0.0.1 template.py \
        --num_bases 50 --increment 10 --data_set cifar100 --data_path <dir>

or single-process: python template.py --data_set synthetic --num_epochs 2
"""

import argparse

from cilfw.config import get_args_parser
from cilfw.engine import run

if __name__ == "__main__":
    parser = argparse.ArgumentParser("cilfw WA class-incremental training",
                                     parents=[get_args_parser()])
    args = parser.parse_args()
    run(args)
