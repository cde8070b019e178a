#!/usr/bin/env python3
"""UNK replacement CLI — parity with scripts/replace_unk.py:4-10."""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nats_amd.decode.replace_unk import replace_unk_files

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("input", type=str)
    parser.add_argument("origin", type=str)
    parser.add_argument("new", type=str)
    args = parser.parse_args()
    replace_unk_files(args.input, args.origin, args.new)
