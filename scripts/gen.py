#!/usr/bin/env python3
"""Summarize a source file — entrypoint parity with scripts/gen.py:138-156
(same flags: -k beam, -p processes, -l/-x/-s distraction lambdas,
-n length-normalize, -c char-level)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nats_amd.decode.driver import generate_file

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("-k", type=int, default=5)
    parser.add_argument("-p", type=int, default=5)
    parser.add_argument("-l", type=float, default=0)
    parser.add_argument("-x", type=float, default=0)
    parser.add_argument("-s", type=float, default=0)
    parser.add_argument("-n", action="store_true", default=False)
    parser.add_argument("-c", action="store_true", default=False)
    parser.add_argument("--maxlen", type=int, default=100,
                        help="summary length cap (framework extension; "
                             "the reference hardcodes 100, gen.py:33)")
    parser.add_argument("model", type=str)
    parser.add_argument("dictionary", type=str)
    parser.add_argument("source", type=str)
    parser.add_argument("saveto", type=str)
    args = parser.parse_args()

    generate_file(args.model, args.dictionary, args.source, args.saveto,
                  k=args.k, normalize=args.n, n_process=args.p,
                  chr_level=args.c, kl_factor=args.l, ctx_factor=args.x,
                  state_factor=args.s, maxlen=args.maxlen)
