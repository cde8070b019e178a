#!/usr/bin/env python3
"""Dictionary builder CLI — parity with data/build_dictionary.py:9-38
(one .pkl per input corpus, eos=0/UNK=1/freq-desc from 2)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nats_amd.data.dictionary import build_dictionary

if __name__ == "__main__":
    for filename in sys.argv[1:]:
        print("Processing", filename)
        build_dictionary(filename)
        print("Done")
