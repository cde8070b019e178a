#!/usr/bin/env python3
"""ROUGE CLI — drop-in for `perl ROUGE.pl NSIZE {N|L} REF SYS`
(ROUGE.pl:5-10); same stdout block."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nats_amd.decode.rouge import format_report

if __name__ == "__main__":
    if len(sys.argv) != 5:
        sys.exit("usage: rouge.py NSIZE {N|L} REFERENCE SYSTEM")
    nsize = int(sys.argv[1])
    metric = sys.argv[2]
    print(format_report(sys.argv[3], sys.argv[4], nsize, metric))
