#!/usr/bin/env python3
"""Single-process word count over stdin — the correctness oracle
(misc/naive.lua:1-7 parity).  Usage: cat files... | python misc/naive.py"""

import collections
import sys

vocab = collections.Counter()
for line in sys.stdin.buffer:
    vocab.update(line.split())
for w, v in vocab.items():
    sys.stdout.write(f"{v}\t{w.decode('utf-8', 'surrogateescape')}\n")
