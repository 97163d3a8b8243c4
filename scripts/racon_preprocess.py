#!/usr/bin/env python3
"""Uniquify Illumina paired-end FASTQ headers for racon.

racon deduplicates reads by name (polisher dedup pass), which collapses
paired-end mates sharing a header. This script appends '1' to the first
occurrence of each header and '2' to a repeat, so both mates survive.
Capability parity with /root/reference/scripts/racon_preprocess.py
(new implementation; also accepts gzipped input and FASTA).

Usage: racon_preprocess.py first.fastq [second.fastq] > combined.fastq
"""

import argparse
import gzip
import sys


def opener(path):
    return gzip.open(path, "rt") if path.endswith(".gz") else open(path)


def emit_fastq(name, data, qual, seen):
    if not name or not data or len(data) != len(qual):
        sys.exit("[racon_preprocess] error: file is not in FASTQ format")
    suffix = "2" if name in seen else "1"
    seen.add(name)
    sys.stdout.write(f"{name}{suffix}\n{data}\n+\n{qual}\n")


def process(path, seen):
    with opener(path) as f:
        first = f.read(1)
        f.seek(0)
        if first == ">":  # FASTA: header uniquify only
            name = None
            for line in f:
                line = line.rstrip()
                if line.startswith(">"):
                    name = line.split(" ")[0]
                    suffix = "2" if name in seen else "1"
                    seen.add(name)
                    sys.stdout.write(f"{name}{suffix}\n")
                else:
                    sys.stdout.write(line + "\n")
            return
        name = data = qual = None
        state = 0  # 0: expect header, 1: reading data, 2: reading quality
        for line in f:
            line = line.rstrip()
            if state == 0:
                name, data, qual = line.split(" ")[0], "", ""
                state = 1
            elif state == 1:
                if line.startswith("+"):
                    state = 2
                else:
                    data += line
            else:
                qual += line
                if len(qual) >= len(data):
                    emit_fastq(name, data, qual, seen)
                    state = 0
        if state != 0:
            emit_fastq(name, data, qual, seen)


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("first", help="first (or interleaved) FASTQ file")
    ap.add_argument("second", nargs="?", help="optional second FASTQ file of the pair")
    args = ap.parse_args()
    seen = set()
    process(args.first, seen)
    if args.second:
        process(args.second, seen)


if __name__ == "__main__":
    main()
