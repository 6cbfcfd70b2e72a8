#!/usr/bin/env python3
"""Render sweep JSONL output (benchmarks/sweep.py --out) as the BASELINE.md
results table: latency + algorithmic/bus bandwidth per message size,
mlsl_amd vs torch.distributed when both were measured."""
import argparse
import json
from collections import defaultdict


def human(nbytes):
    for unit, div in (("GiB", 1 << 30), ("MiB", 1 << 20), ("KiB", 1 << 10)):
        if nbytes >= div:
            v = nbytes / div
            return f"{v:.0f}{unit}" if v == int(v) else f"{v:.1f}{unit}"
    return f"{nbytes}B"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("jsonl", nargs="+")
    args = ap.parse_args()

    rows = defaultdict(dict)   # bytes -> impl -> rec
    world = None
    for path in args.jsonl:
        with open(path) as f:
            for line in f:
                r = json.loads(line)
                rows[r["bytes"]][r["impl"]] = r
                world = r["world"]

    impls = sorted({i for by in rows.values() for i in by})
    hdr = "| size |"
    sep = "|---|"
    for i in impls:
        hdr += f" {i} lat_us | {i} algbw GB/s | {i} busbw GB/s |"
        sep += "---|---|---|"
    print(f"AllReduce fp32, {world} rank(s)\n")
    print(hdr)
    print(sep)
    for b in sorted(rows):
        line = f"| {human(b)} |"
        for i in impls:
            r = rows[b].get(i)
            if r:
                line += f" {r['lat_us']} | {r['algbw_GBps']} | {r['busbw_GBps']} |"
            else:
                line += " - | - | - |"
        print(line)


if __name__ == "__main__":
    main()
