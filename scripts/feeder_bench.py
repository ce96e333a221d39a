#!/usr/bin/env python3
"""Serial-feeder microbenchmark: ZMW decode ceiling of the BAM feeder.

Measures only the serial ZMW streaming path (BamReader decode + subread
grouping + ccs join), which bounds whole-node inference throughput
(ROADMAP: host pipeline item 1). No model, no workers.
"""
import argparse
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scripts.pipeline_bench import make_bams  # noqa: E402
from deepconsensus_amd.preprocess import feeder as pre_feeder  # noqa: E402
from deepconsensus_amd.preprocess.windows import DcConfig  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--zmws", type=int, default=200)
    ap.add_argument("--length", type=int, default=10000)
    ap.add_argument("--subreads", type=int, default=8)
    ap.add_argument("--raw", action="store_true",
                    help="use the raw-record deferred-decode feeder")
    args = ap.parse_args()

    with tempfile.TemporaryDirectory() as td:
        t0 = time.perf_counter()
        sub, ccs = make_bams(td, args.zmws, args.length, args.subreads, 3)
        gen_s = time.perf_counter() - t0
        sz = os.path.getsize(sub) / 1e6
        dc_config = DcConfig(20, 100, False)
        t0 = time.perf_counter()
        proc_feeder, counter = pre_feeder.create_proc_feeder(
            subreads_to_ccs=sub, ccs_bam=ccs, dc_config=dc_config,
            defer_expansion=True, raw_records=args.raw,
        )
        n = 0
        for job in proc_feeder():
            n += 1
        dt = time.perf_counter() - t0
        print(f"gen {gen_s:.1f}s, subreads bam {sz:.1f} MB")
        print(f"feeder ({'raw' if args.raw else 'decoded'}): "
              f"{n} ZMWs in {dt:.2f}s = {n / dt:.1f} ZMW/s "
              f"({sz / dt:.1f} MB/s compressed)")


if __name__ == "__main__":
    main()
