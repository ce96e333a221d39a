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
    ap.add_argument("--shards", type=int, default=0,
                    help="also time every shard of an N-way split, with "
                    "and without index sidecars")
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

        if args.shards:
            from deepconsensus_amd.dcio import bam as bam_lib

            def time_shard(i, n_shards):
                t0 = time.perf_counter()
                pf, _ = pre_feeder.create_proc_feeder(
                    subreads_to_ccs=sub, ccs_bam=ccs, dc_config=dc_config,
                    defer_expansion=True, shard_index=i,
                    shard_count=n_shards,
                )
                k = sum(1 for _ in pf())
                return k, time.perf_counter() - t0

            per = [time_shard(i, args.shards) for i in range(args.shards)]
            tot = sum(t for _, t in per)
            print(f"modulo shards x{args.shards}: "
                  f"{sum(k for k, _ in per)} ZMWs, sum {tot:.2f}s "
                  f"(max {max(t for _, t in per):.2f}s)")
            t0 = time.perf_counter()
            bam_lib.build_zmw_index(sub)
            bam_lib.build_zmw_index(ccs)
            idx_s = time.perf_counter() - t0
            per = [time_shard(i, args.shards) for i in range(args.shards)]
            tot = sum(t for _, t in per)
            print(f"index build {idx_s:.2f}s; byte-range shards "
                  f"x{args.shards}: {sum(k for k, _ in per)} ZMWs, "
                  f"sum {tot:.2f}s (max {max(t for _, t in per):.2f}s)")


if __name__ == "__main__":
    main()
