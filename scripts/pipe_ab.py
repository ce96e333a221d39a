#!/usr/bin/env python3
"""Within-lease A/B of the pipeline stitch modes (one box, one corpus)."""
import os, sys, tempfile, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepconsensus_amd.dcio import bam as bam_lib
from deepconsensus_amd.utils.synth import make_synth_bams
from deepconsensus_amd.inference import quick_inference as qi

ZMWS = int(sys.argv[1]) if len(sys.argv) > 1 else 800
with tempfile.TemporaryDirectory() as td:
    sub, ccs, _ = make_synth_bams(td, ZMWS, 15000, 8, 3)
    bam_lib.build_zmw_index(sub); bam_lib.build_zmw_index(ccs)
    results = {}
    order = ["pool", "serial", "pool", "serial"]  # interleaved rounds
    for i, mode in enumerate(order):
        os.environ["DC_STITCH_MODE"] = mode
        torch.manual_seed(1234)
        options = qi.InferenceOptions(batch_size=16384, batch_zmws=100,
                                      cpus=min(os.cpu_count() or 4, 16),
                                      min_quality=0, skip_windows_above=0)
        t0 = time.perf_counter()
        c = qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
                   output=os.path.join(td, f"o{i}.fastq"), options=options)
        dt = time.perf_counter() - t0
        results.setdefault(mode, []).append(dt)
        print(f"round {i} mode={mode}: {dt:.2f}s ({ZMWS/dt:.0f} ZMW/s), "
              f"success={c.success}", flush=True)
    for mode, ts in results.items():
        best = min(ts)
        print(f"{mode}: best {best:.2f}s = {ZMWS/best:.0f} ZMW/s")
