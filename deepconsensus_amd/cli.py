"""DeepConsensus-AMD command-line dispatcher.

Parity with reference cli.py:50-122: two-stage dispatch with lazy imports,
--version, and the subcommands {run, preprocess, calibrate, filter_reads}
plus the training entry points the reference exposes as python -m modules
(train, distill, eval).
"""
from __future__ import annotations

import argparse
import logging
import sys
from typing import List, Optional

import deepconsensus_amd

COMMANDS = ["run", "preprocess", "calibrate", "filter_reads", "train",
            "distill", "eval", "export", "index"]


def _index_main(argv: List[str]) -> None:
    from deepconsensus_amd.dcio import bam as bam_lib

    ap = argparse.ArgumentParser(
        "deepconsensus index",
        description="Build ZMW byte-range index sidecars (<bam>"
        f"{bam_lib.ZMW_INDEX_SUFFIX}) so sharded `run --shard i/N` "
        "jobs seek directly to their ZMW range instead of streaming "
        "the whole BAM per shard.",
    )
    ap.add_argument("bams", nargs="+", help="ZMW-sorted BAM file(s)")
    args = ap.parse_args(argv)
    for path in args.bams:
        out = bam_lib.build_zmw_index(path)
        idx = bam_lib.load_zmw_index(path)
        print(f"{out}: {len(idx['zmw'])} ZMW groups"
              + ("" if idx["sorted_flag"][0] else " (NOT zm-sorted; "
                 "sharded runs will fall back to full streaming)"))


def _run_main(argv: List[str]) -> None:
    from deepconsensus_amd.inference import quick_inference as qi

    ap = argparse.ArgumentParser("deepconsensus run")
    ap.add_argument("--subreads_to_ccs", required=True)
    ap.add_argument("--ccs_bam", required=True)
    ap.add_argument("--checkpoint", required=True,
                    help="checkpoint directory (params.json + weights), "
                    "or 'random' for a random-init model (testing)")
    ap.add_argument("--output", required=True,
                    help=".fastq or .bam output path")
    ap.add_argument("--batch_size", type=int, default=1024)
    ap.add_argument("--batch_zmws", type=int, default=100)
    ap.add_argument("--prefetch_batches", type=int, default=3,
                    help="ZMW batches preprocessed ahead of the model "
                    "loop (hides device startup and model jitter)")
    ap.add_argument("--max_length", type=int, default=100)
    ap.add_argument("--min_quality", type=int, default=20)
    ap.add_argument("--min_length", type=int, default=0)
    ap.add_argument("--cpus", type=int, default=0)
    ap.add_argument("--skip_windows_above", type=int, default=45)
    ap.add_argument("--ins_trim", type=int, default=5)
    ap.add_argument("--limit", type=int, default=0)
    ap.add_argument("--use_ccs_smart_windows", action="store_true")
    ap.add_argument("--dc_calibration", default=None,
                    help='"threshold,w,b" or "skip"; default from params.json')
    ap.add_argument("--ccs_calibration", default=None)
    ap.add_argument("--device", default=None)
    ap.add_argument("--shard", default=None,
                    help="'i/N': process only shard i of N (one process "
                    "per GPU). With `deepconsensus index` sidecars the "
                    "shard seeks straight to its contiguous ZMW byte "
                    "range; otherwise it streams the whole BAM keeping "
                    "every Nth ZMW (like the reference's ccs --chunk)")
    ap.add_argument("--end_after_stage", default="full",
                    choices=[s.name.lower() for s in qi.DebugStage],
                    help="stop after this stage (debug/runtime testing)")
    ap.add_argument("--use_only_gpu_index", type=int, default=None,
                    help="pin inference to this GPU index "
                    "(shorthand for --device cuda:N)")
    ap.add_argument("--ccs_fasta", default=None,
                    help="deprecated; use --ccs_bam")
    args = ap.parse_args(argv)
    if args.ccs_fasta:
        # Parity with the reference's deprecation (quick_inference.py:968).
        raise NotImplementedError(
            "The --ccs_fasta flag has been deprecated. "
            "Please use --ccs_bam instead."
        )

    options = qi.InferenceOptions(
        max_length=args.max_length,
        min_quality=args.min_quality,
        min_length=args.min_length,
        batch_size=args.batch_size,
        batch_zmws=args.batch_zmws,
        prefetch_batches=args.prefetch_batches,
        cpus=args.cpus,
        skip_windows_above=args.skip_windows_above,
        ins_trim=args.ins_trim,
        use_ccs_smart_windows=args.use_ccs_smart_windows,
        end_after_stage=qi.DebugStage[args.end_after_stage.upper()],
    )
    if args.shard:
        i, n = args.shard.split("/")
        options.shard_index, options.shard_count = int(i), int(n)
    device = args.device
    if args.use_only_gpu_index is not None:
        device = f"cuda:{args.use_only_gpu_index}"
    outcome = qi.run(
        subreads_to_ccs=args.subreads_to_ccs,
        ccs_bam=args.ccs_bam,
        checkpoint=args.checkpoint,
        output=args.output,
        options=options,
        limit=args.limit,
        dc_calibration=args.dc_calibration,
        ccs_calibration=args.ccs_calibration,
        device=device,
    )
    if outcome.success == 0 and options.end_after_stage == qi.DebugStage.FULL:
        sys.exit(1)


def main(argv: Optional[List[str]] = None) -> None:
    argv = list(sys.argv[1:] if argv is None else argv)
    logging.basicConfig(
        level=logging.INFO,
        format="%(levelname)s %(name)s: %(message)s",
    )
    if argv and argv[0] in ("--version", "-v"):
        print(f"deepconsensus-amd {deepconsensus_amd.__version__}")
        return
    if not argv or argv[0] in ("-h", "--help"):
        print(
            "usage: deepconsensus {" + ",".join(COMMANDS) + "} [args]\n\n"
            "DeepConsensus-AMD: MI355X-native consensus calling.\n"
            "  run           polish CCS reads (BAM -> FASTQ/BAM)\n"
            "  preprocess    generate tf.Example TFRecords\n"
            "  calibrate     empirical base-quality calibration stats\n"
            "  filter_reads  filter FASTQ/BAM by average read quality\n"
            "  train         train a model (see also: distill, eval)\n"
            "  index         build ZMW byte-range sidecars for sharded runs\n"
        )
        return
    command, rest = argv[0], argv[1:]
    if command == "run":
        _run_main(rest)
    elif command == "preprocess":
        from deepconsensus_amd.preprocess import preprocess_cli

        preprocess_cli.main(rest)
    elif command == "calibrate":
        from deepconsensus_amd.calibration import calculate_baseq_calibration

        calculate_baseq_calibration.main(rest)
    elif command == "filter_reads":
        from deepconsensus_amd.calibration import filter_reads

        filter_reads.main(rest)
    elif command == "train":
        from deepconsensus_amd.models import train

        train.main(rest)
    elif command == "distill":
        from deepconsensus_amd.models import distill

        distill.main(rest)
    elif command == "eval":
        from deepconsensus_amd.models import infer_eval

        infer_eval.main(rest)
    elif command == "export":
        from deepconsensus_amd.models import export_model

        export_model.main(rest)
    elif command == "index":
        _index_main(rest)
    else:
        print(f"unknown command {command!r}; one of: {', '.join(COMMANDS)}")
        sys.exit(2)


if __name__ == "__main__":
    main()
