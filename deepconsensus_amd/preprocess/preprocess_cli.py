"""`deepconsensus preprocess`: parallel tf.Example generation.

Behavioral parity with reference preprocess.py:60-361: multiprocessing
workers per ZMW plus a dedicated TFRecord-writer process consuming a queue,
gzip TFRecords per split (@split wildcard in --output), and a
summary.{training,inference}.json with counters + config + version that
training later requires for step counts (model_utils.py:182-205).
"""
from __future__ import annotations

import argparse
import json
import logging
import multiprocessing
import os
import time
from typing import Dict, List, Optional


import deepconsensus_amd
from deepconsensus_amd.dcio import tfrecord
from deepconsensus_amd.preprocess import feeder as pre_feeder
from deepconsensus_amd.preprocess.windows import DcConfig

log = logging.getLogger(__name__)


def setup_writers(
    output_fname: str, splits: List[str]
) -> Dict[str, tfrecord.TFRecordWriter]:
    writers = {}
    for split in splits:
        split_fname = output_fname.replace("@split", split)
        d = os.path.dirname(split_fname)
        if d:
            os.makedirs(d, exist_ok=True)
        writers[split] = tfrecord.TFRecordWriter(
            split_fname, compression="gzip"
        )
    return writers


def process_subreads(subreads, ccs_seqname, dc_config, split,
                     window_widths, queue=None, local=False):
    """Subread processing worker (preprocess.py:200-223).

    ``subreads`` may be a deferred ZmwJob/RawZmwJob (inference mode):
    BAM record decode + expansion then run here, in the worker."""
    expand_counter = None
    if isinstance(subreads, pre_feeder.ZmwJob):
        import collections

        expand_counter = collections.Counter()
        subreads = subreads.materialize(expand_counter)
    tf_out = []
    dc_example = pre_feeder.subreads_to_dc_example(
        subreads, ccs_seqname, dc_config, window_widths
    )
    for example in dc_example.iter_examples():
        tf_out.append(example.tf_example())
    if expand_counter:
        dc_example.counter.update(expand_counter)
    dc_example.counter[f"n_examples_{split}"] += len(tf_out)
    dc_example.counter["n_examples"] += len(tf_out)
    if local:
        return tf_out, split, dc_example.counter
    queue.put([tf_out, split])
    return dc_example.counter


def tf_record_writer_proc(output_fname: str, splits: List[str], queue):
    """Dedicated writer process (preprocess.py:184-196)."""
    writers = setup_writers(output_fname, splits)
    while True:
        tf_example_set, split = queue.get()
        if split == "kill":
            break
        for rec in tf_example_set:
            writers[split].write(rec)
    for w in writers.values():
        w.close()
    return True


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus preprocess")
    ap.add_argument("--subreads_to_ccs", required=True)
    ap.add_argument("--ccs_fasta", default=None,
                    help="deprecated; use --ccs_bam")
    ap.add_argument("--ccs_bam", required=True)
    ap.add_argument("--output", required=True,
                    help="must end in .tfrecord.gz; use @split when training")
    ap.add_argument("--truth_to_ccs")
    ap.add_argument("--truth_bed")
    ap.add_argument("--truth_split")
    ap.add_argument("--cpus", "-j", type=int,
                    default=multiprocessing.cpu_count())
    ap.add_argument("--bam_reader_threads", type=int, default=8)
    ap.add_argument("--limit", type=int, default=0)
    ap.add_argument("--ins_trim", type=int, default=5)
    ap.add_argument("--use_ccs_smart_windows", action="store_true")
    ap.add_argument("--use_ccs_bq", action="store_true")
    ap.add_argument("--max_passes", type=int, default=20)
    ap.add_argument("--max_length", type=int, default=100)
    args = ap.parse_args(argv)
    if args.ccs_fasta:
        # Parity with the reference's deprecation (preprocess.py:247).
        raise NotImplementedError(
            "The --ccs_fasta flag has been deprecated. "
            "Please use --ccs_bam instead."
        )

    if args.cpus == 1:
        raise ValueError("Must set cpus to 0 or >=2 for parallel processing.")
    if not args.output.endswith(".tfrecord.gz"):
        raise ValueError("--output must end with .tfrecord.gz")

    is_training = args.truth_to_ccs and args.truth_bed and args.truth_split
    if is_training:
        log.info("Generating tf.Examples in training mode.")
        contig_split = pre_feeder.read_truth_split(args.truth_split)
        splits = sorted(set(contig_split.values()))
        if "@split" not in args.output:
            raise ValueError("You must add @split to --output when training.")
    elif args.truth_to_ccs or args.truth_bed or args.truth_split:
        raise ValueError(
            "You must specify truth_to_ccs, truth_bed, and truth_split "
            "to generate a training dataset."
        )
    else:
        log.info("Generating tf.Examples in inference mode.")
        splits = ["inference"]

    dc_config = DcConfig(
        max_passes=args.max_passes,
        max_length=args.max_length,
        use_ccs_bq=args.use_ccs_bq,
    )
    proc_feeder, main_counter = pre_feeder.create_proc_feeder(
        subreads_to_ccs=args.subreads_to_ccs,
        ccs_bam=args.ccs_bam,
        dc_config=dc_config,
        ins_trim=args.ins_trim,
        use_ccs_smart_windows=args.use_ccs_smart_windows,
        truth_bed=args.truth_bed,
        truth_to_ccs=args.truth_to_ccs,
        truth_split=args.truth_split,
        limit=args.limit,
        bam_reader_threads=args.bam_reader_threads,
        # Inference mode defers record decode + expansion to the workers
        # (raw feeder). Training mode keeps feeder-side expansion: label
        # fetch/filtering needs the expanded reads and the reference
        # counts expansion stats for ZMWs it later drops.
        defer_expansion=not is_training,
    )

    if args.cpus == 0:
        log.info("Using a single cpu.")
        writers = setup_writers(args.output, splits)
        for job in proc_feeder():
            tf_set, split, counter = process_subreads(*job, local=True)
            for rec in tf_set:
                writers[split].write(rec)
            main_counter.update(counter)
            if main_counter["n_zmw_pass"] % 20 == 0:
                log.info("Processed %s ZMWs.", main_counter["n_zmw_pass"])
        for w in writers.values():
            w.close()
    else:
        log.info("Processing in parallel using %s cores", args.cpus)
        manager = multiprocessing.Manager()
        queue = manager.Queue()
        with multiprocessing.Pool(args.cpus) as mp_pool:
            writer_task = mp_pool.apply_async(
                tf_record_writer_proc, (args.output, splits, queue)
            )
            tasks = []
            for job in proc_feeder():
                tasks.append(
                    mp_pool.starmap_async(
                        process_subreads, ([*job, queue],)
                    )
                )
                if main_counter["n_zmw_pass"] % 20 == 0:
                    tasks = _clear_tasks(tasks, main_counter)
            while tasks:
                time.sleep(0.2)
                tasks = _clear_tasks(tasks, main_counter)
            queue.put(["", "kill"])
            writer_task.get()
            manager.shutdown()
            mp_pool.close()
            mp_pool.join()

    log.info("Completed processing %s ZMWs.", main_counter["n_zmw_pass"])
    summary_name = "training" if is_training else "inference"
    dataset_summary = args.output.replace(
        ".tfrecord.gz", f".{summary_name}.json"
    ).replace("@split", "summary")
    d = os.path.dirname(dataset_summary)
    if d:
        os.makedirs(d, exist_ok=True)
    summary = dict(main_counter.items())
    summary.update(dc_config.to_dict())
    for flag in ["subreads_to_ccs", "ccs_bam", "truth_to_ccs", "truth_bed",
                 "truth_split", "max_passes", "max_length", "ins_trim"]:
        summary[flag] = str(getattr(args, flag))
    summary["version"] = deepconsensus_amd.__version__
    with open(dataset_summary, "w") as f:
        json.dump(summary, f, indent=True)
    log.info("Wrote %s.", dataset_summary)


def _clear_tasks(tasks, main_counter):
    for task in list(tasks):
        if task.ready():
            if task.successful():
                counter = task.get()[0]
                main_counter.update(counter)
                tasks.remove(task)
            else:
                raise Exception("A worker process failed.")
    log.info("Processed %s ZMWs.", main_counter["n_zmw_pass"])
    return tasks


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
