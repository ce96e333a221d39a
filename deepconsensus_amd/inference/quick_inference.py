"""`deepconsensus run`: BAM -> polished FASTQ/BAM inference orchestrator.

Behavioral parity with reference quick_inference.py:238-984 — ZMW streaming,
multiprocess window preprocessing, skip-window fast path (overflow windows
and windows whose average CCS base quality exceeds --skip_windows_above adopt
the CCS sequence + calibrated CCS qualities), batched model execution,
stitching, FASTQ/BAM output with ec/np/rq/RG/zm tags, a per-stage runtime CSV
and an inference-stats JSON.

MI355X-native differences: the model path is the InferenceRunner HIP pipeline
(fused embed gather -> bf16 encoder with MFMA banded attention -> fused
LN+head+QV kernel), windows cross to the device as int16 tensors, and
preprocessing of the NEXT ZMW batch overlaps model execution of the current
one (the reference serialized these stages).
"""
from __future__ import annotations

import collections
import concurrent.futures
import dataclasses
import enum
import itertools
import json
import logging
import os
import time
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

from deepconsensus_amd.calibration import calibration as calibration_lib
from deepconsensus_amd.models import checkpoint as ckpt_lib
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.models.runner import InferenceRunner
from deepconsensus_amd.postprocess import stitch as stitch_utils
from deepconsensus_amd.preprocess import feeder as pre_feeder
from deepconsensus_amd.preprocess.windows import DcConfig
from deepconsensus_amd.utils import constants, phred

log = logging.getLogger(__name__)


@enum.unique
class DebugStage(enum.Enum):
    """Stage to end after, for debugging and runtime testing
    (quick_inference.py:68-75)."""

    DC_INPUT = 1
    TF_EXAMPLES = 2
    RUN_MODEL = 3
    FULL = 4


@dataclasses.dataclass
class InferenceOptions:
    """Options shared across inference stages (quick_inference.py:238-275)."""

    max_length: int = 100
    example_height: int = 85
    max_passes: int = 20
    min_quality: int = 20
    min_length: int = 0
    batch_size: int = 1024
    use_ccs_bq: bool = False
    cpus: int = 0
    skip_windows_above: int = 45
    max_base_quality: int = constants.MAX_QUAL
    dc_calibration_values: calibration_lib.QualityCalibrationValues = (
        dataclasses.field(
            default_factory=lambda: calibration_lib.parse_calibration_string(
                "skip"
            )
        )
    )
    ccs_calibration_values: calibration_lib.QualityCalibrationValues = (
        dataclasses.field(
            default_factory=lambda: calibration_lib.parse_calibration_string(
                "skip"
            )
        )
    )
    batch_zmws: int = 100
    # ZMW batches preprocessed ahead of the model loop. >1 hides device/
    # runner startup (~20 s HIP context + extension init on a cold box)
    # and model-time jitter behind the worker pool; memory cost is
    # ~batch_zmws x windows x height x width x 2 bytes per slot.
    prefetch_batches: int = 3
    ins_trim: int = 0
    use_ccs_smart_windows: bool = False
    # ZMW sharding for multi-process / multi-GPU runs ("i/N", like the
    # reference's `ccs --chunk` sharding, docs/quick_start.md:216-248):
    # this process handles ZMWs with index % N == i.
    shard_index: int = 0
    shard_count: int = 1
    # Stop the pipeline early for debugging/runtime testing
    # (quick_inference.py:501,559,715).
    end_after_stage: DebugStage = DebugStage.FULL


@dataclasses.dataclass
class ZmwWindows:
    """Worker output for one ZMW: model windows pre-stacked, skipped
    windows pre-resolved to DCModelOutput.

    The reference crossed the pool boundary with one feature dict per
    window and ran skip triage + CCS passthrough serially
    (quick_inference.py:655-677). Here the worker does the triage
    (avg-Phred + overflow), builds the CCS passthrough outputs, and
    stacks the model-bound windows into ONE int16 [n, R, L] array — the
    serial batch loop only concatenates arrays and runs the model.
    """

    name: str
    window_pos: np.ndarray  # [n_model] int64, model-bound windows only
    rows: Optional[np.ndarray]  # [n_model, R, L] int16, format_rows'd
    skipped: List["stitch_utils.DCModelOutput"]
    counter: collections.Counter
    ec: Optional[float] = None
    np_num_passes: Optional[int] = None
    rq: Optional[float] = None
    rg: Optional[str] = None

    @property
    def n_examples(self) -> int:
        n = len(self.skipped)
        if self.rows is not None:
            n += len(self.rows)
        return n


def preprocess_one_zmw(one_zmw) -> ZmwWindows:
    """Windows + counters for one ZMW (quick_inference.py:535-564).

    ``subreads`` may be a deferred ZmwJob/RawZmwJob (raw BAM records):
    record decode and expansion then run HERE — i.e. in the worker
    pool — instead of the serial feeder.
    """
    zmw, subreads, dc_config, window_widths, options = one_zmw
    stage = options.end_after_stage
    expand_counter = None
    if isinstance(subreads, pre_feeder.ZmwJob):
        expand_counter = collections.Counter()
        subreads = subreads.materialize(expand_counter)
    dc_whole = pre_feeder.subreads_to_dc_example(
        subreads=subreads,
        ccs_seqname=zmw,
        dc_config=dc_config,
        window_widths=window_widths,
    )
    empty = ZmwWindows(name=zmw, window_pos=np.empty(0, np.int64),
                       rows=None, skipped=[], counter=dc_whole.counter)
    if stage == DebugStage.DC_INPUT:
        if expand_counter:
            empty.counter.update(expand_counter)
        return empty
    # Worker-side finishing, keeping the serial batch loop thin: the
    # fast path applies format_rows' PW/IP/SN clipping at ZMW level and
    # emits int16 matrices directly (integral-after-truncation for the
    # model — the embedding casts to int, SN fractions truncate
    # identically — and int16 halves the pickle + H2D volume).
    model_rows: List[np.ndarray] = []
    window_pos: List[int] = []
    skipped: List[stitch_utils.DCModelOutput] = []
    for f in dc_whole.iter_feature_dicts(
        pw_max=255, ip_max=255, sn_max=500,  # config.py:89-92 defaults
        out_dtype=np.int16,
    ):
        skip = bool(f["overflow"])
        if not skip and options.skip_windows_above:
            avg_q = phred.avg_phred(f["ccs_base_quality_scores"])
            skip = avg_q > options.skip_windows_above
        if skip:
            skipped.append(process_skipped_window(f, options))
            continue
        model_rows.append(f["subreads"][:, :, 0])
        window_pos.append(int(f["window_pos"]))
    counter = dc_whole.counter
    if expand_counter:
        counter.update(expand_counter)
    ccs = dc_whole.ccs
    return ZmwWindows(
        name=zmw,
        window_pos=np.asarray(window_pos, np.int64),
        rows=np.stack(model_rows) if model_rows else None,
        skipped=skipped,
        counter=counter,
        ec=ccs.ec, np_num_passes=ccs.np_num_passes, rq=ccs.rq, rg=ccs.rg,
    )


def process_skipped_window(
    feature_dict: Dict[str, Any], options: InferenceOptions
) -> stitch_utils.DCModelOutput:
    """CCS passthrough for skipped windows (quick_inference.py:567-594)."""
    rows = feature_dict["subreads"]
    (_, _, _, _, ccs_index, _, _) = cfg.get_indices(
        options.max_passes, options.use_ccs_bq
    )
    ccs = rows[ccs_index[0], :, 0]
    ccs_seq = phred.encoded_sequence_to_string(ccs)
    ccs_quality_scores = np.asarray(
        feature_dict["ccs_base_quality_scores"], dtype=np.float64
    )
    if options.ccs_calibration_values.enabled:
        ccs_quality_scores = calibration_lib.calibrate_quality_scores(
            ccs_quality_scores, options.ccs_calibration_values
        )
    ccs_quality_scores = np.minimum(
        ccs_quality_scores, options.max_base_quality
    )
    ccs_quality_scores = np.maximum(ccs_quality_scores, 0)
    ccs_quality_scores = ccs_quality_scores.astype(np.int32)
    return stitch_utils.DCModelOutput(
        window_pos=feature_dict["window_pos"],
        molecule_name=feature_dict["name"],
        sequence=ccs_seq,
        quality_string=phred.quality_scores_to_string(ccs_quality_scores),
        ec=feature_dict["ec"],
        np_num_passes=feature_dict["np_num_passes"],
        rq=feature_dict["rq"],
        rg=feature_dict["rg"],
    )


_SEQ_LUT = np.frombuffer(constants.SEQ_VOCAB.encode("ascii"), np.uint8)


def run_model_on_zmws(
    zmws: List[ZmwWindows],
    runner: InferenceRunner,
    options: InferenceOptions,
) -> List[stitch_utils.DCModelOutput]:
    """Batched model execution emitting per-window DCModelOutput.

    Model-bound windows arrive pre-stacked per ZMW; this concatenates
    them across the ZMW batch, runs the device model in batch_size
    chunks, and converts bases/QVs to strings with batch-level numpy
    (one LUT pass + one +33 pass per chunk).
    """
    predictions: List[stitch_utils.DCModelOutput] = []
    # Per-window owner metadata, in concatenation order.
    owners: List[ZmwWindows] = []
    pos_list: List[np.ndarray] = []
    rows_list: List[np.ndarray] = []
    for z in zmws:
        predictions.extend(z.skipped)
        if z.rows is not None and len(z.rows):
            rows_list.append(z.rows)
            pos_list.append(z.window_pos)
            owners.extend([z] * len(z.rows))
    if not rows_list:
        return predictions
    all_rows = np.concatenate(rows_list)
    all_pos = np.concatenate(pos_list)
    # Native path: features are non-negative and <= SN_MAX (500) after
    # clipping, and the embed kernel casts to int anyway, so int16 staging
    # halves H2D traffic with bit-identical results (numpy's truncation
    # toward zero == the kernel's (int)v == the torch model's .long()).
    use_i16 = bool(getattr(runner, "native", False))
    if not use_i16:
        all_rows = all_rows.astype(np.float32)
    pinned: Dict[Tuple[int, ...], torch.Tensor] = {}
    use_graph = use_i16 and os.environ.get("DC_SERVE_GRAPH", "1") != "0"
    for i in range(0, len(all_rows), options.batch_size):
        chunk = np.ascontiguousarray(all_rows[i : i + options.batch_size])
        n_valid = len(chunk)
        rows_t = torch.from_numpy(chunk)
        if use_i16:
            # Fixed-shape pinned buffer: partial tails are zero-padded
            # so the hipGraph-captured step replays at one shape (pad
            # windows cost compute on the final chunk only; their
            # outputs are sliced away).
            shape = (
                (options.batch_size,) + rows_t.shape[1:]
                if use_graph
                else tuple(rows_t.shape)
            )
            buf = pinned.get(shape)
            if buf is None:
                buf = torch.zeros(shape, dtype=rows_t.dtype).pin_memory()
                pinned[shape] = buf
            buf[:n_valid].copy_(rows_t)
            if use_graph and n_valid < shape[0]:
                buf[n_valid:].zero_()
            rows_t = buf
        if use_graph:
            bases_t, quals_t = runner.forward_windows_graphed(rows_t)
            bases_t = bases_t[:n_valid]
            quals_t = quals_t[:n_valid]
        else:
            bases_t, quals_t = runner.forward_windows(rows_t)
        bases_np = bases_t.cpu().numpy()
        # Corruption tripwire: a healthy model rarely predicts majority
        # gap. A batch-size-dependent native-path regression produced
        # silent all-gap output (profiles/r01_perf_journal.md "GAP
        # REGRESSION") — surface it instead of writing garbage FASTQ.
        gap_frac = float((bases_np == 0).mean())
        if gap_frac > 0.6 and len(chunk) > 1 and getattr(
            runner, "native", False
        ):
            # Corruption tripwire turned hard gate: audit a
            # sample of the suspect batch against the fp32 torch path
            # and refuse to write FASTQ from a disagreeing native pass.
            sample = min(32, len(chunk))
            with torch.no_grad():
                probs = runner.model(
                    rows_t[:sample].to(runner.device).float(),
                    training=False,
                )
            torch_bases = probs.argmax(-1).cpu().numpy()
            agree = float((bases_np[:sample] == torch_bases).mean())
            if agree < 0.99:
                raise RuntimeError(
                    f"native inference path disagrees with the torch "
                    f"reference on {100 * (1 - agree):.1f}% of base "
                    f"calls in a {100 * gap_frac:.0f}%-gap batch of "
                    f"{len(chunk)} windows — refusing to write "
                    f"corrupted FASTQ. This indicates a native-kernel "
                    f"regression; re-run with DC_FUSED_FFN=0/"
                    f"smaller --batch_size and file a bug."
                )
            log.info(
                "high gap fraction %.0f%% over %d windows, but the "
                "native path agrees with the torch reference "
                "(%.4f) — input is legitimately gappy",
                100.0 * gap_frac, len(chunk), agree,
            )
        seq_mat = _SEQ_LUT[bases_np.astype(np.int64)]
        qual_mat = (quals_t.cpu().numpy() + 33).astype(np.uint8)
        for j in range(len(chunk)):
            z = owners[i + j]
            predictions.append(
                stitch_utils.DCModelOutput(
                    window_pos=int(all_pos[i + j]),
                    molecule_name=z.name,
                    sequence=seq_mat[j].tobytes().decode("ascii"),
                    quality_string=qual_mat[j].tobytes().decode("ascii"),
                    ec=z.ec,
                    np_num_passes=z.np_num_passes,
                    rq=z.rq,
                    rg=z.rg,
                )
            )
    return predictions


class Timelog:
    """Stage wall-clock CSV (quick_inference.py:278-299,777-783)."""

    def __init__(self):
        self.rows: List[Dict[str, Any]] = []

    def add(self, stage, item, before, num_examples=None, num_subreads=None,
            num_zmws=None):
        self.rows.append(
            dict(
                item=item, stage=stage, runtime=time.time() - before,
                num_zmws=num_zmws, num_examples=num_examples,
                num_subreads=num_subreads,
            )
        )

    def save(self, output_prefix: str):
        cols = ["item", "stage", "runtime", "num_zmws", "num_examples",
                "num_subreads"]
        with open(f"{output_prefix}.csv", "w") as f:
            f.write(",".join(cols) + "\n")
            for r in self.rows:
                f.write(",".join(str(r[c]) for c in cols) + "\n")


def _stitch_batch_to_text(args):
    """Process-pool stitcher: sorts + stitches a batch of predictions
    into FASTQ text, returning (text, outcome_counts). Runs in the
    worker pool so the GIL-heavy stitching never blocks the serial
    model loop or the prefetch threads."""
    predictions, max_length, min_quality, min_length = args
    counter = stitch_utils.OutcomeCounter()
    parts = []
    predictions = sorted(
        predictions, key=lambda dc: (dc.molecule_name, dc.window_pos)
    )
    for zmw, preds in itertools.groupby(
        predictions, lambda p: p.molecule_name
    ):
        fastq_string = stitch_utils.stitch_to_fastq(
            molecule_name=zmw,
            predictions=list(preds),
            max_length=max_length,
            min_quality=min_quality,
            min_length=min_length,
            outcome_counter=counter,
        )
        if fastq_string:
            parts.append(fastq_string)
    return "".join(parts), dataclasses.asdict(counter)


def _write_outputs(
    predictions: List[stitch_utils.DCModelOutput],
    output_writer,
    bam_out,
    options: InferenceOptions,
    outcome_counter: stitch_utils.OutcomeCounter,
):
    """Sorts, stitches and writes per-ZMW outputs
    (quick_inference.py:718-760)."""
    from deepconsensus_amd.dcio import bam as bam_lib

    predictions = sorted(
        predictions, key=lambda dc: (dc.molecule_name, dc.window_pos)
    )
    for zmw, preds in itertools.groupby(
        predictions, lambda p: p.molecule_name
    ):
        preds = list(preds)
        fastq_string = stitch_utils.stitch_to_fastq(
            molecule_name=zmw,
            predictions=preds,
            max_length=options.max_length,
            min_quality=options.min_quality,
            min_length=options.min_length,
            outcome_counter=outcome_counter,
        )
        if not fastq_string:
            continue
        if bam_out is None:
            output_writer.write(fastq_string)
        else:
            name, seq, _, qual = fastq_string.splitlines()
            name = name[1:]
            record = bam_lib.BamRead(
                qname=name,
                flag=4,
                ref_id=-1,
                pos=-1,
                mapq=255,
                seq=seq,
                query_qualities=phred.quality_string_to_array(qual),
                tags={
                    "ec": preds[0].ec if preds[0].ec is not None else -1.0,
                    "np": preds[0].np_num_passes or 0,
                    "rq": preds[0].rq if preds[0].rq is not None else -1.0,
                    "RG": preds[0].rg or "",
                    "zm": int(name.split("/")[1]),
                },
            )
            bam_out.write(record)


def run(
    subreads_to_ccs: str,
    ccs_bam: str,
    checkpoint: str,
    output: str,
    options: Optional[InferenceOptions] = None,
    limit: int = 0,
    dc_calibration: Optional[str] = None,
    ccs_calibration: Optional[str] = None,
    device: Optional[str] = None,
) -> stitch_utils.OutcomeCounter:
    """Performs an inference run (quick_inference.py:794-963)."""
    t_start = time.time()
    options = options or InferenceOptions()

    # Load params + model. checkpoint == 'random' builds a random-init model
    # (testing only); otherwise a directory with params.json + weights.
    if checkpoint == "random":
        params = cfg.get_config("transformer_learn_values+custom")
        params.max_passes = options.max_passes
        cfg.modify_params(
            params, max_length=options.max_length, is_training=False
        )
        model = get_model(params)
    else:
        params = ckpt_lib.load_params(checkpoint)
        cfg.modify_params(
            params, max_length=options.max_length, is_training=False
        )
        model = get_model(params)
        ckpt_lib.load_checkpoint(checkpoint, model)
        options.max_passes = params.max_passes
        options.use_ccs_bq = bool(params.get("use_ccs_bq", False))
    options.example_height = cfg.get_total_rows(
        options.max_passes, options.use_ccs_bq
    )

    # Calibration: explicit flag > params.json dc_calibration > skip.
    if dc_calibration is None:
        dc_calibration = params.get("dc_calibration", "skip")
    options.dc_calibration_values = calibration_lib.parse_calibration_string(
        dc_calibration
    )
    if ccs_calibration is not None:
        options.ccs_calibration_values = (
            calibration_lib.parse_calibration_string(ccs_calibration)
        )

    calib_str = dc_calibration if dc_calibration else "skip"

    # Output writer.
    bam_out = None
    if output.endswith(".bam"):
        from deepconsensus_amd.dcio import bam as bam_lib

        header = bam_lib.BamHeader(
            text="@HD\tVN:1.6\tSO:unknown", references=[]
        )
        bam_out = bam_lib.BamWriter(output, header)
        output_writer = None
    else:
        output_writer = open(output, "w")

    outcome_counter = stitch_utils.OutcomeCounter()
    stats_counter = collections.Counter()
    timelog = Timelog()
    output_prefix = os.path.splitext(output)[0]

    dc_config = DcConfig(
        options.max_passes, options.max_length, options.use_ccs_bq
    )
    proc_feeder, main_counter = pre_feeder.create_proc_feeder(
        subreads_to_ccs=subreads_to_ccs,
        ccs_bam=ccs_bam,
        dc_config=dc_config,
        ins_trim=options.ins_trim,
        use_ccs_smart_windows=options.use_ccs_smart_windows,
        limit=limit,
        defer_expansion=True,
        shard_index=options.shard_index,
        shard_count=options.shard_count,
    )

    pool = None
    stitch_pool = None
    if options.cpus > 0:
        pool = concurrent.futures.ProcessPoolExecutor(options.cpus)
        # Dedicated small stitch pool: submitting stitch work to the
        # preprocess pool queued it AHEAD of later preprocess batches
        # (FIFO) and starved the prefetch pipeline (measured 209 -> 130
        # ZMW/s at 1000 ZMWs).
        stitch_pool = concurrent.futures.ProcessPoolExecutor(
            max(2, options.cpus // 4)
        )

    def zmw_batches():
        # ZMW sharding happens inside the feeder (byte-range seek when a
        # `deepconsensus index` sidecar exists, modulo streaming else).
        batch = []
        for input_data in proc_feeder():
            subreads, zmw, dcc, split, window_widths = input_data
            batch.append((zmw, subreads, dcc, window_widths, options))
            if len(batch) >= options.batch_zmws:
                yield batch
                batch = []
        if batch:
            yield batch

    def preprocess_batch(inputs):
        if pool is not None:
            return list(pool.map(preprocess_one_zmw, inputs))
        return [preprocess_one_zmw(z) for z in inputs]

    def infer_batch(inputs, outputs, batch_name):
        before = time.time()
        for z in outputs:
            stats_counter.update(z.counter)
        n_examples = sum(z.n_examples for z in outputs)
        n_subreads = sum(len(z[1]) for z in inputs)
        timelog.add("preprocess", batch_name, before, n_examples,
                    n_subreads, len(inputs))
        if options.end_after_stage in (DebugStage.DC_INPUT,
                                       DebugStage.TF_EXAMPLES):
            return

        before = time.time()
        n_model = sum(
            len(z.rows) for z in outputs if z.rows is not None
        )
        n_skipped = sum(len(z.skipped) for z in outputs)
        preds = run_model_on_zmws(outputs, runner, options)
        # Per-batch model/skip split (quick_inference.py:688-705).
        total = max(len(preds), 1)
        log.info(
            "Example summary: ran model=%d (%.2f%%) skip=%d (%.2f%%) "
            "total=%d.",
            n_model, 100.0 * n_model / total,
            n_skipped, 100.0 * n_skipped / total, len(preds),
        )
        timelog.add("run_model", batch_name, before, n_examples,
                    n_subreads, len(inputs))
        if options.end_after_stage == DebugStage.RUN_MODEL:
            return

        before = time.time()
        # Within-lease A/B at 800 ZMWs: serial 173 ZMW/s vs pool 162
        # (plus ~2 s stitch-pool spin-up on the first batch) — the
        # overlap does not pay at this scale; serial is the default
        # and DC_STITCH_MODE=pool keeps the experiment reachable.
        stitch_mode = os.environ.get("DC_STITCH_MODE", "serial")
        if (stitch_mode == "pool" and stitch_pool is not None
                and bam_out is None):
            # Stitch in the PROCESS pool (no GIL contention with the
            # prefetch threads), write the returned text on the ordered
            # writer thread. BAM output keeps the in-process path (the
            # writer needs record objects).
            fut = stitch_pool.submit(
                _stitch_batch_to_text,
                (preds, options.max_length, options.min_quality,
                 options.min_length),
            )

            def write_task(fut=fut, batch_name=batch_name,
                           n_examples=n_examples,
                           n_subreads=n_subreads, n_inputs=len(inputs),
                           before=before):
                text, counts = fut.result()
                for k, v in counts.items():
                    setattr(outcome_counter, k,
                            getattr(outcome_counter, k) + v)
                if text:
                    output_writer.write(text)
                timelog.add("stitch_and_write_fastq", batch_name,
                            before, n_examples, n_subreads, n_inputs)

            writer_futs.append(writer_pool.submit(write_task))
        else:
            _write_outputs(preds, output_writer, bam_out, options,
                           outcome_counter)
            timelog.add("stitch_and_write_fastq", batch_name, before,
                        n_examples, n_subreads, len(inputs))

    # Pipelined loop: up to `prefetch_batches` ZMW batches preprocess in
    # the worker pool while the current batch runs the model.
    batch_iter = zmw_batches()
    lookahead = max(1, options.prefetch_batches)
    pending: "collections.deque" = collections.deque()
    n_fed = 0
    n_batches = 0
    writer_pool = concurrent.futures.ThreadPoolExecutor(1)
    writer_futs: List[concurrent.futures.Future] = []
    with concurrent.futures.ThreadPoolExecutor(lookahead) as prefetcher:

        def feed_one() -> bool:
            # Serial ZMW streaming is a real pipeline stage: account it
            # (the reference folded it into its preprocess rows).
            nonlocal n_fed
            before = time.time()
            inputs = next(batch_iter, None)
            timelog.add("feeder", f"batch {n_fed}", before,
                        num_zmws=len(inputs) if inputs else 0)
            if inputs is None:
                return False
            pending.append(
                (inputs, prefetcher.submit(preprocess_batch, inputs))
            )
            n_fed += 1
            return True

        # Prime the pipeline BEFORE device/model setup: the worker pool
        # preprocesses the first batches while InferenceRunner brings up
        # the device, loads the HIP extension and moves weights — on a
        # cold box that startup is ~20 s and fully overlapped.
        while len(pending) < lookahead and feed_one():
            pass
        before = time.time()
        runner = InferenceRunner(
            params, model, device=device, calibration=calib_str,
            max_qual=options.max_base_quality,
        )
        timelog.add("startup_runner", "startup", before)
        log.info("model on %s (native kernels: %s)", runner.device,
                 runner.native)

        while pending:
            prev_inputs, prev_fut = pending.popleft()
            name = f"batch {n_batches}"
            # Time blocked on the overlapped preprocessing (not a
            # reference stage; extends the CSV so wall is accounted).
            before = time.time()
            outputs = prev_fut.result()
            timelog.add("wait_preprocess", name, before, None, None,
                        len(prev_inputs))
            infer_batch(prev_inputs, outputs, name)
            n_batches += 1
            while len(pending) < lookahead and feed_one():
                pass

    writer_pool.shutdown(wait=True)
    for f in writer_futs:
        f.result()  # surface stitch/write errors instead of truncating
    if stitch_pool is not None:
        stitch_pool.shutdown()
    if pool is not None:
        pool.shutdown()
    if bam_out is not None:
        bam_out.close()
    if output_writer is not None:
        output_writer.close()

    stats_counter.update(main_counter)
    timelog.save(f"{output_prefix}.runtime")
    with open(f"{output_prefix}.inference.json", "w") as f:
        json.dump(dict(stats_counter), f, indent=True)

    log.info(
        "Processed %d ZMWs in %.3f seconds",
        main_counter["n_zmw_processed"], time.time() - t_start,
    )
    log.info("Outcome counts: %s", outcome_counter)
    return outcome_counter
