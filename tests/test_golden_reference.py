"""Golden parity vs the reference's bundled real-data testdata.

Runs the full preprocess pipeline (our BAM reader, expand/trim, C++ spacing,
windowing, truth labels, splits) on the reference's human_1m BAMs and
compares against its shipped outputs: every summary counter and every
serialized example tensor must match BIT-EXACTLY.

Skipped when /root/reference is not mounted (e.g. on GPU boxes).
"""
import json
import os

import numpy as np
import pytest

REF = "/root/reference/deepconsensus/testdata/human_1m"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(REF), reason="reference testdata not mounted"
)


@pytest.fixture(scope="module")
def golden_run(tmp_path_factory):
    from deepconsensus_amd.preprocess import preprocess_cli

    tmp = tmp_path_factory.mktemp("golden")
    out = str(tmp / "ex-@split.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", f"{REF}/subreads_to_ccs.bam",
        "--ccs_bam", f"{REF}/ccs.bam",
        "--truth_to_ccs", f"{REF}/truth_to_ccs.bam",
        "--truth_bed", f"{REF}/truth.bed",
        "--truth_split", f"{REF}/truth_split.tsv",
        "--output", out,
        "--cpus", "0",
        "--ins_trim", "5",
    ])
    return str(tmp)


def test_summary_counters_match(golden_run):
    mine = json.load(open(f"{golden_run}/ex-summary.training.json"))
    ref = json.load(
        open(f"{REF}/tf_examples/summary/summary.training.json")
    )
    skip = ("subreads_to_ccs", "ccs_bam", "truth_to_ccs", "truth_bed",
            "truth_split", "version", "ins_trim")
    for k, v in ref.items():
        if k.startswith(skip):
            continue
        assert str(mine.get(k)) == str(v), (k, v, mine.get(k))


@pytest.mark.parametrize("split,n", [("train", 1239), ("eval", 65),
                                     ("test", 203)])
def test_examples_bit_exact(golden_run, split, n):
    from deepconsensus_amd.dcio import example_codec, tfrecord

    def load(path):
        out = {}
        for rec in tfrecord.read_tfrecords(path):
            d = example_codec.decode_example(rec)
            key = (d["name"][1][0], d["window_pos"][1][0])
            out[key] = d
        return out

    ref = load(f"{REF}/tf_examples/{split}/{split}.tfrecord.gz")
    mine = load(f"{golden_run}/ex-{split}.tfrecord.gz")
    assert len(ref) == n and len(mine) == n
    assert set(ref) == set(mine)
    for k in ref:
        a = np.frombuffer(ref[k]["subreads/encoded"][1][0], np.float32)
        b = np.frombuffer(mine[k]["subreads/encoded"][1][0], np.float32)
        assert np.array_equal(a, b), k
        la = np.frombuffer(ref[k]["label/encoded"][1][0], np.float32)
        lb = np.frombuffer(mine[k]["label/encoded"][1][0], np.float32)
        assert np.array_equal(la, lb), k
        assert (ref[k]["ccs_base_quality_scores"][1]
                == mine[k]["ccs_base_quality_scores"][1]), k


def test_inference_mode_counters(tmp_path):
    """Preprocess inference-mode summary matches the reference golden."""
    from deepconsensus_amd.preprocess import preprocess_cli

    out = str(tmp_path / "inf" / "ex.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", f"{REF}/subreads_to_ccs.bam",
        "--ccs_bam", f"{REF}/ccs.bam",
        "--output", out, "--cpus", "0", "--ins_trim", "5",
    ])
    mine = json.load(open(str(tmp_path / "inf" / "ex.inference.json")))
    ref = json.load(
        open(f"{REF}/tf_examples/summary/summary.inference.json")
    )
    skip = ("subreads_to_ccs", "ccs_bam", "truth_to_ccs", "truth_bed",
            "truth_split", "version", "ins_trim")
    for k, v in ref.items():
        if k.startswith(skip):
            continue
        assert str(mine.get(k)) == str(v), (k, v, mine.get(k))


@pytest.mark.parametrize("q", [0, 10, 20, 30, 40, 50])
def test_filter_reads_golden(tmp_path, q):
    """filter_reads reproduces the reference's golden filtered FASTQs."""
    from deepconsensus_amd.calibration import filter_reads as fr
    from deepconsensus_amd.dcio.fastq import read_fastq

    base = "/root/reference/deepconsensus/testdata/filter_fastq"
    inp = f"{base}/m64062_190806_063919_q0_chr20_100reads.fq.gz"
    golden = f"{base}/m64062_190806_063919_q0_chr20_100reads.q{q}.fq.gz"
    out = str(tmp_path / f"out_q{q}.fastq")
    fr.main(["-i", inp, "-o", out, "-q", str(q)])
    got = [(r.name, r.sequence, r.quality) for r in read_fastq(out)]
    want = [(r.name, r.sequence, r.quality) for r in read_fastq(golden)]
    assert got == want


def test_filter_reads_bam_golden(tmp_path):
    """BAM input path against the q30 golden."""
    from deepconsensus_amd.calibration import filter_reads as fr
    from deepconsensus_amd.dcio.fastq import read_fastq

    base = "/root/reference/deepconsensus/testdata/filter_fastq"
    inp = f"{base}/m64062_190806_063919-chr20.dc.small.bam"
    golden = f"{base}/m64062_190806_063919-chr20.dc.small.q30.fq.gz"
    out = str(tmp_path / "out_bam_q30.fastq")
    fr.main(["-i", inp, "-o", out, "-q", "30"])
    got = [(r.name, r.sequence, r.quality) for r in read_fastq(out)]
    want = [(r.name, r.sequence, r.quality) for r in read_fastq(golden)]
    assert got == want


def test_reference_params_json_loads():
    """The reference's shipped params.json resolves through our config."""
    from deepconsensus_amd.models import config as cfg

    p = cfg.read_params_from_json(
        "/root/reference/deepconsensus/testdata/model"
    )
    cfg.modify_params(p, is_training=False)
    assert p.hidden_size == 280
    assert p.total_rows == 85
    assert p.dc_calibration == "0,1.197654,-0.99781"
    from deepconsensus_amd.models.model import get_model

    m = get_model(p)
    n_params = sum(x.numel() for x in m.parameters())
    # 8.94 M trainable parameters; the reference's 38.18 MB checkpoint
    # (docs/quick_start.md:104) implies ~9.5 M fp32 slots incl. bookkeeping.
    assert 8.5e6 < n_params < 10.1e6, n_params


def test_bq_examples_bit_exact(tmp_path):
    """--use_ccs_bq run matches the reference's tf_examples_bq goldens."""
    from deepconsensus_amd.dcio import example_codec, tfrecord
    from deepconsensus_amd.preprocess import preprocess_cli

    out = str(tmp_path / "bq" / "ex-@split.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", f"{REF}/subreads_to_ccs.bam",
        "--ccs_bam", f"{REF}/ccs.bam",
        "--truth_to_ccs", f"{REF}/truth_to_ccs.bam",
        "--truth_bed", f"{REF}/truth.bed",
        "--truth_split", f"{REF}/truth_split.tsv",
        "--output", out, "--cpus", "0", "--ins_trim", "5",
        "--use_ccs_bq",
    ])
    mine_sum = json.load(
        open(str(tmp_path / "bq" / "ex-summary.training.json"))
    )
    ref_sum = json.load(
        open(f"{REF}/tf_examples_bq/summary/summary.training.json")
    )
    for k in ("n_examples_train", "n_examples", "tensor_height"):
        assert str(mine_sum[k]) == str(ref_sum[k]), k

    def load(path):
        out = {}
        for rec in tfrecord.read_tfrecords(path):
            d = example_codec.decode_example(rec)
            out[(d["name"][1][0], d["window_pos"][1][0])] = d
        return out

    ref = load(f"{REF}/tf_examples_bq/train/train.tfrecord.gz")
    mine = load(str(tmp_path / "bq" / "ex-train.tfrecord.gz"))
    assert set(ref) == set(mine)
    for k in ref:
        a = np.frombuffer(ref[k]["subreads/encoded"][1][0], np.float32)
        b = np.frombuffer(mine[k]["subreads/encoded"][1][0], np.float32)
        assert np.array_equal(a, b), k


def test_run_e2e_real_bams_ccs_passthrough(tmp_path):
    """Full `run` on the reference's real BAMs (CPU; skip_windows_above=1
    routes every window through the CCS passthrough, exercising streaming,
    windowing and stitching on real data without a trained model)."""
    from deepconsensus_amd.dcio.fastq import read_fastq
    from deepconsensus_amd.inference import quick_inference as qi

    out = str(tmp_path / "out.fastq")
    options = qi.InferenceOptions(
        batch_size=32, batch_zmws=5, cpus=0, min_quality=0,
        skip_windows_above=1, ins_trim=5,
    )
    counter = qi.run(
        subreads_to_ccs=f"{REF}/subreads_to_ccs.bam",
        ccs_bam=f"{REF}/ccs.bam",
        checkpoint="random", output=out, options=options, device="cpu",
    )
    assert counter.total == 10
    recs = list(read_fastq(out))
    assert len(recs) == counter.success == 10
    # Passthrough: polished seq length matches the CCS length per ZMW.
    from deepconsensus_amd.dcio import bam as bam_lib

    ccs_lens = {r.qname: len(r.seq)
                for r in bam_lib.BamReader(f"{REF}/ccs.bam")}
    for r in recs:
        assert len(r.sequence) == ccs_lens[r.name], r.name


def test_calibrate_tool_real_alignment(tmp_path):
    """calibrate on the reference's real DC-to-truth alignment + CHM13 ref."""
    from deepconsensus_amd.calibration import calculate_baseq_calibration as cc

    base = "/root/reference/deepconsensus/testdata/prediction_assessment"
    out_csv = str(tmp_path / "calib.csv")
    cc.main([
        "--bam", f"{base}/CHM13_chr20_0_200000_dc.to_truth.bam",
        "--ref", f"{base}/CHM13_chr20_0_200000.fa",
        "--output_csv", out_csv, "--cpus", "1",
        "--interval_length", "100000", "--min_mapq", "1",
    ])
    rows = {}
    with open(out_csv) as f:
        next(f)
        for line in f:
            q, m, x = line.strip().split(",")
            rows[int(q)] = (int(m), int(x))
    total_m = sum(m for m, _ in rows.values())
    total_x = sum(x for _, x in rows.values())
    assert total_m > 1_000_000  # real matches counted
    assert total_m > 50 * total_x
    # The curve is calibrated: empirical error at predicted Q30 is ~1e-3,
    # and low-quality bases err orders of magnitude more often.
    m30, x30 = rows[30]
    rate30 = x30 / (m30 + x30)
    assert 2e-4 < rate30 < 5e-3, rate30
    m_lo = sum(rows[q][0] for q in range(2, 7))
    x_lo = sum(rows[q][1] for q in range(2, 7))
    assert x_lo / (m_lo + x_lo) > 0.1


def test_reference_params_json_bq_variant():
    """The ccs_bq model's shipped params.json also resolves (86 rows)."""
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.model import get_model

    p = cfg.read_params_from_json(
        "/root/reference/deepconsensus/testdata/model_bq"
    )
    cfg.modify_params(p, is_training=False)
    assert p.use_ccs_bq
    assert p.total_rows == 86
    m = get_model(p)
    out = m(__import__("torch").zeros(2, 86, 100))
    assert out.shape == (2, 100, 5)
