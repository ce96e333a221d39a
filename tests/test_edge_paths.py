"""Edge-path tests: CCS smart windows, overflow windows, label overflow,
ccs_bq examples — the fidelity corners of pre_lib (SURVEY.md hard part #2)."""

import numpy as np

from deepconsensus_amd.dcio import bam as bam_lib, tfrecord
from deepconsensus_amd.preprocess import read as R
from deepconsensus_amd.preprocess.windows import DcConfig, DcExample
from deepconsensus_amd.utils import constants

from test_io_and_pipeline import make_test_bams
from test_preprocess import make_segment, _spacing_reads


def _spaced_zmw_with_insertions(length=150, ins_at=50, ins_len=8):
    """One subread with a big insertion -> spaced CCS contains gaps."""
    from deepconsensus_amd.preprocess.expand import expand_clip_indent

    rng = np.random.default_rng(0)
    seq = "".join(rng.choice(list("ATCG"), size=length))
    ins = "".join(rng.choice(list("ATCG"), size=ins_len))
    sub_seq = seq[:ins_at] + ins + seq[ins_at:]
    cigar = f"{ins_at}M{ins_len}I{length - ins_at}M"
    seg = make_segment(sub_seq, cigar, ip=[1] * len(sub_seq),
                       pw=[2] * len(sub_seq))
    sub = expand_clip_indent(seg)
    ccs = R.Read(
        name="m/5/ccs",
        bases=np.array(list(seq), dtype="<U1"),
        cigar=np.repeat(np.uint8(0), length),
        pw=np.zeros(length, dtype=np.uint8),
        ip=np.zeros(length, dtype=np.uint8),
        sn=np.zeros(4),
        strand=constants.Strand.UNKNOWN,
        base_quality_scores=np.full(length, 30),
        ccs_idx=np.arange(length),
    )
    return R.space_out_subreads([sub, ccs])


def test_smart_windows_spaced_widths():
    """CCS 'wl' widths are measured in CCS bases; spacing widens windows."""
    reads = _spaced_zmw_with_insertions(length=150, ins_at=50, ins_len=8)
    widths = np.array([60, 60, 30])
    ex = DcExample("m/5/ccs", reads, DcConfig(20, 100),
                   window_widths=widths)
    spaced = ex.calculate_windows(100)
    # First window covers 60 CCS bases + the 8 inserted columns.
    assert spaced[0] == 68
    assert spaced[1] == 60 and spaced[2] == 30
    windows = list(ex.iter_examples())
    assert len(windows) == 3
    # All windows fit max_length -> no overflow.
    assert ex.counter["n_examples_overflow"] == 0


def test_smart_windows_overflow_flagged():
    """A smart window wider than max_length is flagged overflow (inference
    keeps it for CCS passthrough; training drops it)."""
    reads = _spaced_zmw_with_insertions(length=150, ins_at=50, ins_len=8)
    widths = np.array([120, 30])
    ex = DcExample("m/5/ccs", reads, DcConfig(20, 100),
                   window_widths=widths)
    windows = list(ex.iter_examples())
    assert ex.counter["n_examples_overflow"] == 1
    assert any(w._overflow for w in windows)
    ov = [w for w in windows if w._overflow][0]
    feats = ov.to_features_dict()
    assert feats["overflow"] is True


def test_label_overflow_adjusted_and_dropped():
    """Training labels longer than max_length: gaps removed, else dropped
    (pre_lib.py:669-689)."""
    # Label with a 20-bp INTERNAL insertion -> window slice longer than
    # max_length and gap removal cannot shrink it (50 real bases > 40).
    bases = ["T" * 30, "T" * 30, "T" * 15 + "G" * 20 + "T" * 15]
    cigars = ["M" * 30, "M" * 30, "M" * 15 + "I" * 20 + "M" * 15]
    ccs_idx = [list(range(30)), list(range(30)),
               list(range(15)) + [-1] * 20 + list(range(15, 30))]
    reads = _spacing_reads(bases, cigars, ccs_idx,
                           {"contig": "chr1", "begin": 0, "end": 50})
    spaced = R.space_out_subreads(reads)
    # Window of 40: label is 50 long -> gap removal fits it (no gaps, 50>40
    # -> dropped for training).
    ex = DcExample("m/1/ccs", spaced, DcConfig(20, 40))
    windows = list(ex.iter_examples())
    assert ex.counter["n_examples_label_overflow"] == 1
    assert not windows


def test_preprocess_ccs_bq_rows(tmp_path):
    """--use_ccs_bq emits 86-row examples with the bq row populated."""
    from deepconsensus_amd.preprocess import preprocess_cli

    sub, ccs = make_test_bams(tmp_path, n_zmws=2, length=150)
    out = str(tmp_path / "bq" / "ex.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", sub, "--ccs_bam", ccs,
        "--output", out, "--cpus", "0", "--use_ccs_bq",
    ])
    recs = list(tfrecord.read_tfrecords(out))
    assert recs
    from deepconsensus_amd.preprocess.windows import tf_example_to_features_dict

    feats = tf_example_to_features_dict(recs[0], inference=True)
    assert feats["subreads/shape"] == [86, 100, 1]
    bq_row = feats["subreads"][81, :, 0]
    # bq row holds CCS base qualities (20..40) with -1 padding.
    assert bq_row.max() >= 20


def test_quick_inference_smart_windows(tmp_path):
    """`run --use_ccs_smart_windows` consumes the wl tag end to end."""
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=2, length=160, seed=9)
    # Rewrite the ccs bam with wl tags.
    reads = list(bam_lib.BamReader(ccs))
    header = bam_lib.BamReader(ccs).header
    ccs2 = str(tmp_path / "ccs_wl.bam")
    with bam_lib.BamWriter(ccs2, header) as w:
        for r in reads:
            r.tags["wl"] = np.array([80, 80], dtype=np.int32)
            w.write(r)
    out = str(tmp_path / "out.fastq")
    options = qi.InferenceOptions(
        batch_size=8, batch_zmws=2, cpus=0, min_quality=0,
        skip_windows_above=0, use_ccs_smart_windows=True,
    )
    counter = qi.run(subreads_to_ccs=sub, ccs_bam=ccs2, checkpoint="random",
                     output=out, options=options, device="cpu")
    assert counter.total == 2
    from deepconsensus_amd.dcio.fastq import read_fastq

    assert len(list(read_fastq(out))) == 2
