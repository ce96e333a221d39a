"""Tests for the calibration tool, FASTA IO, export tool, analysis utils."""
import os

import torch

from deepconsensus_amd.dcio import bam as bam_lib
from deepconsensus_amd.dcio.fasta import FastaFile, write_fasta
from deepconsensus_amd.utils import analysis


def test_fasta_round_trip(tmp_path):
    p = str(tmp_path / "ref.fa")
    write_fasta(p, {"chr1": "ACGT" * 50, "chr2": "TTTT"})
    f = FastaFile(p)
    assert f.references == ["chr1", "chr2"]
    assert f.get_reference_length("chr1") == 200
    assert f.fetch("chr1", 0, 4) == "ACGT"
    assert f.fetch("chr2") == "TTTT"


def test_calculate_baseq_calibration(tmp_path):
    from deepconsensus_amd.calibration import calculate_baseq_calibration as cc

    ref_seq = "ACGT" * 25  # 100 bp
    fa = str(tmp_path / "ref.fa")
    write_fasta(fa, {"chr1": ref_seq})

    header = bam_lib.BamHeader(text="@HD\tVN:1.6",
                               references=[("chr1", 100)])
    bam_path = str(tmp_path / "aln.bam")
    with bam_lib.BamWriter(bam_path, header) as w:
        # Read 1: perfect match, Q30 -> 100 matches at q30.
        w.write(bam_lib.BamRead(
            qname="r1", flag=0, ref_id=0, pos=0, mapq=60,
            cigartuples=[(0, 100)], seq=ref_seq,
            query_qualities=[30] * 100, tags={},
        ))
        # Read 2: 1 mismatch at pos 0 (A->T), Q20.
        seq2 = "T" + ref_seq[1:]
        w.write(bam_lib.BamRead(
            qname="r2", flag=0, ref_id=0, pos=0, mapq=60,
            cigartuples=[(0, 100)], seq=seq2,
            query_qualities=[20] * 100, tags={},
        ))
        # Read 3: low mapq -> skipped.
        w.write(bam_lib.BamRead(
            qname="r3", flag=0, ref_id=0, pos=0, mapq=10,
            cigartuples=[(0, 100)], seq=ref_seq,
            query_qualities=[30] * 100, tags={},
        ))
    out_csv = str(tmp_path / "out.csv")
    cc.main(["--bam", bam_path, "--ref", fa, "--output_csv", out_csv,
             "--cpus", "1", "--interval_length", "60"])
    rows = {}
    with open(out_csv) as f:
        next(f)
        for line in f:
            q, m, x = line.strip().split(",")
            rows[int(q)] = (int(m), int(x))
    # Position 60 sits in both intervals (inclusive stop, reference
    # semantics), so one duplicate count at the seam: 101 matches.
    assert rows[30][0] == 101
    assert rows[30][1] == 0
    assert rows[20][1] == 1
    assert rows[20][0] == 100


def test_export_and_load(tmp_path, monkeypatch):
    from deepconsensus_amd.models import checkpoint as ckpt_lib
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models import export_model
    from deepconsensus_amd.models.model import get_model

    params = cfg.get_config("transformer_learn_values+test")
    cfg.modify_params(params)
    torch.manual_seed(0)
    model = get_model(params)
    ckpt_dir = str(tmp_path / "ckpt")
    ckpt_lib.save_checkpoint(ckpt_dir, 1, 0, model, None, params)

    out_dir = str(tmp_path / "bundle")
    traced_path = export_model.export(ckpt_dir, out_dir)
    assert os.path.exists(traced_path)
    assert os.path.exists(os.path.join(out_dir, "params.json"))

    loaded = torch.jit.load(traced_path)
    x = torch.zeros(1, params.total_rows, params.max_length)
    with torch.no_grad():
        probs_traced = loaded(x)
        probs_model = model(x)
    torch.testing.assert_close(probs_traced, probs_model, atol=1e-5,
                               rtol=1e-5)
    # The bundle also restores as a checkpoint directory.
    m2 = get_model(params)
    ckpt_lib.load_checkpoint(out_dir, m2)


def test_analysis_utils():
    assert analysis.edit_distance("ACGT", "ACGT") == 0
    assert analysis.edit_distance("ACGT", "AGT") == 1
    assert analysis.edit_distance("AAAA", "TTTT") == 4
    assert analysis.homopolymer_content("AAATCG", min_run=3) == 0.5
    assert analysis.longest_homopolymer("AATTTTGC") == 4
    d = analysis.per_base_error_counts("ACGT", "ACCT")
    assert d["match"] == 3 and d["mismatch"] == 1


def test_cli_dispatch_help(capsys):
    from deepconsensus_amd import cli

    cli.main(["--version"])
    out = capsys.readouterr().out
    assert "deepconsensus-amd" in out
    cli.main([])
    out = capsys.readouterr().out
    assert "run" in out and "preprocess" in out


def test_cli_index_subcommand(tmp_path, capsys):
    """`deepconsensus index <bam>` writes a loadable sidecar."""
    from deepconsensus_amd import cli
    from deepconsensus_amd.dcio import bam as bam_lib
    from tests.test_io_and_pipeline import make_test_bams

    sub, ccs = make_test_bams(tmp_path, n_zmws=3)
    cli.main(["index", sub, ccs])
    out = capsys.readouterr().out
    assert "3 ZMW groups" in out
    idx = bam_lib.load_zmw_index(sub)
    assert idx is not None and len(idx["zmw"]) == 3
    assert bam_lib.load_zmw_index(ccs) is not None


def test_parse_calibration_string():
    from deepconsensus_amd.calibration import calibration as cal

    skip = cal.parse_calibration_string("skip")
    assert not skip.enabled and skip.w == 1.0 and skip.b == 0.0
    v = cal.parse_calibration_string("10,0.99,0.15")
    assert v.enabled and v.threshold == 10 and v.w == 0.99 and v.b == 0.15
    import pytest as _pytest

    with _pytest.raises(ValueError):
        cal.parse_calibration_string("1,2")
    with _pytest.raises(ValueError):
        cal.parse_calibration_string("not,a,number")


def test_calibrate_quality_scores_threshold_semantics():
    import numpy as np

    from deepconsensus_amd.calibration import calibration as cal

    q = np.array([5.0, 10.0, 20.0, 40.0])
    # threshold 0: applied everywhere.
    v0 = cal.parse_calibration_string("0,2,1")
    np.testing.assert_allclose(
        cal.calibrate_quality_scores(q, v0), q * 2 + 1
    )
    # threshold 10: strictly-above only; at/below unchanged.
    v10 = cal.parse_calibration_string("10,2,1")
    np.testing.assert_allclose(
        cal.calibrate_quality_scores(q, v10), [5.0, 10.0, 41.0, 81.0]
    )
    # v1.2 production string maps Q30 to ~Q35 (monotone, finite).
    vp = cal.parse_calibration_string("0,1.197654,-0.99781")
    out = cal.calibrate_quality_scores(np.array([30.0]), vp)
    assert 34.5 < out[0] < 35.1


def test_run_ccs_fasta_deprecated():
    import pytest as _pytest

    from deepconsensus_amd import cli

    with _pytest.raises(NotImplementedError, match="deprecated"):
        cli.main(["run", "--subreads_to_ccs", "x.bam", "--ccs_bam", "y.bam",
                  "--checkpoint", "random", "--output", "o.fastq",
                  "--ccs_fasta", "z.fa"])


def test_run_accepts_exported_bundle(tmp_path):
    """`deepconsensus run` serves from an exported bundle directory
    (reference: SavedModel-or-checkpoint detection, quick_inference
    .py:797-800)."""
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from test_io_and_pipeline import make_test_bams

    from deepconsensus_amd.dcio.fastq import read_fastq
    from deepconsensus_amd.inference import quick_inference as qi
    from deepconsensus_amd.models import checkpoint as ckpt_lib
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models import export_model
    from deepconsensus_amd.models.model import get_model

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params)
    torch.manual_seed(2)
    model = get_model(params)
    ckpt_dir = str(tmp_path / "ckpt")
    ckpt_lib.save_checkpoint(ckpt_dir, 1, 0, model, None, params)
    bundle = str(tmp_path / "bundle")
    export_model.export(ckpt_dir, bundle)

    sub, ccs = make_test_bams(tmp_path, n_zmws=2, length=150, seed=9)
    out = str(tmp_path / "out.fastq")
    counter = qi.run(
        subreads_to_ccs=sub, ccs_bam=ccs, checkpoint=bundle, output=out,
        options=qi.InferenceOptions(batch_size=8, batch_zmws=2, cpus=0,
                                    min_quality=0, skip_windows_above=0),
        device="cpu",
    )
    assert counter.total == 2
    assert len(list(read_fastq(out))) == 2
