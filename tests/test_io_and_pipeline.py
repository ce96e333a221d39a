"""I/O round-trips + end-to-end pipeline tests (`run` and `preprocess` on
synthetic BAMs), mirroring reference preprocess_test.py's E2E style with
in-repo fixtures."""
import json
import os

import numpy as np
import pytest

from deepconsensus_amd.dcio import bam as bam_lib
from deepconsensus_amd.dcio import example_codec, tfrecord
from deepconsensus_amd.postprocess import stitch
from deepconsensus_amd.utils import phred


def _random_seq(rng, n):
    return "".join(rng.choice(list("ATCG"), size=n))


def make_test_bams(tmp_path, n_zmws=3, length=220, n_subreads=4, seed=7,
                   unmapped_zmws=()):
    """Writes synthetic subreads_to_ccs.bam + ccs.bam; returns paths.

    ``unmapped_zmws``: ZMW ordinals (0-based) whose subreads are all
    written FUNMAP — such a zm-run is indexed by build_zmw_index but
    never emitted by the grouper (the sharding accounting regression
    from ADVICE r1)."""
    rng = np.random.default_rng(seed)
    refs = []
    zmw_seqs = {}
    for z in range(n_zmws):
        name = f"m000/{z + 10}/ccs"
        seq = _random_seq(rng, length)
        refs.append((name, length))
        zmw_seqs[name] = seq
    header = bam_lib.BamHeader(text="@HD\tVN:1.6", references=refs)

    sub_path = str(tmp_path / "subreads_to_ccs.bam")
    with bam_lib.BamWriter(sub_path, header) as w:
        for rid, (name, ln) in enumerate(refs):
            zm = int(name.split("/")[1])
            seq = zmw_seqs[name]
            if rid in unmapped_zmws:
                read = bam_lib.BamRead(
                    qname=f"m000/{zm}/0_{ln}",
                    flag=4,
                    ref_id=-1,
                    pos=-1,
                    cigartuples=[],
                    seq=seq,
                    query_qualities=[30] * ln,
                    tags={"zm": zm},
                )
                w.write(read)
                continue
            for i in range(n_subreads):
                # Introduce a small mutation region per subread.
                s = list(seq)
                p = int(rng.integers(0, ln - 1))
                s[p] = rng.choice(list("ATCG"))
                read = bam_lib.BamRead(
                    qname=f"m000/{zm}/{i * 100}_{i * 100 + ln}",
                    flag=16 if i % 2 else 0,
                    ref_id=rid,
                    pos=0,
                    mapq=60,
                    cigartuples=[(0, ln)],
                    seq="".join(s),
                    query_qualities=[30] * ln,
                    tags={
                        "zm": zm,
                        "pw": rng.integers(0, 60, ln).astype(np.uint8),
                        "ip": rng.integers(0, 60, ln).astype(np.uint8),
                        "sn": np.array([6.0, 7.0, 5.5, 9.1],
                                       dtype=np.float32),
                    },
                )
                w.write(read)

    ccs_path = str(tmp_path / "ccs.bam")
    with bam_lib.BamWriter(ccs_path, header) as w:
        for rid, (name, ln) in enumerate(refs):
            zm = int(name.split("/")[1])
            read = bam_lib.BamRead(
                qname=name,
                flag=4,
                ref_id=-1,
                pos=-1,
                cigartuples=[],
                seq=zmw_seqs[name],
                query_qualities=rng.integers(20, 40, ln),
                tags={"zm": zm, "ec": 11.5, "np": n_subreads, "rq": 0.998,
                      "RG": "rg0"},
            )
            w.write(read)
    return sub_path, ccs_path


def test_bam_round_trip(tmp_path):
    sub, ccs = make_test_bams(tmp_path)
    reads = list(bam_lib.BamReader(sub))
    assert len(reads) == 12
    r = reads[0]
    assert r.qname.startswith("m000/10/")
    assert r.cigartuples == [(0, 220)]
    assert len(r.seq) == 220
    assert r.get_tag("zm") == 10
    assert r.get_tag("sn").shape == (4,)
    assert r.reference_name == "m000/10/ccs"
    assert not r.is_unmapped
    reads2 = list(bam_lib.BamReader(ccs))
    assert reads2[0].is_unmapped
    assert abs(reads2[0].get_tag("rq") - 0.998) < 1e-6
    # fetch_index groups by reference name.
    idx = bam_lib.fetch_index(sub)
    assert len(idx) == 3
    assert len(idx["m000/10/ccs"]) == 4


def test_raw_bam_reader_matches_decoded(tmp_path):
    """RawBamReader buffers + decode_record reproduce BamReader exactly,
    and the raw_* peeks agree with the decoded fields."""
    sub, ccs = make_test_bams(tmp_path)
    decoded = list(bam_lib.BamReader(sub))
    raw_reader = bam_lib.RawBamReader(sub)
    raws = list(raw_reader)
    assert len(raws) == len(decoded)
    for buf, ref in zip(raws, decoded):
        assert bam_lib.raw_qname(buf) == ref.qname
        assert bam_lib.raw_flag(buf) == ref.flag
        assert bam_lib.raw_ref_id(buf) == ref.ref_id
        assert bam_lib.raw_tag(buf, "zm") == ref.get_tag("zm")
        np.testing.assert_array_equal(
            bam_lib.raw_tag(buf, "pw"), ref.get_tag("pw")
        )
        assert bam_lib.raw_tag(buf, "absent", default=-1) == -1
        got = bam_lib.decode_record(buf, raw_reader.header)
        assert got.qname == ref.qname
        assert got.cigartuples == ref.cigartuples
        assert got.seq == ref.seq
        np.testing.assert_array_equal(
            got.query_qualities, ref.query_qualities
        )
        assert got.reference_name == ref.reference_name
        assert set(got.tags) == set(ref.tags)


def test_raw_feeder_matches_decoded_feeder(tmp_path):
    """create_proc_feeder(raw_records=True) materializes to the same
    expanded Read stacks as the decoded deferred feeder."""
    from deepconsensus_amd.preprocess import feeder as pre_feeder
    from deepconsensus_amd.preprocess.windows import DcConfig

    sub, ccs = make_test_bams(tmp_path)
    dc_config = DcConfig(20, 100, False)
    jobs = {}
    for raw in (False, True):
        pf, counter = pre_feeder.create_proc_feeder(
            subreads_to_ccs=sub, ccs_bam=ccs, dc_config=dc_config,
            defer_expansion=True, raw_records=raw,
        )
        jobs[raw] = [(z, job) for job, z, *_ in pf()]
        assert counter["n_zmw_processed"] == 3
    assert [z for z, _ in jobs[False]] == [z for z, _ in jobs[True]]
    for (_, a), (_, b) in zip(jobs[False], jobs[True]):
        assert isinstance(b, pre_feeder.RawZmwJob)
        assert len(a) == len(b)
        import collections

        ca, cb = collections.Counter(), collections.Counter()
        reads_a, reads_b = a.materialize(ca), b.materialize(cb)
        assert ca == cb
        assert len(reads_a) == len(reads_b)
        for ra, rb in zip(reads_a, reads_b):
            assert ra.name == rb.name
            np.testing.assert_array_equal(ra.bases, rb.bases)
            np.testing.assert_array_equal(ra.pw, rb.pw)
            np.testing.assert_array_equal(ra.ip, rb.ip)
            np.testing.assert_array_equal(ra.ccs_idx, rb.ccs_idx)


def test_zmw_index_resume_points(tmp_path):
    """build_zmw_index entries land exactly on ZMW group starts: resuming
    a RawBamReader at each entry yields that group's records first."""
    sub, ccs = make_test_bams(tmp_path, n_zmws=5)
    idx_path = bam_lib.build_zmw_index(sub)
    assert idx_path == sub + bam_lib.ZMW_INDEX_SUFFIX
    idx = bam_lib.load_zmw_index(sub)
    assert idx["sorted_flag"][0] == 1
    assert list(idx["zmw"]) == [10, 11, 12, 13, 14]
    all_by_zmw = {}
    for r in bam_lib.BamReader(sub):
        all_by_zmw.setdefault(r.get_tag("zm"), []).append(r.qname)
    for g in range(len(idx["zmw"])):
        start = (int(idx["coffset"][g]), int(idx["uoffset"][g]))
        reader = bam_lib.RawBamReader(sub, start=start)
        want = [q for z in idx["zmw"][g:] for q in all_by_zmw[z]]
        got = [bam_lib.raw_qname(b) for b in reader]
        assert got == want, f"group {g} resume mismatch"


def test_feeder_byte_range_sharding(tmp_path):
    """With index sidecars, shards are contiguous, disjoint, and their
    union covers every ZMW with the same job contents as unsharded."""
    import collections

    from deepconsensus_amd.preprocess import feeder as pre_feeder
    from deepconsensus_amd.preprocess.windows import DcConfig

    sub, ccs = make_test_bams(tmp_path, n_zmws=5)
    bam_lib.build_zmw_index(sub)
    bam_lib.build_zmw_index(ccs)
    dc_config = DcConfig(20, 100, False)

    def collect(shard_index, shard_count):
        pf, counter = pre_feeder.create_proc_feeder(
            subreads_to_ccs=sub, ccs_bam=ccs, dc_config=dc_config,
            defer_expansion=True, shard_index=shard_index,
            shard_count=shard_count,
        )
        out = {}
        for job, zmw, *_ in pf():
            reads = job.materialize(collections.Counter())
            out[zmw] = [r.name for r in reads]
        return out

    full = collect(0, 1)
    assert len(full) == 5
    sharded = [collect(i, 3) for i in range(3)]
    seen = {}
    for s in sharded:
        assert not (set(s) & set(seen))
        seen.update(s)
    assert seen == full


def test_feeder_byte_range_sharding_with_unmapped_zmw(tmp_path):
    """An all-unmapped zm-run is indexed but never emitted; shards must
    still be disjoint and complete (the run counts toward max_groups —
    ADVICE r1: a shard otherwise read past its boundary and duplicated
    the next shard's first ZMW)."""
    import collections

    from deepconsensus_amd.preprocess import feeder as pre_feeder
    from deepconsensus_amd.preprocess.windows import DcConfig

    # ZMW ordinal 1 sits inside shard 0's range of the 6-group index.
    sub, ccs = make_test_bams(tmp_path, n_zmws=6, unmapped_zmws=(1, 4))
    bam_lib.build_zmw_index(sub)
    bam_lib.build_zmw_index(ccs)
    idx = bam_lib.load_zmw_index(sub)
    assert len(idx["zmw"]) == 6  # all-unmapped runs ARE indexed
    dc_config = DcConfig(20, 100, False)

    def collect(shard_index, shard_count):
        pf, _ = pre_feeder.create_proc_feeder(
            subreads_to_ccs=sub, ccs_bam=ccs, dc_config=dc_config,
            defer_expansion=True, shard_index=shard_index,
            shard_count=shard_count,
        )
        out = {}
        for job, zmw, *_ in pf():
            reads = job.materialize(collections.Counter())
            out[zmw] = [r.name for r in reads]
        return out

    full = collect(0, 1)
    assert len(full) == 4  # unmapped ZMWs never emit
    for n_shards in (2, 3):
        seen = {}
        for i in range(n_shards):
            s = collect(i, n_shards)
            dup = set(s) & set(seen)
            assert not dup, f"duplicate ZMWs across shards: {dup}"
            seen.update(s)
        assert seen == full, (n_shards, sorted(seen), sorted(full))


def test_quick_inference_sharding_with_index(tmp_path):
    """E2E sharded run over index sidecars covers all ZMWs once."""
    from deepconsensus_amd.dcio.fastq import read_fastq
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=4, length=120, seed=11)
    bam_lib.build_zmw_index(sub)
    bam_lib.build_zmw_index(ccs)
    names = []
    for i in range(2):
        out = str(tmp_path / f"outi{i}.fastq")
        options = qi.InferenceOptions(
            batch_size=8, batch_zmws=2, cpus=0, min_quality=0,
            skip_windows_above=0, shard_index=i, shard_count=2,
        )
        qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
               output=out, options=options, device="cpu")
        names.append({r.name for r in read_fastq(out)})
    assert names[0] and names[1]
    assert not (names[0] & names[1])
    assert len(names[0] | names[1]) == 4


def test_example_codec_round_trip():
    feats = {
        "subreads/encoded": (example_codec.BYTES, [b"\x00\x01\x02\x03"]),
        "subreads/shape": (example_codec.INT64, [85, 100, 1]),
        "name": (example_codec.BYTES, [b"m0/1/ccs"]),
        "floats": (example_codec.FLOAT, [1.5, -2.25]),
    }
    enc = example_codec.encode_example(feats)
    dec = example_codec.decode_example(enc)
    assert dec["subreads/shape"] == (example_codec.INT64, [85, 100, 1])
    assert dec["name"][1][0] == b"m0/1/ccs"
    assert dec["floats"][0] == example_codec.FLOAT
    np.testing.assert_allclose(dec["floats"][1], [1.5, -2.25])


def test_tfrecord_round_trip(tmp_path):
    path = str(tmp_path / "x.tfrecord.gz")
    records = [b"alpha", b"beta" * 100, b""]
    with tfrecord.TFRecordWriter(path, compression="gzip") as w:
        for r in records:
            w.write(r)
    got = list(tfrecord.read_tfrecords(path, check_crc=True))
    assert got == records


def test_stitch_basic():
    outs = [
        stitch.DCModelOutput(
            molecule_name="m/1/ccs", window_pos=0,
            sequence="AAAA TT", quality_string="IIIIIII",
        ),
        stitch.DCModelOutput(
            molecule_name="m/1/ccs", window_pos=7,
            sequence="GGGG", quality_string="IIII",
        ),
    ]
    counter = stitch.OutcomeCounter()
    fq = stitch.stitch_to_fastq("m/1/ccs", outs, max_length=7,
                                min_quality=10, min_length=1,
                                outcome_counter=counter)
    lines = fq.splitlines()
    assert lines[0] == "@m/1/ccs"
    assert lines[1] == "AAAATTGGGG"  # gap removed
    assert len(lines[3]) == 10
    assert counter.success == 1


def test_stitch_missing_window_discards():
    outs = [
        stitch.DCModelOutput(
            molecule_name="m/1/ccs", window_pos=100,
            sequence="GGGG", quality_string="IIII",
        ),
    ]
    counter = stitch.OutcomeCounter()
    fq = stitch.stitch_to_fastq("m/1/ccs", outs, max_length=100,
                                min_quality=10, min_length=1,
                                outcome_counter=counter)
    assert fq is None
    assert counter.empty_sequence == 1


def test_stitch_quality_rounding():
    # all-Q10 read passes a min_quality=10 filter (rounding caveat,
    # stitch_utils.py:101-109).
    q10 = phred.quality_scores_to_string(np.full(50, 10))
    outs = [stitch.DCModelOutput(molecule_name="m", window_pos=0,
                                 sequence="A" * 50, quality_string=q10)]
    counter = stitch.OutcomeCounter()
    fq = stitch.stitch_to_fastq("m", outs, max_length=50, min_quality=10,
                                min_length=1, outcome_counter=counter)
    assert fq is not None


def test_quick_inference_e2e_fastq(tmp_path):
    """Full `run` on synthetic BAMs with a random-init model (CPU)."""
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=3, length=220)
    out = str(tmp_path / "out.fastq")
    options = qi.InferenceOptions(
        batch_size=16, batch_zmws=2, cpus=0, min_quality=0,
        skip_windows_above=0,
    )
    counter = qi.run(
        subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
        output=out, options=options, device="cpu",
    )
    assert counter.total == 3
    # Runtime CSV + stats JSON written (incl. the prefetch-wait stage).
    runtime = open(str(tmp_path / "out.runtime.csv")).read()
    assert "wait_preprocess" in runtime and "run_model" in runtime
    # Full wall accounting: serial feeder pulls and device/runner
    # startup are stages too (gpurun_out/pipe3.log attribution).
    assert "feeder" in runtime and "startup_runner" in runtime
    stats = json.load(open(tmp_path / "out.inference.json"))
    assert stats["n_zmw_processed"] == 3
    # Every ZMW produced output (min_quality=0 disables filtering).
    from deepconsensus_amd.dcio.fastq import read_fastq

    recs = list(read_fastq(out))
    assert len(recs) == 3
    for r in recs:
        assert len(r.sequence) == len(r.quality)
        assert len(r.sequence) > 0


def test_quick_inference_skip_windows(tmp_path):
    """skip_windows_above adopts CCS bases for high-quality windows."""
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=2, length=150, seed=3)
    out = str(tmp_path / "out.fastq")
    options = qi.InferenceOptions(
        batch_size=16, batch_zmws=2, cpus=0, min_quality=0,
        skip_windows_above=20,  # CCS quals are 20-40 -> most windows skip
    )
    counter = qi.run(
        subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
        output=out, options=options, device="cpu",
    )
    assert counter.total == 2
    stats = json.load(open(tmp_path / "out.inference.json"))
    assert stats["n_zmw_processed"] == 2


def test_quick_inference_bam_output(tmp_path):
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=2, length=120, seed=5)
    out = str(tmp_path / "out.bam")
    options = qi.InferenceOptions(batch_size=8, batch_zmws=2, cpus=0,
                                  min_quality=0, skip_windows_above=0)
    qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
           output=out, options=options, device="cpu")
    reads = list(bam_lib.BamReader(out))
    assert len(reads) == 2
    r = reads[0]
    assert r.flag == 4
    assert r.has_tag("zm") and r.has_tag("rq") and r.has_tag("RG")
    assert len(r.seq) > 0


def test_preprocess_cli_e2e(tmp_path):
    """`preprocess` inference mode: serial + parallel produce same count."""
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models import data as data_lib
    from deepconsensus_amd.preprocess import preprocess_cli

    sub, ccs = make_test_bams(tmp_path, n_zmws=3, length=220)
    out0 = str(tmp_path / "serial" / "ex.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", sub, "--ccs_bam", ccs,
        "--output", out0, "--cpus", "0",
    ])
    recs = list(tfrecord.read_tfrecords(out0))
    assert len(recs) == 9  # 3 ZMWs x ceil(220/100) windows
    summary = json.load(
        open(str(tmp_path / "serial" / "ex.inference.json"))
    )
    assert summary["n_zmw_processed"] == 3
    assert summary["max_passes"] == "20"

    out2 = str(tmp_path / "par" / "ex.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", sub, "--ccs_bam", ccs,
        "--output", out2, "--cpus", "2",
    ])
    recs2 = list(tfrecord.read_tfrecords(out2))
    assert len(recs2) == 9
    # Pool mode produces the same records (order may differ through the
    # writer queue) and the same summary counters as serial mode.
    assert sorted(recs) == sorted(recs2)
    summary2 = json.load(open(str(tmp_path / "par" / "ex.inference.json")))
    assert {k: v for k, v in summary2.items() if k != "cpus"} == {
        k: v for k, v in summary.items() if k != "cpus"
    }

    # Records parse through the data layer.
    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params, is_training=False)
    ex = data_lib.process_input(recs[0], params, inference=True)
    assert ex["rows"].shape == (85, 100, 1)
    # Window positions monotonic within one ZMW.
    by_name = {}
    for r in recs:
        d = data_lib.process_input(r, params, inference=True)
        by_name.setdefault(d["name"], []).append(int(d["window_pos"]))
    for name, poss in by_name.items():
        assert poss == sorted(poss)


def test_filter_reads_fastq(tmp_path):
    from deepconsensus_amd.calibration import filter_reads as fr
    from deepconsensus_amd.dcio.fastq import FastqRecord, read_fastq, write_fastq

    inp = str(tmp_path / "in.fastq")
    write_fastq(inp, [
        FastqRecord("good", "ACGT" * 10, "I" * 40),   # Q40
        FastqRecord("bad", "ACGT" * 10, "+" * 40),    # Q10
    ])
    outp = str(tmp_path / "out.fastq")
    fr.main(["-i", inp, "-o", outp, "-q", "20"])
    recs = list(read_fastq(outp))
    assert [r.name for r in recs] == ["good"]
    # Threshold exactly at the quality keeps the read (rounding).
    fr.main(["-i", inp, "-o", outp, "-q", "10"])
    assert len(list(read_fastq(outp))) == 2


def test_quick_inference_sharding(tmp_path):
    """Two shards cover disjoint ZMWs whose union is the full set."""
    from deepconsensus_amd.inference import quick_inference as qi
    from deepconsensus_amd.dcio.fastq import read_fastq

    sub, ccs = make_test_bams(tmp_path, n_zmws=4, length=120, seed=11)
    names = []
    for i in range(2):
        out = str(tmp_path / f"out{i}.fastq")
        options = qi.InferenceOptions(
            batch_size=8, batch_zmws=2, cpus=0, min_quality=0,
            skip_windows_above=0, shard_index=i, shard_count=2,
        )
        qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
               output=out, options=options, device="cpu")
        names.append({r.name for r in read_fastq(out)})
    assert names[0] and names[1]
    assert not (names[0] & names[1])
    assert len(names[0] | names[1]) == 4


def test_quick_inference_multiprocess_cpus(tmp_path):
    """cpus=2 routes preprocessing through a ProcessPoolExecutor (pickling
    of BamRead/Read across process boundaries)."""
    from deepconsensus_amd.inference import quick_inference as qi
    from deepconsensus_amd.dcio.fastq import read_fastq

    sub, ccs = make_test_bams(tmp_path, n_zmws=3, length=150, seed=21)
    out = str(tmp_path / "out.fastq")
    options = qi.InferenceOptions(
        batch_size=8, batch_zmws=2, cpus=2, min_quality=0,
        skip_windows_above=0,
    )
    counter = qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
                     output=out, options=options, device="cpu")
    assert counter.total == 3
    assert len(list(read_fastq(out))) == 3


@pytest.mark.parametrize("stage,expect_fastq", [
    ("dc_input", False), ("tf_examples", False), ("run_model", False),
])
def test_run_end_after_stage(tmp_path, stage, expect_fastq):
    """--end_after_stage stops the pipeline early (reference DebugStage)."""
    from deepconsensus_amd import cli

    sub, ccs = make_test_bams(tmp_path, n_zmws=2, length=150, seed=5)
    out = str(tmp_path / "out.fastq")
    cli.main(["run", "--subreads_to_ccs", sub, "--ccs_bam", ccs,
              "--checkpoint", "random", "--output", out,
              "--batch_size", "8", "--min_quality", "0",
              "--skip_windows_above", "0", "--device", "cpu",
              "--end_after_stage", stage])
    fq = open(out).read()
    assert bool(fq.strip()) == expect_fastq
    # runtime CSV is still written with the stages that ran
    runtime = open(str(tmp_path / "out.runtime.csv")).read()
    assert "preprocess" in runtime
    if stage in ("dc_input", "tf_examples"):
        assert "run_model" not in runtime


def test_bam_round_trip_fuzz():
    """Property fuzz: random reads survive BAM write->read bit-exactly."""
    from hypothesis import given, settings, strategies as st

    header = bam_lib.BamHeader(text="@HD\tVN:1.6", references=[("r0", 5000)])

    @settings(max_examples=40, deadline=None)
    @given(
        seq_len=st.integers(1, 300),
        flag=st.sampled_from([0, 4, 16]),
        n_ins=st.integers(0, 3),
        mapq=st.integers(0, 254),
        data=st.data(),
    )
    def check(seq_len, flag, n_ins, mapq, data):
        rng = np.random.default_rng(data.draw(st.integers(0, 2**31)))
        seq = "".join(rng.choice(list("ATCGN"), size=seq_len))
        if flag == 4:
            cig = []
        else:
            cig = [(0, seq_len)]
            for _ in range(n_ins):
                cig.append((1, int(rng.integers(1, 5))))
                seq += "".join(
                    rng.choice(list("ATCG"), size=cig[-1][1])
                )
        read = bam_lib.BamRead(
            qname="m0/1/0_%d" % len(seq), flag=flag,
            ref_id=-1 if flag == 4 else 0, pos=-1 if flag == 4 else 3,
            mapq=mapq, cigartuples=cig, seq=seq,
            query_qualities=rng.integers(0, 94, len(seq)),
            tags={
                "zm": 1,
                "pw": rng.integers(0, 256, len(seq)).astype(np.uint8),
                "sn": np.array([1.5, 2.5, 3.5, 4.5], np.float32),
                "rq": 0.99,
                "RG": "rg1",
            },
        )
        import io as _io
        import tempfile

        with tempfile.TemporaryDirectory() as td:
            p = os.path.join(td, "f.bam")
            with bam_lib.BamWriter(p, header) as w:
                w.write(read)
            got = list(bam_lib.BamReader(p))
        assert len(got) == 1
        g = got[0]
        assert g.qname == read.qname and g.flag == flag
        assert g.seq == seq and g.mapq == mapq
        assert g.cigartuples == cig
        np.testing.assert_array_equal(
            g.query_qualities, read.query_qualities
        )
        np.testing.assert_array_equal(g.get_tag("pw"), read.tags["pw"])
        np.testing.assert_allclose(g.get_tag("sn"), read.tags["sn"])
        assert abs(g.get_tag("rq") - 0.99) < 1e-6
        assert g.get_tag("RG") == "rg1"

    check()


def test_example_codec_fuzz():
    """Random feature dicts survive the tf.Example wire codec."""
    from hypothesis import given, settings, strategies as st

    keys = st.text(
        alphabet="abcdefghij_/", min_size=1, max_size=20
    )

    @settings(max_examples=60, deadline=None)
    @given(
        d=st.dictionaries(
            keys,
            st.one_of(
                st.tuples(st.just(example_codec.BYTES),
                          st.lists(st.binary(max_size=64), max_size=4)),
                st.tuples(st.just(example_codec.INT64),
                          st.lists(st.integers(-2**62, 2**62), max_size=8)),
                st.tuples(st.just(example_codec.FLOAT),
                          st.lists(st.floats(-1e9, 1e9, width=32),
                                   max_size=8)),
            ),
            max_size=6,
        )
    )
    def check(d):
        enc = example_codec.encode_example(d)
        dec = example_codec.decode_example(enc)
        assert set(dec) == set(d)
        for k, (kind, vals) in d.items():
            dkind, dvals = dec[k]
            assert dkind == kind
            if kind == example_codec.FLOAT:
                np.testing.assert_allclose(dvals, vals, rtol=1e-6)
            else:
                assert list(dvals) == list(vals)

    check()


def test_bam_header_larger_than_bgzf_block(tmp_path):
    """Real subreads_to_ccs.bam headers carry one reference per ZMW and
    exceed the 64 KB BGZF block payload; the header must span blocks."""
    refs = [(f"m0/{i}/ccs", 10000) for i in range(20000)]  # ~400 KB header
    header = bam_lib.BamHeader(text="@HD\tVN:1.6", references=refs)
    p = str(tmp_path / "big.bam")
    with bam_lib.BamWriter(p, header) as w:
        w.write(bam_lib.BamRead(
            qname="m0/19999/0_100", flag=0, ref_id=19999, pos=0, mapq=60,
            cigartuples=[(0, 100)], seq="A" * 100,
            query_qualities=[30] * 100, tags={"zm": 19999},
        ))
    rd = bam_lib.BamReader(p)
    assert len(rd.header.references) == 20000
    assert next(iter(rd)).reference_name == "m0/19999/ccs"


def test_stitch_min_quality_and_min_length_filters():
    """Reads below min_quality / min_length are filtered with the right
    outcome counters (stitch_utils.py:101-121)."""
    lowq = phred.quality_scores_to_string(np.full(50, 5))
    outs = [stitch.DCModelOutput(molecule_name="m", window_pos=0,
                                 sequence="A" * 50, quality_string=lowq)]
    counter = stitch.OutcomeCounter()
    assert stitch.stitch_to_fastq("m", outs, max_length=50, min_quality=20,
                                  min_length=1,
                                  outcome_counter=counter) is None
    assert counter.failed_quality_filter == 1
    okq = phred.quality_scores_to_string(np.full(50, 40))
    outs2 = [stitch.DCModelOutput(molecule_name="m", window_pos=0,
                                  sequence="A" * 50, quality_string=okq)]
    counter2 = stitch.OutcomeCounter()
    assert stitch.stitch_to_fastq("m", outs2, max_length=50, min_quality=20,
                                  min_length=100,
                                  outcome_counter=counter2) is None
    assert counter2.failed_length_filter == 1


def test_quick_inference_pool_matches_serial(tmp_path):
    """cpus=2 produces byte-identical FASTQ to cpus=0 (same model seed)."""
    import torch

    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=3, length=150, seed=33)
    outs = []
    for cpus in (0, 2):
        out = str(tmp_path / f"out{cpus}.fastq")
        torch.manual_seed(123)
        qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
               output=out,
               options=qi.InferenceOptions(batch_size=8, batch_zmws=2,
                                           cpus=cpus, min_quality=0,
                                           skip_windows_above=0),
               device="cpu")
        outs.append(open(out, "rb").read())
    assert outs[0] == outs[1]


def test_stitch_fill_n_for_missing_windows():
    """fill_n=True inserts N*max_length with EMPTY_QUAL for gaps
    (stitch_utils.py:61-74)."""
    outs = [
        stitch.DCModelOutput(molecule_name="m", window_pos=0,
                             sequence="AAAAA", quality_string="IIIII"),
        stitch.DCModelOutput(molecule_name="m", window_pos=10,
                             sequence="GGGGG", quality_string="IIIII"),
    ]
    seq, qual = stitch.get_full_sequence(outs, max_length=5, fill_n=True)
    assert seq == "AAAAA" + "N" * 5 + "GGGGG"
    assert qual[5:10] == phred.quality_scores_to_string(np.zeros(5))
    seq2, qual2 = stitch.get_full_sequence(outs, max_length=5)
    assert seq2 is None and qual2 == ""


def test_quick_inference_mixed_skip_and_model_windows(tmp_path):
    """A ZMW whose windows split between the model path and the CCS skip
    path still stitches into one complete read (the skipped predictions
    are appended out of order; _write_outputs must sort before grouping)."""
    from deepconsensus_amd.dcio import bam as bam_lib
    from deepconsensus_amd.dcio.fastq import read_fastq
    from deepconsensus_amd.inference import quick_inference as qi

    rng = np.random.default_rng(55)
    length = 250  # 3 windows per ZMW
    name = "m000/10/ccs"
    seq = _random_seq(rng, length)
    header = bam_lib.BamHeader(text="@HD\tVN:1.6", references=[(name, length)])
    sub = str(tmp_path / "subreads_to_ccs.bam")
    with bam_lib.BamWriter(sub, header) as w:
        for i in range(3):
            w.write(bam_lib.BamRead(
                qname=f"m000/10/{i * 300}_{i * 300 + length}",
                flag=16 if i % 2 else 0, ref_id=0, pos=0, mapq=60,
                cigartuples=[(0, length)], seq=seq,
                query_qualities=[30] * length,
                tags={"zm": 10,
                      "pw": rng.integers(0, 60, length).astype(np.uint8),
                      "ip": rng.integers(0, 60, length).astype(np.uint8),
                      "sn": np.array([6.0, 7.0, 5.5, 9.1], np.float32)},
            ))
    ccs = str(tmp_path / "ccs.bam")
    # CCS qualities: window 0 (0..99) high (skipped via skip_windows_above),
    # windows 1-2 low (routed through the model).
    quals = np.full(length, 10)
    quals[:100] = 60
    with bam_lib.BamWriter(ccs, header) as w:
        w.write(bam_lib.BamRead(
            qname=name, flag=4, ref_id=-1, pos=-1, cigartuples=[], seq=seq,
            query_qualities=quals,
            tags={"zm": 10, "ec": 9.5, "np": 3, "rq": 0.99, "RG": "rg0"},
        ))
    out = str(tmp_path / "out.fastq")
    counter = qi.run(
        subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random", output=out,
        options=qi.InferenceOptions(batch_size=8, batch_zmws=2, cpus=0,
                                    min_quality=0, skip_windows_above=45),
        device="cpu",
    )
    assert counter.success == 1
    recs = list(read_fastq(out))
    assert len(recs) == 1
    # Window 0 adopted the CCS sequence verbatim (skip path).
    assert recs[0].sequence[:100] == seq[:100]
    assert len(recs[0].sequence) >= 100


def test_quick_inference_stitch_modes_equivalent(tmp_path, monkeypatch):
    """DC_STITCH_MODE=pool produces byte-identical FASTQ and outcome
    counts to the serial default."""
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=4, length=150, seed=3)
    outs = {}
    counters = {}
    for mode in ("serial", "pool"):
        monkeypatch.setenv("DC_STITCH_MODE", mode)
        import torch

        torch.manual_seed(7)
        out = str(tmp_path / f"out_{mode}.fastq")
        options = qi.InferenceOptions(
            batch_size=64, batch_zmws=2, cpus=2, min_quality=0,
            skip_windows_above=0,
        )
        c = qi.run(subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
                   output=out, options=options)
        outs[mode] = open(out).read()
        counters[mode] = dataclasses_asdict(c)
    assert outs["serial"] == outs["pool"]
    assert counters["serial"] == counters["pool"]


def dataclasses_asdict(c):
    import dataclasses

    return dataclasses.asdict(c)
