"""Preprocessing tests mirroring the reference pre_lib_test case structure:
expand_clip_indent cigar cases, insertion trimming, multi-read spacing
(python and C++ paths), windowing, and feature extraction."""
import numpy as np
import pytest

from deepconsensus_amd.dcio.bam import BamRead
from deepconsensus_amd.preprocess import read as R
from deepconsensus_amd.preprocess.expand import expand_clip_indent, trim_insertions
from deepconsensus_amd.preprocess.windows import DcConfig, DcExample, dc_config_from_shape
from deepconsensus_amd.utils import constants

C = constants.CIGAR_OPS


def make_segment(bases, cigar, ip=None, pw=None, pos=0, reverse=False):
    """Builds a BamRead from a cigar string like '4M4I4M'."""
    cigartuples = []
    num = ""
    for ch in cigar:
        if ch.isdigit():
            num += ch
        else:
            cigartuples.append((C[ch], int(num)))
            num = ""
    n = len(bases)
    flag = 0x10 if reverse else 0
    tags = {
        "ip": np.array(ip if ip is not None else [0] * n, dtype=np.int64),
        "pw": np.array(pw if pw is not None else [0] * n, dtype=np.int64),
        "sn": np.array([0.1, 0.2, 0.3, 0.4], dtype=np.float32),
        "zm": 1,
    }
    return BamRead(
        qname="m/1/0_10", flag=flag, ref_id=0, pos=pos,
        cigartuples=cigartuples, seq=bases, query_qualities=[30] * n,
        tags=tags,
    )


EXPAND_CASES = [
    # (name, bases, cigar, ip, pw, expected_bases, expected_cigar_ops,
    #  expected_ip, expected_pw)
    ("match", "ATCG", "4M", None, None, "ATCG", [C["M"]] * 4, None, None),
    (
        "insertion", "AAAATTTTAAAA", "4M4I4M", [1] * 12, [2] * 12,
        "AAAATTTTAAAA",
        [C["M"]] * 4 + [C["I"]] * 4 + [C["M"]] * 4, [1] * 12, [2] * 12,
    ),
    (
        "deletion", "AAAAAAAA", "4M4D4M", [1] * 8, [2] * 4 + [0] * 4,
        "AAAA    AAAA",
        [C["M"]] * 4 + [C["D"]] * 4 + [C["M"]] * 4,
        [1] * 4 + [0] * 4 + [1] * 4, [2] * 4 + [0] * 8,
    ),
    (
        "skip_region", "AAAAAAAA", "4N8M", [1] * 8, [2] * 8,
        "    AAAAAAAA",
        [C["N"]] * 4 + [C["M"]] * 8, [0] * 4 + [1] * 8, [0] * 4 + [2] * 8,
    ),
    (
        "complex", "TTTTCGGAACTTGGGAAGGG", "5M5D5M5I5M", [1] * 20, [2] * 20,
        "TTTTC     GGAACTTGGGAAGGG",
        [C["M"]] * 5 + [C["D"]] * 5 + [C["M"]] * 5 + [C["I"]] * 5
        + [C["M"]] * 5,
        [1] * 5 + [0] * 5 + [1] * 15, [2] * 5 + [0] * 5 + [2] * 15,
    ),
    (
        "soft_clip", "AAAATTTTAAAA", "4S4M4S",
        [0] * 4 + [1] * 4 + [0] * 4, [0] * 4 + [2] * 4 + [0] * 4,
        "TTTT", [C["M"]] * 4, [1] * 4, [2] * 4,
    ),
    (
        "hard_clip", "TTTT", "4H4M4H", [1] * 4, [2] * 4,
        "TTTT", [C["M"]] * 4, [1] * 4, [2] * 4,
    ),
    (
        "eq_and_diff", "AAAATTTTAAAA", "4=4X4=", [1] * 12, [2] * 12,
        "AAAATTTTAAAA",
        [C["="]] * 4 + [C["X"]] * 4 + [C["="]] * 4, [1] * 12, [2] * 12,
    ),
]


@pytest.mark.parametrize(
    "name,bases,cigar,ip,pw,eb,ec,ei,ep", EXPAND_CASES,
    ids=[c[0] for c in EXPAND_CASES],
)
def test_expand_clip_indent(name, bases, cigar, ip, pw, eb, ec, ei, ep):
    seg = make_segment(bases, cigar, ip=ip, pw=pw)
    out = expand_clip_indent(seg)
    assert "".join(out.bases) == eb
    np.testing.assert_array_equal(out.cigar, ec)
    if ei is not None:
        np.testing.assert_array_equal(out.ip, ei)
    if ep is not None:
        np.testing.assert_array_equal(out.pw, ep)


def test_expand_clip_indent_indent():
    """Alignment starting at pos > 0 is gap-indented, cigar N-prefixed."""
    seg = make_segment("ATCG", "4M", pos=3)
    out = expand_clip_indent(seg)
    assert "".join(out.bases) == "   ATCG"
    np.testing.assert_array_equal(out.cigar, [C["N"]] * 3 + [C["M"]] * 4)
    np.testing.assert_array_equal(out.ccs_idx, [-1, -1, -1, 3, 4, 5, 6])


def test_expand_clip_indent_reverse_strand_flips_pw_ip():
    seg = make_segment("ATCG", "4M", ip=[1, 2, 3, 4], pw=[5, 6, 7, 8],
                       reverse=True)
    out = expand_clip_indent(seg)
    assert out.strand == constants.Strand.REVERSE
    np.testing.assert_array_equal(out.ip, [4, 3, 2, 1])
    np.testing.assert_array_equal(out.pw, [8, 7, 6, 5])


def test_trim_insertions():
    # 5M5I5M with ins_trim=4: the 5-bp insertion is removed entirely.
    seg = make_segment("AAAAATTTTTGGGGG", "5M5I5M", ip=list(range(15)),
                       pw=list(range(15)))
    import collections

    counter = collections.Counter()
    out = trim_insertions(seg, ins_trim=4, counter=counter)
    assert out.seq == "AAAAAGGGGG"
    assert out.cigartuples == [(C["M"], 5), (C["M"], 5)]
    np.testing.assert_array_equal(
        out.get_tag("pw"), list(range(5)) + list(range(10, 15))
    )
    assert counter["zmw_trimmed_insertions"] == 1
    assert counter["zmw_trimmed_insertions_bp"] == 5


def _spacing_reads(bases_rows, cigar_rows, ccs_idx=None, truth_range=None):
    reads = []
    n = len(bases_rows)
    for i, (bases, cigar) in enumerate(zip(bases_rows, cigar_rows)):
        cig = np.array([C[x] for x in cigar], dtype=np.uint8)
        idx = (
            np.array(ccs_idx[i]) if ccs_idx else np.arange(len(bases))
        )
        reads.append(
            R.Read(
                name="",
                bases=np.array(list(bases), dtype="<U1"),
                cigar=cig,
                ip=np.zeros(len(bases), dtype=np.uint8),
                pw=np.zeros(len(bases), dtype=np.uint8),
                sn=np.zeros(4),
                strand=constants.Strand.UNKNOWN,
                ccs_idx=idx,
                truth_range=truth_range if i == n - 1 else None,
            )
        )
    return reads


SPACING_CASES = [
    # (bases rows, cigar rows, expected spaced rows, ccs_idx, truth_range)
    (["AAAA", "AAAA"], ["MMMM", "MMMM"], ["AAAA", "AAAA"], None, None),
    (["ACTA", "ACTAG"], ["MMMM", "MMMMM"], ["ACTA", "ACTAG"], None, None),
    (["ACTG", "ACTAG"], ["MMMM", "MMMIM"], ["ACT G", "ACTAG"], None, None),
    (["ACTGG", "ACT G"], ["MMMMM", "MMMDM"], ["ACTGG", "ACT G"], None, None),
    (
        ["TTTTT", "TTTTT", "TTTTT"],
        ["MIMIM", "MMMMM", "MIMIM"],
        ["TTTTT", "T T TTT", "TTTTT"],
        None, None,
    ),
    (
        ["TTTTT", "TTTTT", "TTTTT"],
        ["MIIIM", "MMMMM", "MIIIM"],
        ["TTTTT", "T   TTTT", "TTTTT"],
        None, None,
    ),
    (
        ["TTTTT", "TTTTT", "TTTTT", "TTGGGTTT"],
        ["MMMMM", "MMMMM", "MMMMM", "MMIIIMMM"],
        ["TTTTT", "TTTTT", "TTTTT", "TTGGGTTT"],
        [
            [0, 1, 2, 3, 4],
            [0, 1, 2, 3, 4],
            [0, 1, 2, 3, 4],
            [0, 1, 2, -1, -1, -1, 3, 4],
        ],
        {"contig": "chr1", "begin": 0, "end": 8},
    ),
    (
        ["TTTTT", "TTTTT", "TTTTT", "TTTTTGG"],
        ["MMMMM", "MMMMM", "MMMMM", "MMMMMII"],
        ["TTTTT", "TTTTT", "TTTTT", "TTTTTGG"],
        [
            [0, 1, 2, 3, 4],
            [0, 1, 2, 3, 4],
            [0, 1, 2, 3, 4],
            [0, 1, 2, 3, 4, -1, -1],
        ],
        {"contig": "chr1", "begin": 0, "end": 7},
    ),
]


@pytest.mark.parametrize("force_python", [True, False],
                         ids=["python", "cpp"])
@pytest.mark.parametrize("case", range(len(SPACING_CASES)))
def test_space_out_subreads(case, force_python):
    bases, cigars, expected, ccs_idx, truth_range = SPACING_CASES[case]
    reads = _spacing_reads(bases, cigars, ccs_idx, truth_range)
    spaced = R.space_out_subreads(reads, force_python=force_python)
    got = ["".join(r.bases).rstrip() for r in spaced]
    assert got == expected


def test_spacing_cpp_matches_python_random():
    """Fuzz: C++ and Python spacing agree on random cigar mixes."""
    rng = np.random.default_rng(0)
    for trial in range(10):
        n_reads = int(rng.integers(2, 6))
        rows, cigs = [], []
        for _ in range(n_reads):
            n = int(rng.integers(5, 40))
            ops = rng.choice(list("MMMMID"), size=n)
            # bases exist for M and I, not D... keep all; just mirror shapes.
            cigs.append("".join(ops))
            rows.append("".join(rng.choice(list("ATCG "), size=n)))
        r1 = _spacing_reads(rows, cigs)
        r2 = _spacing_reads(rows, cigs)
        s1 = R.space_out_subreads(r1, force_python=True)
        s2 = R.space_out_subreads(r2, force_python=False)
        for a, b in zip(s1, s2):
            np.testing.assert_array_equal(a.bases, b.bases)
            np.testing.assert_array_equal(a.ccs_idx, b.ccs_idx)


def test_dc_config():
    cfg = DcConfig(20, 100)
    assert cfg.tensor_height == 85
    assert cfg.indices("bases", 5) == slice(0, 5)
    assert cfg.indices("pw", 30) == slice(20, 40)
    assert cfg.indices("ccs") == slice(80, 81)
    assert cfg.indices("sn") == slice(81, 85)
    cfg_bq = DcConfig(20, 100, use_ccs_bq=True)
    assert cfg_bq.tensor_height == 86
    assert cfg_bq.indices("ccs_bq") == slice(81, 82)
    assert cfg_bq.indices("sn") == slice(82, 86)


def test_dc_config_from_shape():
    cfg = dc_config_from_shape((85, 100, 1))
    assert cfg.max_passes == 20 and cfg.max_length == 100
    cfg = dc_config_from_shape((86, 100, 1), use_ccs_bq=True)
    assert cfg.max_passes == 20
    with pytest.raises(ValueError):
        dc_config_from_shape((87, 100, 1))


def _make_zmw(n_subreads=3, length=250, seed=0):
    """Builds a spaced DcExample resembling one ZMW."""
    rng = np.random.default_rng(seed)
    seq = "".join(rng.choice(list("ATCG"), size=length))
    reads = []
    for i in range(n_subreads):
        seg = make_segment(
            seq, f"{length}M", ip=[1] * length, pw=[2] * length
        )
        seg.qname = f"m/7/{i}"
        reads.append(expand_clip_indent(seg))
    ccs = R.Read(
        name="m/7/ccs",
        bases=np.array(list(seq), dtype="<U1"),
        cigar=np.repeat(np.uint8(C["M"]), length),
        pw=np.zeros(length, dtype=np.uint8),
        ip=np.zeros(length, dtype=np.uint8),
        sn=np.zeros(4),
        strand=constants.Strand.UNKNOWN,
        base_quality_scores=np.full(length, 30),
        ccs_idx=np.arange(length),
        ec=12.0, np_num_passes=10, rq=0.999, rg="rg1",
    )
    reads.append(ccs)
    spaced = R.space_out_subreads(reads)
    return DcExample("m/7/ccs", spaced, DcConfig(20, 100))


def _make_zmw_with_insertions(length=250, seed=3, use_ccs_bq=False,
                              window_widths=None):
    """Spaced DcExample whose subreads carry insertions (real gap
    columns) and varied strands/pw/ip."""
    rng = np.random.default_rng(seed)
    seq = "".join(rng.choice(list("ATCG"), size=length))
    reads = []
    for i in range(4):
        ins_pos = int(rng.integers(1, length - 1))
        ins_len = int(rng.integers(1, 4))
        ins = "".join(rng.choice(list("ATCG"), size=ins_len))
        full = seq[:ins_pos] + ins + seq[ins_pos:]
        cig = f"{ins_pos}M{ins_len}I{length - ins_pos}M"
        n = len(full)
        seg = make_segment(
            full, cig,
            ip=rng.integers(0, 60, n).tolist(),
            pw=rng.integers(0, 60, n).tolist(),
            reverse=bool(i % 2),
        )
        seg.qname = f"m/7/{i}"
        reads.append(expand_clip_indent(seg))
    ccs = R.Read(
        name="m/7/ccs",
        bases=np.array(list(seq), dtype="<U1"),
        cigar=np.repeat(np.uint8(C["M"]), length),
        pw=np.zeros(length, dtype=np.uint8),
        ip=np.zeros(length, dtype=np.uint8),
        sn=np.zeros(4),
        strand=constants.Strand.UNKNOWN,
        base_quality_scores=rng.integers(10, 50, length).astype(np.int16),
        ccs_idx=np.arange(length),
        ec=12.0, np_num_passes=10, rq=0.999, rg="rg1",
    )
    reads.append(ccs)
    spaced = R.space_out_subreads(reads)
    return DcExample("m/7/ccs", spaced, DcConfig(20, 100, use_ccs_bq),
                     window_widths=window_widths)


@pytest.mark.parametrize("case", [
    "plain", "ccs_bq", "smart_overflow", "smart_mixed",
])
def test_iter_feature_dicts_matches_slow(case):
    """The vectorized inference fast path emits dicts identical to the
    per-window iter_examples()/to_features_dict() path, counters too.
    (CCS-only ZMWs are not covered: the per-window path itself rejects
    n_subreads=0, and the grouper never yields such a ZMW.)"""
    kwargs = {}
    if case == "ccs_bq":
        kwargs["use_ccs_bq"] = True
    elif case == "smart_overflow":
        kwargs["window_widths"] = np.array([120, 80, 50])
    elif case == "smart_mixed":
        kwargs["window_widths"] = np.array([90, 150, 10])
    ex_fast = _make_zmw_with_insertions(**kwargs)
    ex_slow = _make_zmw_with_insertions(**kwargs)
    fast = list(ex_fast.iter_feature_dicts())
    slow = [x.to_features_dict() for x in ex_slow.iter_examples()]
    assert ex_fast.counter == ex_slow.counter
    assert len(fast) == len(slow) and fast
    for f, s in zip(fast, slow):
        assert set(f) == set(s)
        for key in s:
            if isinstance(s[key], np.ndarray):
                assert f[key].shape == s[key].shape, key
                np.testing.assert_array_equal(f[key], s[key], err_msg=key)
            else:
                assert f[key] == s[key], key


@pytest.mark.parametrize("case", ["plain", "smart_overflow"])
def test_iter_feature_dicts_formatted_matches_format_rows(case):
    """Fast path with clipping/int16 == slow dicts + format_rows +
    astype(int16) (the worker's previous finishing steps)."""
    from deepconsensus_amd.models import data as data_lib
    from deepconsensus_amd.models.config import Params

    kwargs = {}
    if case == "smart_overflow":
        kwargs["window_widths"] = np.array([120, 80, 50])
    ex_fast = _make_zmw_with_insertions(**kwargs)
    ex_slow = _make_zmw_with_insertions(**kwargs)
    fast = list(ex_fast.iter_feature_dicts(
        pw_max=255, ip_max=255, sn_max=500, out_dtype=np.int16
    ))
    fmt_params = Params(
        max_passes=20, use_ccs_bq=False,
        total_rows=ex_slow.config.tensor_height,
        PW_MAX=255, IP_MAX=255, SN_MAX=500,
    )
    slow = [x.to_features_dict() for x in ex_slow.iter_examples()]
    assert len(fast) == len(slow) and fast
    for f, s in zip(fast, slow):
        assert f["fmt"] is True
        want = data_lib.format_rows(s["subreads"], fmt_params).astype(
            np.int16
        )
        assert f["subreads"].dtype == np.int16
        np.testing.assert_array_equal(f["subreads"], want)


def test_dc_example_windows_and_features():
    ex = _make_zmw(length=250)
    assert ex.n_subreads == 3
    windows = list(ex.iter_examples())
    assert len(windows) == 3  # 250 bp -> 100+100+50
    for w in windows:
        feat = w.extract_features()
        assert feat.shape == (85, 100, 1)
    # Window positions increase.
    pos = [w.to_features_dict()["window_pos"] for w in windows]
    assert pos == sorted(pos)
    assert pos[0] == 0
    # Feature content: CCS row matches the sequence encoding.
    f0 = windows[0].extract_features()[:, :, 0]
    ccs_row = f0[80]
    assert ccs_row.max() <= 4 and ccs_row.min() >= 0
    # sn rows constant per-row (values from the sn tag).
    np.testing.assert_allclose(f0[81], 0.1, atol=1e-6)
    np.testing.assert_allclose(f0[84], 0.4, atol=1e-6)


def test_dc_example_tf_example_round_trip():
    from deepconsensus_amd.preprocess.windows import tf_example_to_features_dict

    ex = _make_zmw(length=120)
    w = next(ex.iter_examples())
    serialized = w.tf_example()
    feats = tf_example_to_features_dict(serialized, inference=True)
    assert feats["name"] == "m/7/ccs"
    assert feats["subreads/shape"] == [85, 100, 1]
    np.testing.assert_array_equal(
        feats["subreads"], w.extract_features()
    )
    assert feats["subreads/num_passes"] == 3


def test_dc_config_from_shape_invalid_raises():
    """Decoding an 86-row (ccs_bq) tensor with use_ccs_bq=False raises
    (preprocess_test.py::test_invalid_tf_examples semantics)."""
    from deepconsensus_amd.preprocess.windows import dc_config_from_shape

    ok = dc_config_from_shape((86, 100, 1), use_ccs_bq=True)
    assert ok.max_passes == 20
    with pytest.raises(ValueError, match="Invalid subreads shape"):
        dc_config_from_shape((86, 100, 1), use_ccs_bq=False)


def test_expand_native_matches_python(tmp_path):
    """C++ expand_read must reproduce the python expand_clip_indent
    field-for-field on reads with soft clips, indels, indent and
    reverse strand."""
    import numpy as np

    from deepconsensus_amd.dcio.bam import BamRead
    from deepconsensus_amd.preprocess import expand as expand_lib

    ext = expand_lib._spacing_ext()
    assert ext, "spacing extension must build in CI"

    rng = np.random.default_rng(5)

    def mk(cigar, pos=0, flag=0):
        qlen = sum(n for op, n in cigar if op in (0, 1, 4, 7, 8))
        seq = "".join(rng.choice(list("ATCG"), size=qlen))
        return BamRead(
            qname="q", flag=flag, ref_id=0, pos=pos, mapq=60,
            cigartuples=cigar, seq=seq,
            query_qualities=[30] * qlen,
            tags={
                "zm": 1,
                "pw": rng.integers(0, 300, qlen),
                "ip": rng.integers(0, 300, qlen),
                "sn": np.array([6.0, 7.0, 5.5, 9.1], np.float32),
            },
        )

    cases = [
        ([(0, 50)], 0, 0),
        ([(0, 30), (1, 3), (0, 10), (2, 4), (0, 20)], 0, 0),
        ([(4, 7), (0, 40), (4, 5)], 0, 0),
        ([(5, 9), (4, 3), (0, 25), (1, 2), (2, 2), (0, 10), (4, 6),
          (5, 2)], 12, 0),
        ([(0, 30), (2, 5), (0, 30)], 4, 16),          # reverse strand
        ([(4, 4), (0, 20), (1, 5), (0, 15), (4, 8)], 9, 16),
    ]
    for cigar, pos, flag in cases:
        read = mk(cigar, pos, flag)
        nat = expand_lib._expand_native(read, ext)
        # Force the python path by monkeying the ext lookup.
        saved = expand_lib._SPACING
        expand_lib._SPACING = False
        try:
            ref = expand_lib.expand_clip_indent(read)
        finally:
            expand_lib._SPACING = saved
        np.testing.assert_array_equal(nat.bases, ref.bases,
                                      err_msg=str(cigar))
        np.testing.assert_array_equal(nat.cigar, ref.cigar)
        np.testing.assert_array_equal(nat.pw, ref.pw)
        np.testing.assert_array_equal(nat.ip, ref.ip)
        np.testing.assert_array_equal(nat.ccs_idx, ref.ccs_idx)
        np.testing.assert_array_equal(nat.sn, ref.sn)
        assert nat.strand == ref.strand
