"""Extended fuzz/soak suites (opt-in: pytest --run-soak -m soak).

These re-run the session's long verification sweeps: spacing C++/python
parity over random cigar mixes, DatasetIterator sharding invariants, and
randomized end-to-end pipeline configurations."""
import json
import os
import sys
import tempfile
from pathlib import Path

import numpy as np
import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.soak
def test_soak_spacing_parity_1000():
    from test_preprocess import _spacing_reads

    from deepconsensus_amd.preprocess import read as R

    rng = np.random.default_rng(777)
    for trial in range(1000):
        n_reads = int(rng.integers(2, 8))
        rows, cigs = [], []
        for _ in range(n_reads):
            n = int(rng.integers(3, 80))
            cigs.append("".join(rng.choice(list("MMMMMIDIS"), size=n)))
            rows.append("".join(rng.choice(list("ATCG "), size=n)))
        s1 = R.space_out_subreads(_spacing_reads(rows, cigs),
                                  force_python=True)
        s2 = R.space_out_subreads(_spacing_reads(rows, cigs),
                                  force_python=False)
        for a, b in zip(s1, s2):
            np.testing.assert_array_equal(a.bases, b.bases)
            np.testing.assert_array_equal(a.ccs_idx, b.ccs_idx)
            np.testing.assert_array_equal(a.pw, b.pw)


@pytest.mark.soak
def test_soak_dataset_iterator_invariants():
    from test_data import _make_example, _params

    from deepconsensus_amd.dcio import tfrecord
    from deepconsensus_amd.models import data as data_lib

    p = _params()
    rng = np.random.default_rng(4)
    for trial in range(60):
        n = int(rng.integers(1, 40))
        bs = int(rng.integers(1, 9))
        world = int(rng.choice([1, 2, 3]))
        shuffle = bool(rng.integers(0, 2))
        drop = bool(rng.integers(0, 2))
        limit = int(rng.choice([-1, 0, 5, 17]))
        with tempfile.TemporaryDirectory() as td:
            path = os.path.join(td, "d.tfrecord.gz")
            with tfrecord.TFRecordWriter(path, compression="gzip") as w:
                for i in range(n):
                    enc, _ = _make_example(p, i)
                    w.write(enc)
            seen = []
            for rank in range(world):
                it = data_lib.DatasetIterator(
                    [path], p, batch_size=bs, shuffle=shuffle, seed=trial,
                    rank=rank, world_size=world, limit=limit,
                    drop_remainder=drop,
                )
                for b in it.iterate(epoch=trial % 3):
                    assert b["rows"].shape[0] <= bs
                    if drop:
                        assert b["rows"].shape[0] == bs
                    seen.extend(b["window_pos"].tolist())
            assert len(seen) == len(set(seen)), trial
            assert set(seen) <= {i * 100 for i in range(n)}, trial


@pytest.mark.soak
def test_soak_pipeline_random_configs():
    from test_io_and_pipeline import make_test_bams

    from deepconsensus_amd.dcio.fastq import read_fastq
    from deepconsensus_amd.inference import quick_inference as qi

    rng = np.random.default_rng(9)
    for trial in range(25):
        n_zmws = int(rng.integers(1, 6))
        length = int(rng.integers(40, 500))
        n_sub = int(rng.integers(1, 9))
        cpus = int(rng.choice([0, 2]))
        skip = int(rng.choice([0, 30, 45]))
        with tempfile.TemporaryDirectory() as td:
            sub, ccs = make_test_bams(Path(td), n_zmws=n_zmws,
                                      length=length, n_subreads=n_sub,
                                      seed=trial)
            out = os.path.join(td, "o.fastq")
            torch.manual_seed(trial)
            c = qi.run(
                subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
                output=out,
                options=qi.InferenceOptions(
                    batch_size=16, batch_zmws=2, cpus=cpus, min_quality=0,
                    skip_windows_above=skip,
                ),
                device="cpu",
            )
            assert c.total == n_zmws, (trial, c)
            for r in read_fastq(out):
                assert len(r.sequence) == len(r.quality) > 0
            stats = json.load(open(os.path.join(td, "o.inference.json")))
            assert stats["n_zmw_processed"] == n_zmws


@pytest.mark.soak
def test_soak_fast_featurizer_parity():
    """Randomized ZMWs: iter_feature_dicts (both plain and fmt/int16
    modes) vs the per-window path, including smart windows."""
    from test_preprocess import _make_zmw_with_insertions

    from deepconsensus_amd.models import data as data_lib
    from deepconsensus_amd.models.config import Params

    rng = np.random.default_rng(4242)
    for trial in range(150):
        length = int(rng.integers(40, 400))
        use_bq = bool(rng.integers(0, 2))
        window_widths = None
        if rng.integers(0, 2):
            # Random smart widths summing to length, some > max_length.
            widths = []
            left = length
            while left > 0:
                w = int(min(left, rng.integers(10, 160)))
                widths.append(w)
                left -= w
            window_widths = np.array(widths)
        kwargs = dict(length=length, seed=trial, use_ccs_bq=use_bq,
                      window_widths=window_widths)
        fast = list(_make_zmw_with_insertions(**kwargs).iter_feature_dicts())
        ex_slow = _make_zmw_with_insertions(**kwargs)
        slow = [x.to_features_dict() for x in ex_slow.iter_examples()]
        assert len(fast) == len(slow)
        for f, s in zip(fast, slow):
            for key in s:
                if isinstance(s[key], np.ndarray):
                    np.testing.assert_array_equal(f[key], s[key],
                                                  err_msg=key)
                else:
                    assert f[key] == s[key], key
        # fmt/int16 mode vs format_rows over the slow dicts.
        fast_fmt = list(_make_zmw_with_insertions(**kwargs).iter_feature_dicts(
            pw_max=255, ip_max=255, sn_max=500, out_dtype=np.int16
        ))
        fmt_params = Params(max_passes=20, use_ccs_bq=use_bq,
                            total_rows=ex_slow.config.tensor_height,
                            PW_MAX=255, IP_MAX=255, SN_MAX=500)
        for f, s in zip(fast_fmt, slow):
            want = data_lib.format_rows(s["subreads"], fmt_params)
            np.testing.assert_array_equal(
                f["subreads"], want.astype(np.int16)
            )
