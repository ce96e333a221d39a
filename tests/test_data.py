"""Data-provider tests (reference data_providers_test.py counterpart).

Covers format_rows clipping/assembly, process_input parse semantics
(incl. remove_label_gaps), and DatasetIterator shuffling/sharding/limit
behavior on real TFRecord files written by our codec.
"""
import numpy as np
import pytest

from deepconsensus_amd.dcio import example_codec, tfrecord
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import data as data_lib


def _params(use_ccs_bq=False):
    p = cfg.get_config("transformer_learn_values+custom")
    p.use_ccs_bq = use_ccs_bq
    cfg.modify_params(p)
    return p


def test_format_rows_clips_and_assembles():
    p = _params()
    R, L, mp = p.total_rows, 7, p.max_passes
    sub = np.zeros((R, L), dtype=np.float32)
    sub[0, 0] = 4  # base row untouched
    sub[mp, 0] = 999  # PW row -> clip to PW_MAX
    sub[2 * mp, 0] = 777  # IP row -> clip to IP_MAX
    sub[3 * mp, 0] = 2  # strand untouched
    sub[-1, :] = 1e6  # SN row -> clip to SN_MAX
    rows = data_lib.format_rows(sub, p)
    assert rows.shape == (R, L)
    assert rows[0, 0] == 4
    assert rows[mp, 0] == p.PW_MAX == 255
    assert rows[2 * mp, 0] == p.IP_MAX == 255
    assert rows[3 * mp, 0] == 2
    assert rows[-1, 0] == p.SN_MAX == 500


def test_format_rows_ccs_bq_height():
    p = _params(use_ccs_bq=True)
    sub = np.zeros((p.total_rows, 5), dtype=np.float32)
    rows = data_lib.format_rows(sub, p)
    assert rows.shape[0] == p.total_rows == 86


def _make_example(p, seed, label_with_gaps=False):
    rng = np.random.default_rng(seed)
    R, L = p.total_rows, p.max_length
    sub = rng.integers(0, 5, size=(R, L, 1)).astype(np.float32)
    if label_with_gaps:
        label = np.array([0, 3, 0, 1, 2] + [0] * (L - 5), np.float32)
    else:
        label = rng.integers(0, 5, size=(L,)).astype(np.float32)
    feats = {
        "subreads/encoded": (example_codec.BYTES, [sub.tobytes()]),
        "subreads/shape": (example_codec.INT64, [R, L, 1]),
        "subreads/num_passes": (example_codec.INT64, [7]),
        "name": (example_codec.BYTES, [f"m0/{seed}/ccs".encode()]),
        "window_pos": (example_codec.INT64, [seed * 100]),
        "label/encoded": (example_codec.BYTES, [label.tobytes()]),
        "label/shape": (example_codec.INT64, [L]),
        "ccs_base_quality_scores": (example_codec.INT64, [30] * L),
    }
    return example_codec.encode_example(feats), label


def test_process_input_train_and_label_shift():
    p = _params()
    enc, label = _make_example(p, 3, label_with_gaps=True)
    out = data_lib.process_input(enc, p, inference=False)
    assert out["rows"].shape == (p.total_rows, p.max_length, 1)
    assert out["window_pos"] == 300
    np.testing.assert_array_equal(out["label"], label)
    # remove_label_gaps: internal gaps removed, left-shifted.
    p2 = _params()
    p2.remove_label_gaps = True
    out2 = data_lib.process_input(enc, p2, inference=False)
    assert list(out2["label"][:3]) == [3, 1, 2]
    assert not out2["label"][3:].any()


def test_process_input_inference_has_no_label():
    p = _params()
    enc, _ = _make_example(p, 4)
    out = data_lib.process_input(enc, p, inference=True)
    assert "label" not in out
    assert out["ccs_base_quality_scores"].shape == (p.max_length,)


def _write_dataset(tmp_path, p, n, name="d.tfrecord.gz"):
    path = str(tmp_path / name)
    with tfrecord.TFRecordWriter(path, compression="gzip") as w:
        for i in range(n):
            enc, _ = _make_example(p, i)
            w.write(enc)
    return path


def test_dataset_iterator_sharding_disjoint_and_complete(tmp_path):
    p = _params()
    path = _write_dataset(tmp_path, p, 24)
    seen = []
    for rank in (0, 1):
        it = data_lib.DatasetIterator(
            [path], p, batch_size=4, shuffle=False, rank=rank, world_size=2
        )
        for b in it.iterate():
            seen.extend(b["window_pos"].tolist())
    assert sorted(seen) == [i * 100 for i in range(24)]
    assert len(set(seen)) == 24


def test_dataset_iterator_shuffle_deterministic_and_epoch_varies(tmp_path):
    p = _params()
    p.buffer_size = 8
    path = _write_dataset(tmp_path, p, 20)

    def order(epoch, seed=5):
        it = data_lib.DatasetIterator([path], p, batch_size=5, seed=seed)
        return [w for b in it.iterate(epoch) for w in b["window_pos"]]

    assert order(0) == order(0)  # deterministic per (seed, epoch)
    assert order(0) != order(1)  # varies across epochs
    assert sorted(order(0)) == [i * 100 for i in range(20)]


def test_dataset_iterator_drop_remainder_and_limit(tmp_path):
    p = _params()
    path = _write_dataset(tmp_path, p, 10)
    it = data_lib.DatasetIterator([path], p, batch_size=4, shuffle=False)
    assert sum(1 for _ in it.iterate()) == 2  # 10//4, remainder dropped
    it2 = data_lib.DatasetIterator(
        [path], p, batch_size=4, shuffle=False, drop_remainder=False
    )
    batches = list(it2.iterate())
    assert [b["rows"].shape[0] for b in batches] == [4, 4, 2]
    assert batches[0]["rows"].shape[1:] == (p.total_rows, p.max_length, 1)
    it3 = data_lib.DatasetIterator(
        [path], p, batch_size=2, shuffle=False, limit=4
    )
    assert sum(1 for _ in it3.iterate()) == 2
    assert it3.count_examples() == 10


def test_dataset_iterator_missing_files_raises(tmp_path):
    p = _params()
    with pytest.raises(FileNotFoundError):
        data_lib.DatasetIterator([str(tmp_path / "none-*.gz")], p, 4)
