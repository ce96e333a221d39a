"""Data providers: tf.Example -> model tensors, dataset iteration.

Behavioral parity with reference data_providers.py: the proto schema
(:41-58), per-feature row indices (get_indices, via models.config),
format_rows value clipping (:128-184), per-example parsing (:249-297), and
batched shuffled repeated iteration (:307-425) — re-implemented on numpy +
the in-repo TFRecord/Example codec, with per-rank sharding for DP training.
"""
from __future__ import annotations

import glob as globlib
import random
from typing import Dict, Iterator, List

import numpy as np

from deepconsensus_amd.dcio import example_codec, tfrecord
from deepconsensus_amd.models.config import Params, get_indices
from deepconsensus_amd.utils import constants, phred


def format_rows(subreads: np.ndarray, params: Params) -> np.ndarray:
    """Clips PW/IP/SN rows and reassembles the input matrix
    (data_providers.py:128-184)."""
    (
        base_idx, pw_idx, ip_idx, strand_idx, ccs_idx, ccs_bq_idx, sn_idx
    ) = get_indices(params.max_passes, params.use_ccs_bq)
    base_rows = subreads[slice(*base_idx)]
    pw_rows = subreads[slice(*pw_idx)]
    ip_rows = subreads[slice(*ip_idx)]
    strand_rows = subreads[slice(*strand_idx)]
    ccs_rows = subreads[slice(*ccs_idx)]
    ccs_bq_rows = subreads[slice(*ccs_bq_idx)]
    sn_rows = subreads[slice(*sn_idx)]
    if params.PW_MAX:
        pw_rows = np.clip(pw_rows, 0, params.PW_MAX)
    if params.IP_MAX:
        ip_rows = np.clip(ip_rows, 0, params.IP_MAX)
    if params.SN_MAX:
        sn_rows = np.clip(sn_rows, 0, params.SN_MAX)
    if params.use_ccs_bq:
        feats = [base_rows, pw_rows, ip_rows, strand_rows, ccs_rows,
                 ccs_bq_rows, sn_rows]
    else:
        feats = [base_rows, pw_rows, ip_rows, strand_rows, ccs_rows,
                 sn_rows]
    rows = np.concatenate(feats, axis=0)
    assert rows.shape[0] == params.total_rows, rows.shape
    return rows


def remove_internal_gaps_and_shift(label: np.ndarray) -> np.ndarray:
    """Gap-removing left shift (data_providers.py:116-125)."""
    label = np.squeeze(label)
    return phred.left_shift_seq(label)


def process_feature_dict(
    features: Dict[str, object], params: Params
) -> Dict[str, object]:
    """Inference-side feature prep (data_providers.py:187-246)."""
    subreads = features["subreads"]
    rows = format_rows(np.asarray(subreads), params)
    return {
        "rows": rows,
        "label": np.array([]),
        "num_passes": features["subreads/num_passes"],
        "window_pos": features["window_pos"],
        "name": features["name"],
        "ccs_base_quality_scores": features["ccs_base_quality_scores"],
        "ec": features["ec"],
        "np_num_passes": features["np_num_passes"],
        "rq": features["rq"],
        "rg": features["rg"],
    }


def process_input(
    serialized: bytes, params: Params, inference: bool
) -> Dict[str, np.ndarray]:
    """Parses one serialized tf.Example (data_providers.py:249-297)."""
    decoded = example_codec.decode_example(serialized)
    shape = list(decoded["subreads/shape"][1])
    flat = np.frombuffer(
        decoded["subreads/encoded"][1][0], dtype=constants.NP_DATA_TYPE
    )
    subreads = flat.reshape(shape)
    num_passes = float(decoded["subreads/num_passes"][1][0])
    out: Dict[str, np.ndarray] = {}
    if not inference:
        label = np.frombuffer(
            decoded["label/encoded"][1][0], dtype=constants.NP_DATA_TYPE
        ).reshape(list(decoded["label/shape"][1]))
        if params.get("remove_label_gaps"):
            label = remove_internal_gaps_and_shift(label)
        out["label"] = label.astype(constants.NP_DATA_TYPE)
    rows = format_rows(subreads, params)
    out["rows"] = rows
    out["num_passes"] = np.float32(num_passes)
    out["window_pos"] = np.int64(decoded["window_pos"][1][0])
    out["name"] = decoded["name"][1][0]
    out["ccs_base_quality_scores"] = np.array(
        decoded["ccs_base_quality_scores"][1], dtype=np.int64
    )
    return out


def get_total_rows(max_passes: int, use_ccs_bq: bool) -> int:
    from deepconsensus_amd.models.config import get_total_rows as f

    return f(max_passes, use_ccs_bq)


class DatasetIterator:
    """Batched, shuffled, repeated, rank-sharded TFRecord dataset.

    Mirrors data_providers.get_dataset/create_input_fn semantics
    (data_providers.py:307-425) without TF: file-level interleave with a
    record-level shuffle buffer; shards records across DP ranks by
    round-robin (record_index % world_size == rank).
    """

    def __init__(
        self,
        file_patterns: List[str],
        params: Params,
        batch_size: int,
        inference: bool = False,
        shuffle: bool = True,
        seed: int = 1,
        rank: int = 0,
        world_size: int = 1,
        limit: int = -1,
        drop_remainder: bool = True,
    ):
        if isinstance(file_patterns, str):
            file_patterns = [file_patterns]
        self.files: List[str] = []
        for p in file_patterns:
            self.files.extend(sorted(globlib.glob(p)))
        if not self.files:
            raise FileNotFoundError(f"no files match {file_patterns}")
        self.params = params
        self.batch_size = batch_size
        self.inference = inference
        self.shuffle = shuffle
        self.seed = seed
        self.rank = rank
        self.world_size = world_size
        self.limit = limit
        self.drop_remainder = drop_remainder
        self.buffer_size = min(int(params.get("buffer_size", 1000)), 100_000)

    def _record_stream(self, epoch: int) -> Iterator[bytes]:
        files = list(self.files)
        rng = random.Random(self.seed + epoch)
        if self.shuffle:
            rng.shuffle(files)
        idx = 0
        for f in files:
            for rec in tfrecord.read_tfrecords(f):
                if idx % self.world_size == self.rank:
                    yield rec
                idx += 1

    def __iter__(self):
        return self.iterate(epoch=0)

    def iterate(self, epoch: int = 0) -> Iterator[Dict[str, np.ndarray]]:
        """One epoch of batches."""
        rng = random.Random(self.seed * 7919 + epoch)
        buffer: List[bytes] = []
        batch: List[Dict[str, np.ndarray]] = []
        n_yielded = 0

        def emit(example_bytes):
            nonlocal batch, n_yielded
            ex = process_input(example_bytes, self.params, self.inference)
            batch.append(ex)
            if len(batch) == self.batch_size:
                out = self._collate(batch)
                batch = []
                return out
            return None

        for rec in self._record_stream(epoch):
            if self.limit >= 0 and n_yielded * self.batch_size >= self.limit:
                break
            if self.shuffle:
                if len(buffer) < self.buffer_size:
                    buffer.append(rec)
                    continue
                j = rng.randrange(len(buffer))
                rec, buffer[j] = buffer[j], rec
            b = emit(rec)
            if b is not None:
                n_yielded += 1
                yield b
        if self.shuffle:
            rng.shuffle(buffer)
            for rec in buffer:
                b = emit(rec)
                if b is not None:
                    n_yielded += 1
                    yield b
        if batch and not self.drop_remainder:
            yield self._collate(batch)

    @staticmethod
    def _collate(batch: List[Dict[str, np.ndarray]]) -> Dict[str, np.ndarray]:
        out: Dict[str, np.ndarray] = {}
        for key in batch[0]:
            vals = [b[key] for b in batch]
            if key == "name":
                out[key] = np.array(vals)
            else:
                out[key] = np.stack(vals)
        return out

    def count_examples(self) -> int:
        n = 0
        for f in self.files:
            for _ in tfrecord.read_tfrecords(f):
                n += 1
        return n
