"""tf.train.Example wire-format codec (no TensorFlow dependency).

The reference's record schema (data_providers.py:41-58, pre_lib.py:764-787)
is plain protobuf: Example{ Features{ map<string, Feature> } } with Feature a
oneof of BytesList(1)/FloatList(2)/Int64List(3). This module hand-rolls that
wire format so records written here are byte-compatible with TensorFlow's
parser and vice versa.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Tuple, Union

FeatureValue = Union[List[bytes], List[int], List[float]]

# Kind tags used in the decoded dict.
BYTES, FLOAT, INT64 = "bytes", "float", "int64"


def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _read_varint(buf: bytes, off: int) -> Tuple[int, int]:
    shift = 0
    result = 0
    while True:
        b = buf[off]
        off += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, off
        shift += 7


def _len_delim(field: int, payload: bytes) -> bytes:
    return _varint((field << 3) | 2) + _varint(len(payload)) + payload


def _encode_feature(kind: str, values: FeatureValue) -> bytes:
    if kind == BYTES:
        inner = b"".join(_len_delim(1, v) for v in values)
        return _len_delim(1, inner)
    if kind == FLOAT:
        # packed floats (field 1, wire type 2)
        packed = struct.pack(f"<{len(values)}f", *values)
        inner = _len_delim(1, packed)
        return _len_delim(2, inner)
    if kind == INT64:
        packed = b"".join(_varint(v & 0xFFFFFFFFFFFFFFFF) for v in values)
        inner = _len_delim(1, packed)
        return _len_delim(3, inner)
    raise ValueError(kind)


def encode_example(features: Dict[str, Tuple[str, FeatureValue]]) -> bytes:
    """Encodes {name: (kind, values)} into a serialized tf.train.Example."""
    feats = bytearray()
    for name, (kind, values) in features.items():
        entry = _len_delim(1, name.encode()) + _len_delim(
            2, _encode_feature(kind, values)
        )
        feats += _len_delim(1, entry)
    return bytes(_len_delim(1, bytes(feats)))


def _decode_list(kind_field: int, payload: bytes) -> Tuple[str, FeatureValue]:
    off = 0
    values: FeatureValue = []
    if kind_field == 1:  # BytesList
        while off < len(payload):
            tag, off = _read_varint(payload, off)
            assert tag == (1 << 3) | 2, tag
            ln, off = _read_varint(payload, off)
            values.append(payload[off:off + ln])
            off += ln
        return BYTES, values
    if kind_field == 2:  # FloatList
        while off < len(payload):
            tag, off = _read_varint(payload, off)
            wt = tag & 7
            if wt == 2:  # packed
                ln, off = _read_varint(payload, off)
                values.extend(
                    struct.unpack(f"<{ln // 4}f", payload[off:off + ln])
                )
                off += ln
            else:  # unpacked fixed32
                values.append(struct.unpack_from("<f", payload, off)[0])
                off += 4
        return FLOAT, values
    if kind_field == 3:  # Int64List
        while off < len(payload):
            tag, off = _read_varint(payload, off)
            wt = tag & 7
            if wt == 2:
                ln, off = _read_varint(payload, off)
                end = off + ln
                while off < end:
                    v, off = _read_varint(payload, off)
                    if v >= 1 << 63:
                        v -= 1 << 64
                    values.append(v)
            else:
                v, off = _read_varint(payload, off)
                if v >= 1 << 63:
                    v -= 1 << 64
                values.append(v)
        return INT64, values
    raise ValueError(f"unknown Feature kind field {kind_field}")


def decode_example(buf: bytes) -> Dict[str, Tuple[str, FeatureValue]]:
    """Decodes a serialized tf.train.Example into {name: (kind, values)}."""
    out: Dict[str, Tuple[str, FeatureValue]] = {}
    off = 0
    # Example -> features (field 1).
    tag, off = _read_varint(buf, off)
    assert tag == (1 << 3) | 2, "expected Example.features"
    flen, off = _read_varint(buf, off)
    feats = buf[off:off + flen]
    off = 0
    while off < len(feats):
        tag, off = _read_varint(feats, off)
        assert tag == (1 << 3) | 2, "expected map entry"
        elen, off = _read_varint(feats, off)
        entry = feats[off:off + elen]
        off += elen
        eoff = 0
        name = None
        kind_values = None
        while eoff < len(entry):
            etag, eoff = _read_varint(entry, eoff)
            fieldno, wt = etag >> 3, etag & 7
            assert wt == 2
            ln, eoff = _read_varint(entry, eoff)
            payload = entry[eoff:eoff + ln]
            eoff += ln
            if fieldno == 1:
                name = payload.decode()
            else:  # Feature message: one oneof field
                if len(payload) == 0:
                    kind_values = (BYTES, [])
                    continue
                ftag, foff = _read_varint(payload, 0)
                kf, fwt = ftag >> 3, ftag & 7
                assert fwt == 2
                flen2, foff = _read_varint(payload, foff)
                kind_values = _decode_list(
                    kf, payload[foff:foff + flen2]
                )
        assert name is not None and kind_values is not None
        out[name] = kind_values
    return out
