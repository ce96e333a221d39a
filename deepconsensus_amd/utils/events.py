"""TensorBoard-compatible event files without TensorFlow.

Parity with the reference's training observability
(model_utils.py:549-583, model_train_custom_loop.py:164-166): train/eval
summary writers holding scalar series (loss, lr, epoch progress, eval
metrics). Files are standard `events.out.tfevents.*` records readable by
TensorBoard: TFRecord framing (dcio.tfrecord's masked CRC) around
hand-encoded `tensorflow.Event` protos — same approach as the hand-rolled
tf.train.Example codec (dcio/example_codec.py).

Proto schema (tensorflow/core/util/event.proto):
  Event { double wall_time = 1; int64 step = 2;
          oneof { string file_version = 3; Summary summary = 5; } }
  Summary { repeated Value value = 1; }
  Summary.Value { string tag = 1; float simple_value = 2; }
"""
from __future__ import annotations

import os
import socket
import struct
import time
from typing import Dict, Optional

from deepconsensus_amd.dcio.example_codec import _len_delim, _varint
from deepconsensus_amd.dcio.tfrecord import _masked_crc


def _encode_event(
    wall_time: float,
    step: int = 0,
    file_version: Optional[str] = None,
    scalars: Optional[Dict[str, float]] = None,
) -> bytes:
    out = b"\x09" + struct.pack("<d", wall_time)  # field 1, fixed64
    if step:
        out += b"\x10" + _varint(step)  # field 2, varint
    if file_version is not None:
        out += _len_delim(3, file_version.encode())
    if scalars:
        summary = b""
        for tag, value in scalars.items():
            val = _len_delim(1, tag.encode()) + b"\x15" + struct.pack(
                "<f", float(value)
            )  # tag (field 1) + simple_value (field 2, fixed32)
            summary += _len_delim(1, val)
        out += _len_delim(5, summary)
    return out


class EventWriter:
    """Appends scalar events to one tfevents file (unbuffered record
    framing; flush() makes partial runs readable)."""

    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        fname = (
            f"events.out.tfevents.{int(time.time())}."
            f"{socket.gethostname()}"
        )
        self._fh = open(os.path.join(logdir, fname), "wb")
        self._write(_encode_event(time.time(),
                                  file_version="brain.Event:2"))

    def _write(self, record: bytes) -> None:
        header = struct.pack("<Q", len(record))
        self._fh.write(
            header
            + struct.pack("<I", _masked_crc(header))
            + record
            + struct.pack("<I", _masked_crc(record))
        )

    def add_scalars(self, step: int, scalars: Dict[str, float]) -> None:
        self._write(_encode_event(time.time(), step=step, scalars=scalars))

    def add_scalar(self, step: int, tag: str, value: float) -> None:
        self.add_scalars(step, {tag: value})

    def flush(self) -> None:
        self._fh.flush()

    def close(self) -> None:
        if not self._fh.closed:
            self._fh.flush()
            self._fh.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
