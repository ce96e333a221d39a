

def test_event_writer_tensorboard_format(tmp_path):
    """EventWriter emits CRC-valid TFRecord framing whose first record is
    the brain.Event:2 version event and whose scalar records decode back
    (minimal proto walk) to the written tag/value/step."""
    import glob
    import struct

    from deepconsensus_amd.dcio import tfrecord
    from deepconsensus_amd.utils.events import EventWriter

    logdir = str(tmp_path / "summaries" / "train")
    with EventWriter(logdir) as w:
        w.add_scalars(7, {"train/loss": 1.5, "train/learning_rate": 1e-3})
        w.add_scalar(8, "train/loss", 1.25)
    files = glob.glob(logdir + "/events.out.tfevents.*")
    assert len(files) == 1
    records = list(tfrecord.read_tfrecords(files[0]))  # validates CRCs
    assert len(records) == 3
    assert b"brain.Event:2" in records[0]

    def parse_event(buf):
        # Minimal proto walk: field 1 fixed64 wall_time, 2 varint step,
        # 5 len-delim summary of (tag, fixed32 simple_value) values.
        off, step, scalars = 0, 0, {}
        while off < len(buf):
            key = buf[off]
            field, wire = key >> 3, key & 7
            off += 1
            if wire == 1:
                off += 8
            elif wire == 0:
                v, shift = 0, 0
                while True:
                    b = buf[off]
                    off += 1
                    v |= (b & 0x7F) << shift
                    shift += 7
                    if not b & 0x80:
                        break
                if field == 2:
                    step = v
            elif wire == 2:
                ln, shift = 0, 0
                while True:
                    b = buf[off]
                    off += 1
                    ln |= (b & 0x7F) << shift
                    shift += 7
                    if not b & 0x80:
                        break
                payload = buf[off : off + ln]
                off += ln
                if field == 5:
                    # Summary: repeated Value (field 1, len-delim).
                    so = 0
                    while so < len(payload):
                        assert payload[so] == 0x0A
                        so += 1
                        vl = payload[so]
                        so += 1
                        val = payload[so : so + vl]
                        so += vl
                        # Value: tag (field 1 len), simple_value (field 2
                        # fixed32).
                        assert val[0] == 0x0A
                        tl = val[1]
                        tag = val[2 : 2 + tl].decode()
                        assert val[2 + tl] == 0x15
                        (fv,) = struct.unpack("<f", val[3 + tl : 7 + tl])
                        scalars[tag] = fv
            elif wire == 5:
                off += 4
        return step, scalars

    step, scalars = parse_event(records[1])
    assert step == 7
    assert abs(scalars["train/loss"] - 1.5) < 1e-6
    assert abs(scalars["train/learning_rate"] - 1e-3) < 1e-9
    step, scalars = parse_event(records[2])
    assert step == 8 and abs(scalars["train/loss"] - 1.25) < 1e-6


def test_trace_ranges_noop_on_cpu():
    """trace.range is a no-op context manager without a GPU."""
    from deepconsensus_amd.utils import trace

    trace.enable(True)
    with trace.range("K2_embed_gather"):
        x = 1 + 1
    trace.mark("point")
    trace.enable(False)
    assert x == 2


def test_train_writes_tensorboard_events(tmp_path):
    """train_model emits TB event files for train and eval scalars."""
    import glob

    from deepconsensus_amd.models import train as train_lib
    from test_train import make_training_data, _tiny_params

    train_file, _ = make_training_data(tmp_path)
    params = _tiny_params(train_file)
    out = str(tmp_path / "out")
    train_lib.train_model(
        out, params, device="cpu", eval_every=2, limit_steps=2,
    )
    tfiles = glob.glob(out + "/summaries/train/events.out.tfevents.*")
    efiles = glob.glob(out + "/summaries/eval/events.out.tfevents.*")
    assert tfiles and efiles
    from deepconsensus_amd.dcio import tfrecord

    erecs = list(tfrecord.read_tfrecords(efiles[0]))
    assert len(erecs) >= 2  # version header + >=1 eval summary
    assert any(b"eval/per_example_accuracy" in r for r in erecs)


def test_enable_tuned_gemms_cpu_noop(monkeypatch):
    """On CPU the tuned-GEMM loader is a no-op returning False."""
    import deepconsensus_amd.utils.tuned_gemm as tg

    monkeypatch.setattr(tg, "_ENABLED", None)
    assert tg.enable_tuned_gemms() is False
    # Disabled via env as well.
    monkeypatch.setattr(tg, "_ENABLED", None)
    monkeypatch.setenv("DC_TUNED_GEMM", "0")
    assert tg.enable_tuned_gemms() is False
