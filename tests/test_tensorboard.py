"""The native TensorBoard scalar writer produces well-formed event files:
TFRecord framing with valid masked CRC32C, a file_version header event, and
scalar summaries that decode back to the logged (tag, value, step)."""
import glob
import struct

from perceiver_amd.utils.tensorboard import ScalarWriter, _masked_crc


def _read_records(path):
    records = []
    with open(path, "rb") as f:
        while True:
            header = f.read(8)
            if len(header) < 8:
                break
            (length,) = struct.unpack("<Q", header)
            (hcrc,) = struct.unpack("<I", f.read(4))
            assert hcrc == _masked_crc(header)
            payload = f.read(length)
            (pcrc,) = struct.unpack("<I", f.read(4))
            assert pcrc == _masked_crc(payload)
            records.append(payload)
    return records


def _decode_varint(buf, i):
    shift, val = 0, 0
    while True:
        b = buf[i]
        i += 1
        val |= (b & 0x7F) << shift
        if not b & 0x80:
            return val, i
        shift += 7


def _parse_event(buf):
    i, out = 0, {}
    while i < len(buf):
        key, i = _decode_varint(buf, i)
        field, wire = key >> 3, key & 7
        if wire == 1:
            out[field] = struct.unpack("<d", buf[i:i + 8])[0]; i += 8
        elif wire == 0:
            out[field], i = _decode_varint(buf, i)
        elif wire == 5:
            out[field] = struct.unpack("<f", buf[i:i + 4])[0]; i += 4
        elif wire == 2:
            ln, i = _decode_varint(buf, i)
            out[field] = buf[i:i + ln]; i += ln
    return out


def test_scalar_writer_roundtrip(tmp_path):
    w = ScalarWriter(str(tmp_path))
    w.add_scalar("train_loss", 1.25, step=3)
    w.add_scalar("lr", 0.001, step=3)
    w.close()

    files = glob.glob(str(tmp_path / "events.out.tfevents.*"))
    assert len(files) == 1
    records = _read_records(files[0])
    assert len(records) == 3
    head = _parse_event(records[0])
    assert head[3] == b"brain.Event:2"

    ev = _parse_event(records[1])
    assert ev[2] == 3  # step
    summary = _parse_event(ev[5])
    value = _parse_event(summary[1])
    assert value[1] == b"train_loss"
    assert abs(value[2] - 1.25) < 1e-6


def test_trainer_writes_tb_events(tmp_path):
    import torch

    from perceiver_amd.train.trainer import Trainer, TrainConfig

    t = Trainer(TrainConfig(max_steps=2, log_every=1, lr=1e-2, lr_schedule="none",
                            out_dir=str(tmp_path)))
    model = torch.nn.Linear(4, 2)
    t.fit_steps(model, [torch.randn(2, 4) for _ in range(2)], lambda m, b: m(b).square().mean())
    files = glob.glob(str(tmp_path / "tb" / "events.out.tfevents.*"))
    assert files and len(_read_records(files[0])) >= 3
