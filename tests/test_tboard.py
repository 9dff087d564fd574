"""tfevents writer: record framing (CRC), protobuf decode, directory layout
(reference TB layout — SURVEY §5.5)."""
import os
import struct

from ddp_tricks_amd.utils.tboard import SummaryWriter, _crc32c, _masked_crc


def test_crc32c_vectors():
    # Known CRC-32C test vectors
    assert _crc32c(b"") == 0x00000000
    assert _crc32c(b"123456789") == 0xE3069283


def _read_records(path):
    out = []
    with open(path, "rb") as f:
        while True:
            header = f.read(8)
            if len(header) < 8:
                break
            (length,) = struct.unpack("<Q", header)
            (hcrc,) = struct.unpack("<I", f.read(4))
            assert hcrc == _masked_crc(header)
            data = f.read(length)
            (dcrc,) = struct.unpack("<I", f.read(4))
            assert dcrc == _masked_crc(data)
            out.append(data)
    return out


def _decode_varint(buf, i):
    shift, val = 0, 0
    while True:
        b = buf[i]
        i += 1
        val |= (b & 0x7F) << shift
        if not b & 0x80:
            return val, i
        shift += 7


def _parse_event(data):
    """Tiny protobuf parser for the fields we emit."""
    i, out = 0, {}
    while i < len(data):
        key, i = _decode_varint(data, i)
        field, wire = key >> 3, key & 7
        if wire == 1:
            val = struct.unpack("<d", data[i:i + 8])[0]
            i += 8
        elif wire == 5:
            val = struct.unpack("<f", data[i:i + 4])[0]
            i += 4
        elif wire == 2:
            ln, i = _decode_varint(data, i)
            val = data[i:i + ln]
            i += ln
        elif wire == 0:
            val, i = _decode_varint(data, i)
        else:
            raise AssertionError(f"wire {wire}")
        out[field] = val
    return out


def test_scalar_roundtrip(tmp_path):
    logdir = os.path.join(tmp_path, "run")
    w = SummaryWriter(logdir)
    w.add_scalar("LR", 0.1, 3)
    w.close()
    files = os.listdir(logdir)
    evfiles = [f for f in files if f.startswith("events.out.tfevents.")]
    assert len(evfiles) == 1
    recs = _read_records(os.path.join(logdir, evfiles[0]))
    assert len(recs) == 2  # file_version + scalar
    first = _parse_event(recs[0])
    assert first[3] == b"brain.Event:2"
    ev = _parse_event(recs[1])
    assert ev[2] == 3  # step
    summary = _parse_event(ev[5])
    value = _parse_event(summary[1])
    assert value[1] == b"LR"
    assert abs(value[2] - 0.1) < 1e-6


def test_add_scalars_layout(tmp_path):
    logdir = os.path.join(tmp_path, "DDP_warmup")
    w = SummaryWriter(logdir)
    w.add_scalars("Loss", {"train": 1.0, "valid": 2.0}, 0)
    w.add_scalars("Acc", {"train": 0.5, "valid": 0.6}, 0)
    w.close()
    subdirs = sorted(d for d in os.listdir(logdir)
                     if os.path.isdir(os.path.join(logdir, d)))
    assert subdirs == ["Acc_train", "Acc_valid", "Loss_train", "Loss_valid"]
    for d in subdirs:
        sub = os.path.join(logdir, d)
        assert any(f.startswith("events.out.tfevents.") for f in os.listdir(sub))
