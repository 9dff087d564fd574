"""Minimal TensorBoard event-file writer (no tensorboard/TF dependency).

The reference logs per-epoch scalars through torch's SummaryWriter
(reference utils/train.py:64-65,95-98) producing tfevents files under
``{model_path}/logs/{exp_name}`` with tags ``LR`` (add_scalar) and grouped
``Loss``/``Acc`` (add_scalars → per-key subdirectories ``Loss_train/`` etc.,
SURVEY §5.5).  This module reproduces that on-disk layout by hand-encoding
the protobuf wire format of tensorflow.Event / Summary / Summary.Value
(only wall_time, step, file_version, tag and simple_value are needed) and
the TFRecord framing (length, masked CRC32C).  TensorBoard reads the output
directly.
"""
from __future__ import annotations

import os
import socket
import struct
import time
from typing import Dict

# ---------------------------------------------------------------- crc32c ---

_CRC32C_POLY = 0x82F63B78
_CRC_TABLE = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ _CRC32C_POLY if (_c & 1) else (_c >> 1)
    _CRC_TABLE.append(_c)


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ------------------------------------------------------- protobuf encoding ---

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


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _pb_string(field: int, s: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(s)) + s


def _pb_double(field: int, v: float) -> bytes:
    return _tag(field, 1) + struct.pack("<d", v)


def _pb_float(field: int, v: float) -> bytes:
    return _tag(field, 5) + struct.pack("<f", v)


def _pb_varint(field: int, v: int) -> bytes:
    return _tag(field, 0) + _varint(v & 0xFFFFFFFFFFFFFFFF)


def _event_bytes(wall_time: float, step: int = 0, file_version: str = None,
                 scalar_tag: str = None, scalar_value: float = None) -> bytes:
    ev = bytearray()
    ev += _pb_double(1, wall_time)              # Event.wall_time
    if step:
        ev += _pb_varint(2, step)               # Event.step
    if file_version is not None:
        ev += _pb_string(3, file_version.encode())  # Event.file_version
    if scalar_tag is not None:
        value = _pb_string(1, scalar_tag.encode()) + _pb_float(2, float(scalar_value))
        summary = _pb_string(1, value)          # Summary.value (repeated)
        ev += _pb_string(5, summary)            # Event.summary
    return bytes(ev)


class _EventFile:
    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        fname = (
            f"events.out.tfevents.{int(time.time())}."
            f"{socket.gethostname()}.{os.getpid()}.0"
        )
        self._f = open(os.path.join(logdir, fname), "wb")
        self._write_record(_event_bytes(time.time(), file_version="brain.Event:2"))
        self._f.flush()

    def _write_record(self, data: bytes) -> None:
        header = struct.pack("<Q", len(data))
        self._f.write(header)
        self._f.write(struct.pack("<I", _masked_crc(header)))
        self._f.write(data)
        self._f.write(struct.pack("<I", _masked_crc(data)))

    def write_scalar(self, tag: str, value: float, step: int) -> None:
        self._write_record(_event_bytes(time.time(), step=step,
                                        scalar_tag=tag, scalar_value=value))
        self._f.flush()

    def close(self) -> None:
        if not self._f.closed:
            self._f.flush()
            self._f.close()


class SummaryWriter:
    """API-compatible subset of torch.utils.tensorboard.SummaryWriter."""

    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        self._main = _EventFile(log_dir)
        self._children: Dict[str, _EventFile] = {}

    def add_scalar(self, tag: str, scalar_value: float, global_step: int = 0) -> None:
        self._main.write_scalar(tag, float(scalar_value), global_step)

    def add_scalars(self, main_tag: str, tag_scalar_dict: Dict[str, float],
                    global_step: int = 0) -> None:
        # torch's add_scalars writes each series into the subdirectory
        # {log_dir}/{main_tag}_{key} with the scalar tag = main_tag — this is
        # what produced the golden runs' Loss_train/ Loss_valid/ ... layout.
        for key, value in tag_scalar_dict.items():
            sub = f"{main_tag}_{key}"
            if sub not in self._children:
                self._children[sub] = _EventFile(os.path.join(self.log_dir, sub))
            self._children[sub].write_scalar(main_tag, float(value), global_step)

    def flush(self) -> None:
        pass  # every record is flushed on write

    def close(self) -> None:
        self._main.close()
        for child in self._children.values():
            child.close()
