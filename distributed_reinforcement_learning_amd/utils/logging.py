"""TensorBoard-compatible scalar logging without external dependencies.

The reference logs through ``tensorboardX.SummaryWriter`` (train_impala.py:91,
:130). tensorboardX is not installable in this offline image, so this module
writes genuine TensorBoard event files (``events.out.tfevents.*``) directly:
each record is [len:u64][masked_crc32c(len):u32][payload][masked_crc32c(payload):u32]
with a hand-rolled protobuf encoding of the Event/Summary messages — scalars
only, which is all the reference ever logs. A JSONL mirror is written next to
the event file so scalars stay greppable without TensorBoard.

Scalar names and run-dir layout match the reference exactly
(learner: runs/learner; actors: runs/{env}/actor_{task}) so dashboards are
comparable side by side.
"""

from __future__ import annotations

import json
import os
import struct
import time


# ---------------------------------------------------------------------------
# crc32c (Castagnoli), table-driven, + TensorFlow record masking
# ---------------------------------------------------------------------------

_CRC_TABLE = []


def _build_table() -> None:
    poly = 0x82F63B78
    for i in range(256):
        crc = i
        for _ in range(8):
            crc = (crc >> 1) ^ poly if crc & 1 else crc >> 1
        _CRC_TABLE.append(crc)


_build_table()


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# minimal protobuf writers for Event{wall_time, step, Summary{value{tag,
# simple_value}}}
# ---------------------------------------------------------------------------


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


def _tag(field_no: int, wire: int) -> bytes:
    return _varint((field_no << 3) | wire)


def _pb_double(field_no: int, v: float) -> bytes:
    return _tag(field_no, 1) + struct.pack("<d", v)


def _pb_float(field_no: int, v: float) -> bytes:
    return _tag(field_no, 5) + struct.pack("<f", v)


def _pb_int(field_no: int, v: int) -> bytes:
    return _tag(field_no, 0) + _varint(v)


def _pb_bytes(field_no: int, v: bytes) -> bytes:
    return _tag(field_no, 2) + _varint(len(v)) + v


def _scalar_event(tag: str, value: float, step: int, wall: float) -> bytes:
    # Summary.Value: tag=1 (string), simple_value=2 (float)
    sv = _pb_bytes(1, tag.encode()) + _pb_float(2, float(value))
    # Summary: value=1 (repeated message)
    summary = _pb_bytes(1, sv)
    # Event: wall_time=1 (double), step=2 (int64), summary=5 (message)
    return _pb_double(1, wall) + _pb_int(2, int(step)) + _pb_bytes(5, summary)


def _record(payload: bytes) -> bytes:
    header = struct.pack("<Q", len(payload))
    return (header + struct.pack("<I", _masked_crc(header))
            + payload + struct.pack("<I", _masked_crc(payload)))


class SummaryWriter:
    """Drop-in for ``tensorboardX.SummaryWriter`` (scalars only)."""

    def __init__(self, logdir: str):
        self.logdir = logdir
        os.makedirs(logdir, exist_ok=True)
        stamp = int(time.time())
        host = os.uname().nodename
        self._path = os.path.join(
            logdir, f"events.out.tfevents.{stamp}.{host}.{os.getpid()}")
        self._f = open(self._path, "ab", buffering=0)
        self._jsonl = open(os.path.join(logdir, "scalars.jsonl"), "a")
        # file-version header event
        ver = _pb_double(1, time.time()) + _pb_bytes(3, b"brain.Event:2")
        self._f.write(_record(ver))

    def add_scalar(self, tag: str, value, step: int) -> None:
        wall = time.time()
        self._f.write(_record(_scalar_event(tag, float(value), step, wall)))
        self._jsonl.write(json.dumps(
            {"tag": tag, "value": float(value), "step": int(step),
             "wall_time": wall}) + "\n")
        self._jsonl.flush()

    def flush(self) -> None:
        self._jsonl.flush()

    def close(self) -> None:
        try:
            self._f.close()
            self._jsonl.close()
        except Exception:
            pass

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
