"""The hand-rolled TensorBoard event files (utils/logging.py) must be
readable by an INDEPENDENT decoder of the TFRecord framing
([len u64][masked_crc(len) u32][payload][masked_crc(payload) u32]) and the
Event/Summary protobuf subset — i.e. real TensorBoard could load them."""

import glob
import os
import struct

from distributed_reinforcement_learning_amd.utils.logging import (
    SummaryWriter, _masked_crc,
)


def _read_records(path):
    out = []
    with open(path, "rb") as f:
        data = f.read()
    i = 0
    while i < len(data):
        (ln,) = struct.unpack_from("<Q", data, i)
        (lcrc,) = struct.unpack_from("<I", data, i + 8)
        assert lcrc == _masked_crc(data[i:i + 8]), "length crc mismatch"
        payload = data[i + 12:i + 12 + ln]
        (pcrc,) = struct.unpack_from("<I", data, i + 12 + ln)
        assert pcrc == _masked_crc(payload), "payload crc mismatch"
        out.append(payload)
        i += 12 + ln + 4
    return out


def _pb_fields(buf):
    """Minimal protobuf walk: returns {field_no: [values]} with varints,
    fixed64 (as raw), and length-delimited (as bytes)."""
    fields = {}
    i = 0
    while i < len(buf):
        key = 0
        shift = 0
        while True:
            b = buf[i]
            i += 1
            key |= (b & 0x7F) << shift
            shift += 7
            if not b & 0x80:
                break
        field_no, wire = key >> 3, key & 7
        if wire == 0:  # varint
            v = 0
            shift = 0
            while True:
                b = buf[i]
                i += 1
                v |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
        elif wire == 1:  # fixed64
            v = buf[i:i + 8]
            i += 8
        elif wire == 2:  # length-delimited
            ln = 0
            shift = 0
            while True:
                b = buf[i]
                i += 1
                ln |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
            v = buf[i:i + ln]
            i += ln
        elif wire == 5:  # fixed32
            v = buf[i:i + 4]
            i += 4
        else:
            raise AssertionError(f"unexpected wire type {wire}")
        fields.setdefault(field_no, []).append(v)
    return fields


def test_event_file_framing_and_scalars(tmp_path):
    w = SummaryWriter(str(tmp_path / "run"))
    w.add_scalar("data/pi_loss", -1.5, 3)
    w.add_scalar("perf/env_frames_per_sec", 901000.0, 4)
    w.close()
    files = glob.glob(str(tmp_path / "run" / "events.out.tfevents.*"))
    assert len(files) == 1
    records = _read_records(files[0])
    # record 0 is the file-version event; then our two scalars
    assert len(records) >= 3
    scalars = {}
    for payload in records[1:]:
        ev = _pb_fields(payload)
        # Event: 1=wall_time(double), 2=step(varint), 5=summary(msg)
        step = ev.get(2, [0])[0]
        summary = _pb_fields(ev[5][0])
        value = _pb_fields(summary[1][0])  # Summary.value
        tag = value[1][0].decode()         # Value.tag
        (sv,) = struct.unpack("<f", value[2][0])  # Value.simple_value
        scalars[tag] = (step, sv)
    assert scalars["data/pi_loss"][0] == 3
    assert abs(scalars["data/pi_loss"][1] + 1.5) < 1e-6
    assert scalars["perf/env_frames_per_sec"][0] == 4
    assert abs(scalars["perf/env_frames_per_sec"][1] - 901000.0) < 1.0
