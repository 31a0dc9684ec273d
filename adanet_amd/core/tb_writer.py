"""TensorBoard event-file (tfevents) writer with zero TF dependency.

"TensorBoard is AdaNet's UI" (reference docs/source/tensorboard.md:1-8); the
reference's _ScopedSummaryV2 writes real per-scope event files
(adanet/core/summary.py:375-637). This module produces files TensorBoard
loads natively:

  * TFRecord framing: [len u64][masked crc32c(len)][payload][masked
    crc32c(payload)] per record.
  * payload = hand-encoded `tensorflow.Event` protobuf (varint wire format):
    Event{wall_time=1(double), step=2(int64), file_version=3(string),
    summary=5(Summary)}; Summary{value=1(repeated Value)};
    Value{tag=1, simple_value=2(float), histo=5(HistogramProto),
    tensor=8(TensorProto), metadata=9(SummaryMetadata)}.

The inverse parser (`read_tfevents`) exists for tests: CRC-checked framing
round-trip without TensorFlow/TensorBoard installed.
"""

from __future__ import annotations

import os
import socket
import struct
import time
from typing import List, Optional, Tuple

# ---------------------------------------------------------------- crc32c

_CRC_TABLE = []


def _crc_table():
    global _CRC_TABLE
    if _CRC_TABLE:
        return _CRC_TABLE
    poly = 0x82F63B78  # Castagnoli, reflected
    table = []
    for i in range(256):
        c = i
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        table.append(c)
    _CRC_TABLE = table
    return table


def crc32c(data: bytes) -> int:
    table = _crc_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = table[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def masked_crc32c(data: bytes) -> int:
    crc = crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ------------------------------------------------------------- protobuf

def _varint(n: int) -> bytes:
    out = bytearray()
    n &= (1 << 64) - 1
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _key(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _f_double(field: int, v: float) -> bytes:
    return _key(field, 1) + struct.pack("<d", v)


def _f_float(field: int, v: float) -> bytes:
    return _key(field, 5) + struct.pack("<f", v)


def _f_varint(field: int, v: int) -> bytes:
    return _key(field, 0) + _varint(v)


def _f_bytes(field: int, v: bytes) -> bytes:
    return _key(field, 2) + _varint(len(v)) + v


def _f_packed_doubles(field: int, vals) -> bytes:
    payload = b"".join(struct.pack("<d", v) for v in vals)
    return _f_bytes(field, payload)


def _value_scalar(tag: str, value: float) -> bytes:
    return _f_bytes(1, _f_bytes(1, tag.encode()) + _f_float(2, value))


def _value_histo(tag: str, mn, mx, num, total, sumsq, limits,
                 counts) -> bytes:
    histo = (_f_double(1, mn) + _f_double(2, mx) + _f_double(3, num) +
             _f_double(4, total) + _f_double(5, sumsq) +
             _f_packed_doubles(6, limits) + _f_packed_doubles(7, counts))
    return _f_bytes(1, _f_bytes(1, tag.encode()) + _f_bytes(5, histo))


def _value_text(tag: str, text: str) -> bytes:
    # TensorProto: dtype=7 (DT_STRING), shape [1], string_val
    shape = _f_bytes(2, _f_varint(1, 1))  # TensorShapeProto{dim{size:1}}
    tensor = _f_varint(1, 7) + shape + _f_bytes(8, text.encode())
    # SummaryMetadata{plugin_data{plugin_name:"text"}}
    meta = _f_bytes(1, _f_bytes(1, b"text"))
    return _f_bytes(1, (_f_bytes(1, tag.encode()) + _f_bytes(8, tensor) +
                        _f_bytes(9, meta)))


def _event(wall_time: float, step: int, summary_values: bytes = b"",
           file_version: Optional[str] = None) -> bytes:
    out = _f_double(1, wall_time) + _f_varint(2, step)
    if file_version is not None:
        out += _f_bytes(3, file_version.encode())
    if summary_values:
        out += _f_bytes(5, summary_values)
    return out


def _record(payload: bytes) -> bytes:
    header = struct.pack("<Q", len(payload))
    return (header + struct.pack("<I", masked_crc32c(header)) + payload +
            struct.pack("<I", masked_crc32c(payload)))


# --------------------------------------------------------------- writer

class TBEventWriter(object):
    """Appends TensorBoard events to <logdir>/events.out.tfevents.*"""

    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        host = socket.gethostname() or "local"
        self._path = os.path.join(
            logdir, "events.out.tfevents.%d.%s" % (int(time.time()), host))
        with open(self._path, "ab") as f:
            f.write(_record(_event(time.time(), 0,
                                   file_version="brain.Event:2")))

    @property
    def path(self):
        return self._path

    def _append(self, rec: bytes):
        with open(self._path, "ab") as f:
            f.write(rec)

    def scalar(self, tag: str, value: float, step: int,
               wall_time: Optional[float] = None):
        self._append(_record(_event(wall_time or time.time(), step,
                                    _value_scalar(tag, float(value)))))

    def histogram(self, tag: str, values, step: int,
                  wall_time: Optional[float] = None, bins: int = 30):
        vals = [float(v) for v in values]
        if not vals:
            return
        mn, mx = min(vals), max(vals)
        total = sum(vals)
        sumsq = sum(v * v for v in vals)
        if mx == mn:
            limits, counts = [mx], [float(len(vals))]
        else:
            width = (mx - mn) / bins
            limits = [mn + width * (i + 1) for i in range(bins)]
            counts = [0.0] * bins
            for v in vals:
                idx = min(int((v - mn) / width), bins - 1)
                counts[idx] += 1.0
        self._append(_record(_event(
            wall_time or time.time(), step,
            _value_histo(tag, mn, mx, len(vals), total, sumsq, limits,
                         counts))))

    def text(self, tag: str, text: str, step: int,
             wall_time: Optional[float] = None):
        self._append(_record(_event(wall_time or time.time(), step,
                                    _value_text(tag, text))))


# --------------------------------------------------------------- reader

def _parse_varint(buf: bytes, i: int) -> Tuple[int, int]:
    shift = 0
    out = 0
    while True:
        b = buf[i]
        i += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, i
        shift += 7


def _parse_fields(buf: bytes):
    i = 0
    while i < len(buf):
        key, i = _parse_varint(buf, i)
        field, wire = key >> 3, key & 7
        if wire == 0:
            v, i = _parse_varint(buf, i)
        elif wire == 1:
            v = buf[i:i + 8]
            i += 8
        elif wire == 5:
            v = buf[i:i + 4]
            i += 4
        elif wire == 2:
            n, i = _parse_varint(buf, i)
            v = buf[i:i + n]
            i += n
        else:
            raise ValueError("unsupported wire type %d" % wire)
        yield field, wire, v


def read_tfevents(path: str) -> List[dict]:
    """CRC-checked parse of a tfevents file into event dicts (test helper;
    implements the inverse of the writer — no TF needed)."""
    out = []
    with open(path, "rb") as f:
        data = f.read()
    i = 0
    while i < len(data):
        (ln,) = struct.unpack_from("<Q", data, i)
        (hcrc,) = struct.unpack_from("<I", data, i + 8)
        if masked_crc32c(data[i:i + 8]) != hcrc:
            raise ValueError("header CRC mismatch at offset %d" % i)
        payload = data[i + 12:i + 12 + ln]
        (pcrc,) = struct.unpack_from("<I", data, i + 12 + ln)
        if masked_crc32c(payload) != pcrc:
            raise ValueError("payload CRC mismatch at offset %d" % i)
        i += 16 + ln
        ev = {"step": 0, "values": []}
        for field, wire, v in _parse_fields(payload):
            if field == 1 and wire == 1:
                ev["wall_time"] = struct.unpack("<d", v)[0]
            elif field == 2:
                ev["step"] = v
            elif field == 3:
                ev["file_version"] = v.decode()
            elif field == 5:
                for f2, w2, v2 in _parse_fields(v):
                    if f2 != 1:
                        continue
                    val = {}
                    for f3, w3, v3 in _parse_fields(v2):
                        if f3 == 1:
                            val["tag"] = v3.decode()
                        elif f3 == 2:
                            val["simple_value"] = struct.unpack("<f", v3)[0]
                        elif f3 == 5:
                            histo = {}
                            for f4, w4, v4 in _parse_fields(v3):
                                if f4 in (1, 2, 3, 4, 5):
                                    histo[{1: "min", 2: "max", 3: "num",
                                           4: "sum", 5: "sum_squares"}[f4]] \
                                        = struct.unpack("<d", v4)[0]
                            val["histo"] = histo
                        elif f3 == 8:
                            for f4, w4, v4 in _parse_fields(v3):
                                if f4 == 8:
                                    val["text"] = v4.decode()
                    ev["values"].append(val)
        out.append(ev)
    return out
