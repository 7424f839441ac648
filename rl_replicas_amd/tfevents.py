"""Minimal native TensorBoard event-file (tfevents) writer.

The reference logs scalars through `torch.utils.tensorboard.
SummaryWriter` (reference: src/rl_replicas/metrics_manager.py:21,35),
producing `events.out.tfevents.*` files that the TensorBoard UI and the
benchmark converter (reference benchmarks/convert.py:28-113) consume.
The tensorboard package isn't available in this stack, so this module
writes the SAME on-disk format directly:

  TFRecord framing (one record per Event):
      uint64 length (LE) | uint32 masked_crc32c(length bytes)
      | data | uint32 masked_crc32c(data)
  Event / Summary protobuf messages, hand-encoded on the wire:
      Event:   wall_time (field 1, double), step (field 2, int64),
               file_version (field 3, string — first record only),
               summary (field 5, message)
      Summary: value (field 1, repeated message)
      Value:   tag (field 1, string), simple_value (field 2, float)

CRC32C is the Castagnoli polynomial with TensorFlow's rotate-and-add
masking.  Pure Python, table-driven — scalar logging writes a few
hundred bytes per epoch, so speed is irrelevant.  Output verified
byte-compatible with SummaryWriter's framing by the round-trip reader
in tests/test_tfevents.py.
"""
from __future__ import annotations

import os
import socket
import struct
import time
from typing import List, Optional, Tuple

_CRC_TABLE: List[int] = []


def _build_table() -> None:
    poly = 0x82F63B78  # Castagnoli, reversed
    for i in range(256):
        crc = i
        for _ in range(8):
            crc = (crc >> 1) ^ poly if crc & 1 else crc >> 1
        _CRC_TABLE.append(crc)


_build_table()


def crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = (crc >> 8) ^ _CRC_TABLE[(crc ^ b) & 0xFF]
    return crc ^ 0xFFFFFFFF


def masked_crc32c(data: bytes) -> int:
    crc = crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# protobuf wire-format helpers (no generated code needed)
# ---------------------------------------------------------------------------
def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        bits = n & 0x7F
        n >>= 7
        if n:
            out.append(bits | 0x80)
        else:
            out.append(bits)
            return bytes(out)


def _key(field: int, wire_type: int) -> bytes:
    return _varint((field << 3) | wire_type)


def _len_delim(field: int, payload: bytes) -> bytes:
    return _key(field, 2) + _varint(len(payload)) + payload


def _double(field: int, value: float) -> bytes:
    return _key(field, 1) + struct.pack("<d", value)


def _float(field: int, value: float) -> bytes:
    return _key(field, 5) + struct.pack("<f", value)


def _int64(field: int, value: int) -> bytes:
    return _key(field, 0) + _varint(value & 0xFFFFFFFFFFFFFFFF)


def encode_scalar_event(tag: str, value: float, step: int, wall_time: float) -> bytes:
    summary_value = _len_delim(1, tag.encode("utf-8")) + _float(2, float(value))
    summary = _len_delim(1, summary_value)
    return _double(1, wall_time) + _int64(2, int(step)) + _len_delim(5, summary)


def encode_file_version(wall_time: float) -> bytes:
    return _double(1, wall_time) + _len_delim(3, b"brain.Event:2")


def frame_record(data: bytes) -> bytes:
    header = struct.pack("<Q", len(data))
    return (
        header
        + struct.pack("<I", masked_crc32c(header))
        + data
        + struct.pack("<I", masked_crc32c(data))
    )


class EventFileWriter:
    """Append scalar events to an `events.out.tfevents.*` file."""

    def __init__(self, log_dir: str):
        os.makedirs(log_dir, exist_ok=True)
        fname = "events.out.tfevents.{:.0f}.{}.{}.0".format(
            time.time(), socket.gethostname(), os.getpid()
        )
        self.path = os.path.join(log_dir, fname)
        self._file = open(self.path, "ab")
        self._file.write(frame_record(encode_file_version(time.time())))
        self._file.flush()

    def add_scalar(self, tag: str, value: float, step: Optional[int]) -> None:
        event = encode_scalar_event(tag, value, int(step or 0), time.time())
        self._file.write(frame_record(event))

    def flush(self) -> None:
        self._file.flush()

    def close(self) -> None:
        self._file.close()


# ---------------------------------------------------------------------------
# reader (verification + the benchmark converter's tfevents input path)
# ---------------------------------------------------------------------------
def read_scalar_events(path: str) -> List[Tuple[str, float, int]]:
    """Parse (tag, value, step) scalars from a tfevents file, verifying
    the masked-CRC framing of every record."""
    out: List[Tuple[str, float, int]] = []
    with open(path, "rb") as f:
        data = f.read()
    pos = 0
    while pos < len(data):
        (length,) = struct.unpack_from("<Q", data, pos)
        (hcrc,) = struct.unpack_from("<I", data, pos + 8)
        assert hcrc == masked_crc32c(data[pos : pos + 8]), "header CRC mismatch"
        payload = data[pos + 12 : pos + 12 + length]
        (dcrc,) = struct.unpack_from("<I", data, pos + 12 + length)
        assert dcrc == masked_crc32c(payload), "data CRC mismatch"
        pos += 12 + length + 4
        out.extend(_parse_event(payload))
    return out


def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _parse_event(buf: bytes) -> List[Tuple[str, float, int]]:
    step = 0
    scalars: List[Tuple[str, float]] = []
    pos = 0
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        field, wt = key >> 3, key & 7
        if wt == 0:
            val, pos = _read_varint(buf, pos)
            if field == 2:
                step = val
        elif wt == 1:
            pos += 8
        elif wt == 5:
            pos += 4
        elif wt == 2:
            ln, pos = _read_varint(buf, pos)
            sub = buf[pos : pos + ln]
            pos += ln
            if field == 5:  # summary
                scalars.extend(_parse_summary(sub))
        else:  # pragma: no cover
            raise ValueError(f"unexpected wire type {wt}")
    return [(tag, value, step) for tag, value in scalars]


def _parse_summary(buf: bytes) -> List[Tuple[str, float]]:
    out = []
    pos = 0
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        field, wt = key >> 3, key & 7
        if wt == 2:
            ln, pos = _read_varint(buf, pos)
            sub = buf[pos : pos + ln]
            pos += ln
            if field == 1:  # Summary.Value
                tag, value = None, None
                p = 0
                while p < len(sub):
                    k, p = _read_varint(sub, p)
                    f, w = k >> 3, k & 7
                    if w == 2:
                        ln2, p = _read_varint(sub, p)
                        if f == 1:
                            tag = sub[p : p + ln2].decode("utf-8")
                        p += ln2
                    elif w == 5:
                        if f == 2:
                            (value,) = struct.unpack_from("<f", sub, p)
                        p += 4
                    elif w == 1:
                        p += 8
                    elif w == 0:
                        _, p = _read_varint(sub, p)
                if tag is not None and value is not None:
                    out.append((tag, value))
        else:  # pragma: no cover
            raise ValueError(f"unexpected wire type {wt} in summary")
    return out
