"""Native TensorBoard event-file writer (no tensorboard package).

The reference writes TF summary event files a stock TensorBoard can open
(`base_model.py:46-47`); this image has no tensorboard/protobuf wheel for
it, so the Event protobuf and TFRecord framing are hand-encoded here:

  * TFRecord frame: uint64 length (LE) + masked CRC32C(length bytes) +
    payload + masked CRC32C(payload)  (mask = ((c >> 15 | c << 17) +
    0xa282ead8) mod 2^32);
  * Event message:  1: double wall_time, 2: int64 step,
    3: string file_version ("brain.Event:2", first record only),
    5: Summary { repeated 1: Value { 1: string tag,
    2: float simple_value } }.

Scalar and histogram summaries are emitted — the same per-variable
set the reference logs (model.py:534-542).
"""

import os
import socket
import struct
import time

_CRC_TABLE = []


def _crc_table():
    global _CRC_TABLE
    if _CRC_TABLE:
        return _CRC_TABLE
    poly = 0x82F63B78  # CRC32C (Castagnoli), reflected
    tab = []
    for n in range(256):
        c = n
        for _ in range(8):
            c = (c >> 1) ^ (poly if c & 1 else 0)
        tab.append(c)
    _CRC_TABLE = tab
    return tab


def _crc32c(data):
    tab = _crc_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = tab[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data):
    c = _crc32c(data)
    return (((c >> 15) | (c << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def _varint(n):
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


def _field_bytes(num, payload):
    return _varint((num << 3) | 2) + _varint(len(payload)) + payload


def _event(wall_time, step=None, file_version=None, summary=None):
    msg = _varint((1 << 3) | 1) + struct.pack('<d', wall_time)
    if step is not None:
        msg += _varint((2 << 3) | 0) + _varint(step)
    if file_version is not None:
        msg += _field_bytes(3, file_version.encode())
    if summary is not None:
        msg += _field_bytes(5, summary)
    return msg


def _scalar_value(tag, value):
    v = _field_bytes(1, tag.encode()) \
        + _varint((2 << 3) | 5) + struct.pack('<f', float(value))
    return _field_bytes(1, v)


def _histo_value(tag, values):
    """Summary.Value with a HistogramProto (field 5): fields
    1 min, 2 max, 3 num, 4 sum, 5 sum_squares, 6 bucket_limit,
    7 bucket (both packed repeated double)."""
    import math
    n = len(values)
    vmin = min(values) if n else 0.0
    vmax = max(values) if n else 0.0
    vsum = float(sum(values))
    vsq = float(sum(x * x for x in values))
    # exponential bucket edges covering [vmin, vmax], TB-style
    limits = []
    edge = 1e-12
    while edge < max(abs(vmin), abs(vmax), 1e-12) * 1.1:
        edge *= 1.1
        limits.append(edge)
    edges = sorted({-e for e in limits} | set(limits) | {0.0})
    counts = [0] * (len(edges) + 1)
    for x in values:
        lo, hi = 0, len(edges)
        while lo < hi:
            mid = (lo + hi) // 2
            if x <= edges[mid]:
                hi = mid
            else:
                lo = mid + 1
        counts[lo] += 1
    # trim empty tail buckets (keeps files small)
    last = max((i for i, c in enumerate(counts) if c), default=0)
    edges = edges[:last + 1]
    counts = counts[:last + 1]

    h = _varint((1 << 3) | 1) + struct.pack('<d', vmin)
    h += _varint((2 << 3) | 1) + struct.pack('<d', vmax)
    h += _varint((3 << 3) | 1) + struct.pack('<d', float(n))
    h += _varint((4 << 3) | 1) + struct.pack('<d', vsum)
    h += _varint((5 << 3) | 1) + struct.pack('<d', vsq)
    h += _field_bytes(6, b''.join(struct.pack('<d', e) for e in edges))
    h += _field_bytes(7, b''.join(struct.pack('<d', float(c))
                                  for c in counts))
    v = _field_bytes(1, tag.encode()) + _field_bytes(5, h)
    return _field_bytes(1, v)


class TBEventWriter(object):
    """Writes events.out.tfevents.* files TensorBoard can load."""

    def __init__(self, log_dir):
        os.makedirs(log_dir, exist_ok=True)
        name = 'events.out.tfevents.%d.%s' % (
            int(time.time()), socket.gethostname())
        self.path = os.path.join(log_dir, name)
        self._f = open(self.path, 'ab')
        self._record(_event(time.time(), file_version='brain.Event:2'))

    def _record(self, payload):
        hdr = struct.pack('<Q', len(payload))
        self._f.write(hdr)
        self._f.write(struct.pack('<I', _masked_crc(hdr)))
        self._f.write(payload)
        self._f.write(struct.pack('<I', _masked_crc(payload)))

    def add_scalar(self, tag, value, step):
        self._record(_event(time.time(), step=int(step),
                            summary=_scalar_value(tag, value)))

    def add_scalars(self, scalars, step):
        summary = b''.join(_scalar_value(t, v)
                           for t, v in scalars.items())
        self._record(_event(time.time(), step=int(step),
                            summary=summary))

    def add_histogram(self, tag, values, step):
        self._record(_event(time.time(), step=int(step),
                            summary=_histo_value(tag, list(values))))

    def flush(self):
        self._f.flush()

    def close(self):
        self._f.close()
