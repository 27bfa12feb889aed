"""Prometheus remote-write (same wire contract as the reference's
cmd/tuning/prometheus/metrics.py: snappy-compressed protobuf WriteRequest
POSTed to <addr>/api/v1/write, metric names train_metrics/eval_metrics,
labels uid/current_steps/total_steps/loss/learning_rate/epoch).

Self-contained: the protobuf wire format and a valid literal-only snappy
frame are emitted directly (the offline image has neither python-snappy
nor the generated prometheus_pb2).
"""

from __future__ import annotations

import struct
import time
from typing import Dict, List, Tuple


# ---------------------------------------------------------- protobuf wire
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


def _len_delim(field: int, payload: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(payload)) + payload


def _label(name: str, value: str) -> bytes:
    return (_len_delim(1, name.encode()) + _len_delim(2, value.encode()))


def _sample(value: float, ts_ms: int) -> bytes:
    return (_tag(1, 1) + struct.pack("<d", value) +
            _tag(2, 0) + _varint(ts_ms if ts_ms >= 0 else ts_ms + (1 << 64)))


def _timeseries(labels: List[Tuple[str, str]],
                samples: List[Tuple[float, int]]) -> bytes:
    body = b"".join(_len_delim(1, _label(n, v)) for n, v in labels)
    body += b"".join(_len_delim(2, _sample(v, t)) for v, t in samples)
    return body


def encode_write_request(series: List[Tuple[List[Tuple[str, str]],
                                            List[Tuple[float, int]]]]) -> bytes:
    return b"".join(_len_delim(1, _timeseries(lbls, smps))
                    for lbls, smps in series)


# ------------------------------------------------------------- snappy enc
def snappy_compress(data: bytes) -> bytes:
    """Valid snappy stream using literal-only encoding (spec-conformant;
    any snappy decoder accepts it)."""
    out = bytearray(_varint(len(data)))
    i = 0
    while i < len(data):
        chunk = data[i:i + 65536]
        n = len(chunk)
        if n <= 60:
            out.append((n - 1) << 2)
        elif n <= 0xFF:
            out.append(60 << 2)
            out.append(n - 1)
        else:
            out.append(61 << 2)
            out += struct.pack("<H", n - 1)
        out += chunk
        i += n
    return bytes(out)


# --------------------------------------------------------------- exporter
class RemoteWriteExporter:
    """POSTs metric dicts to <address>/api/v1/write (metrics.py:21-39
    parity). Failures are swallowed (training never blocks on metrics)."""

    def __init__(self, address: str, uid: str, timeout: float = 2.0):
        self.address = address.rstrip("/")
        self.uid = uid
        self.timeout = timeout

    def _post(self, payload: bytes) -> bool:
        import requests
        try:
            r = requests.post(
                self.address + "/api/v1/write",
                data=snappy_compress(payload),
                headers={
                    "Content-Type": "application/x-protobuf",
                    "Content-Encoding": "snappy",
                    "X-Prometheus-Remote-Write-Version": "0.1.0",
                },
                timeout=self.timeout)
            return r.status_code < 300
        except Exception:
            return False

    def export(self, metric_name: str, metrics: Dict[str, object]) -> bool:
        ts_ms = int(time.time() * 1000)
        labels = [("__name__", metric_name), ("uid", self.uid)]
        labels += [(k, str(v)) for k, v in sorted(metrics.items())]
        payload = encode_write_request([(labels, [(1.0, ts_ms)])])
        return self._post(payload)

    def export_train_metrics(self, metrics: Dict[str, object]) -> bool:
        return self.export("train_metrics", metrics)

    def export_eval_metrics(self, metrics: Dict[str, object]) -> bool:
        return self.export("eval_metrics", metrics)
