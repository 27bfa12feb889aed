"""Wire-format tests for the hand-rolled Prometheus remote-write
(snappy + protobuf WriteRequest — metrics/remote_write.py). The test
implements its OWN snappy literal decoder and protobuf walker from the
public specs, so encoder bugs can't hide behind a shared
implementation."""

import struct
import threading

from datatunerx_amd.metrics.remote_write import (RemoteWriteExporter,
                                                 encode_write_request,
                                                 snappy_compress)


def snappy_decompress_literalonly(buf: bytes) -> bytes:
    """Independent decoder for the snappy format restricted to literal
    elements (the only kind our encoder emits)."""
    # preamble: uncompressed length varint
    n, shift, i = 0, 0, 0
    while True:
        b = buf[i]
        n |= (b & 0x7F) << shift
        i += 1
        if not b & 0x80:
            break
        shift += 7
    out = bytearray()
    while i < len(buf):
        tag = buf[i]
        i += 1
        assert tag & 3 == 0, "non-literal element"
        ln = tag >> 2
        if ln < 60:
            length = ln + 1
        elif ln == 60:
            length = buf[i] + 1
            i += 1
        elif ln == 61:
            length = struct.unpack("<H", buf[i:i + 2])[0] + 1
            i += 2
        else:
            raise AssertionError("3/4-byte literal lengths unused")
        out += buf[i:i + length]
        i += length
    assert len(out) == n
    return bytes(out)


def pb_walk(buf: bytes):
    """Yield (field, wire, value) from a protobuf message body."""
    i = 0
    while i < len(buf):
        key, shift = 0, 0
        while True:
            b = buf[i]
            key |= (b & 0x7F) << shift
            i += 1
            if not b & 0x80:
                break
            shift += 7
        field, wire = key >> 3, key & 7
        if wire == 0:                      # varint
            v, shift = 0, 0
            while True:
                b = buf[i]
                v |= (b & 0x7F) << shift
                i += 1
                if not b & 0x80:
                    break
                shift += 7
            yield field, wire, v
        elif wire == 1:                    # 64-bit
            yield field, wire, buf[i:i + 8]
            i += 8
        elif wire == 2:                    # length-delimited
            ln, shift = 0, 0
            while True:
                b = buf[i]
                ln |= (b & 0x7F) << shift
                i += 1
                if not b & 0x80:
                    break
                shift += 7
            yield field, wire, buf[i:i + ln]
            i += ln
        else:
            raise AssertionError(f"unexpected wire type {wire}")


def decode_write_request(buf: bytes):
    series = []
    for f, w, ts in pb_walk(buf):
        assert f == 1 and w == 2
        labels, samples = [], []
        for f2, w2, v2 in pb_walk(ts):
            if f2 == 1:                    # Label
                d = {ff: vv for ff, _, vv in pb_walk(v2)}
                labels.append((d[1].decode(), d[2].decode()))
            elif f2 == 2:                  # Sample
                val = ts_ms = None
                for ff, ww, vv in pb_walk(v2):
                    if ff == 1:
                        val = struct.unpack("<d", vv)[0]
                    elif ff == 2:
                        ts_ms = vv
                samples.append((val, ts_ms))
        series.append((labels, samples))
    return series


def test_snappy_roundtrip_various_sizes():
    for size in (0, 1, 59, 60, 61, 255, 256, 70000, 200001):
        data = bytes(range(256)) * (size // 256 + 1)
        data = data[:size]
        assert snappy_decompress_literalonly(snappy_compress(data)) == \
            data, size


def test_write_request_wire_format():
    series = [([("__name__", "train_metrics"), ("uid", "abc"),
                ("loss", "1.25")], [(1.0, 1700000000123)]),
              ([("__name__", "eval_metrics")], [(2.5, 42)])]
    got = decode_write_request(encode_write_request(series))
    assert got == series


def test_exporter_posts_decodable_payload():
    """End-to-end: the exporter's POST body snappy-decompresses and
    protobuf-decodes into the labeled series, with the remote-write
    headers set."""
    from http.server import BaseHTTPRequestHandler, HTTPServer
    seen = {}

    class H(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_POST(self):
            n = int(self.headers["Content-Length"])
            seen["path"] = self.path
            seen["body"] = self.rfile.read(n)
            seen["enc"] = self.headers.get("Content-Encoding")
            seen["ct"] = self.headers.get("Content-Type")
            seen["ver"] = self.headers.get(
                "X-Prometheus-Remote-Write-Version")
            self.send_response(200)
            self.end_headers()

    httpd = HTTPServer(("127.0.0.1", 0), H)
    port = httpd.server_address[1]
    t = threading.Thread(target=httpd.handle_request, daemon=True)
    t.start()
    ex = RemoteWriteExporter(f"http://127.0.0.1:{port}", uid="u1")
    ok = ex.export_train_metrics({"loss": 0.5, "current_steps": 3})
    t.join(timeout=30)
    httpd.server_close()
    assert ok
    assert seen["path"] == "/api/v1/write"
    assert seen["enc"] == "snappy"
    assert seen["ct"] == "application/x-protobuf"
    assert seen["ver"] == "0.1.0"
    series = decode_write_request(
        snappy_decompress_literalonly(seen["body"]))
    assert len(series) == 1
    labels = dict(series[0][0])
    assert labels["__name__"] == "train_metrics"
    assert labels["uid"] == "u1"
    assert labels["loss"] == "0.5"
    assert series[0][1][0][0] == 1.0       # sample value


def test_exporter_swallows_connection_failure():
    ex = RemoteWriteExporter("http://127.0.0.1:9", uid="x", timeout=0.3)
    assert ex.export_train_metrics({"loss": 1}) is False
