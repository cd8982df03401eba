"""Minimal OTLP/HTTP protobuf decoder for the metrics-ingestion subset.

The reference accepts both encodings (api/metrics.go:25-99); OTel SDK
default exporters send protobuf, so a JSON-only endpoint breaks the most
common client.  There is no protobuf runtime schema
here — this decodes the proto3 wire format directly for exactly the
message shapes the ingestion path consumes
(opentelemetry/proto/metrics/v1/metrics.proto):

  ExportMetricsServiceRequest{ repeated ResourceMetrics resource_metrics=1 }
  ResourceMetrics { repeated ScopeMetrics scope_metrics = 2 }
  ScopeMetrics    { repeated Metric metrics = 2 }
  Metric          { string name = 1; Histogram histogram = 9 }
  Histogram       { repeated HistogramDataPoint data_points = 1;
                    AggregationTemporality aggregation_temporality = 2 }
  HistogramDataPoint { repeated KeyValue attributes = 9;
                       repeated uint64 bucket_counts = 6 (packed);
                       repeated double explicit_bounds = 7 (packed) }
  KeyValue        { string key = 1; AnyValue value = 2 }
  AnyValue        { string string_value = 1 }

The output is the SAME dict shape the OTLP/JSON path parses, so both
encodings share one ingestion code path (otel/ingest.py)."""
from __future__ import annotations

import struct
from typing import Iterator, Tuple


class ProtoError(ValueError):
    pass


def _varint(buf: bytes, i: int) -> Tuple[int, int]:
    out = 0
    shift = 0
    while True:
        if i >= len(buf):
            raise ProtoError("truncated varint")
        b = buf[i]
        i += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, i
        shift += 7
        if shift > 63:
            raise ProtoError("varint too long")


def _fields(buf: bytes) -> Iterator[Tuple[int, int, object]]:
    """Yield (field_number, wire_type, value) over a message buffer.
    wire types: 0 varint, 1 fixed64, 2 length-delimited, 5 fixed32."""
    i = 0
    n = len(buf)
    while i < n:
        tag, i = _varint(buf, i)
        fno, wt = tag >> 3, tag & 7
        if wt == 0:
            v, i = _varint(buf, i)
        elif wt == 1:
            if i + 8 > n:
                raise ProtoError("truncated fixed64")
            v = buf[i:i + 8]
            i += 8
        elif wt == 2:
            ln, i = _varint(buf, i)
            if i + ln > n:
                raise ProtoError("truncated bytes")
            v = buf[i:i + ln]
            i += ln
        elif wt == 5:
            if i + 4 > n:
                raise ProtoError("truncated fixed32")
            v = buf[i:i + 4]
            i += 4
        else:
            raise ProtoError(f"unsupported wire type {wt}")
        yield fno, wt, v


def _packed_varints(buf: bytes) -> list:
    out = []
    i = 0
    while i < len(buf):
        v, i = _varint(buf, i)
        out.append(v)
    return out


def _packed_doubles(buf: bytes) -> list:
    if len(buf) % 8:
        raise ProtoError("bad packed doubles")
    return list(struct.unpack(f"<{len(buf) // 8}d", buf))


def _any_value(buf: bytes) -> dict:
    for fno, wt, v in _fields(buf):
        if fno == 1 and wt == 2:
            return {"stringValue": v.decode("utf-8", "replace")}
    return {}


def _key_value(buf: bytes) -> dict:
    out = {"key": "", "value": {}}
    for fno, wt, v in _fields(buf):
        if fno == 1 and wt == 2:
            out["key"] = v.decode("utf-8", "replace")
        elif fno == 2 and wt == 2:
            out["value"] = _any_value(v)
    return out


def _histogram_dp(buf: bytes) -> dict:
    dp = {"attributes": [], "bucketCounts": [], "explicitBounds": []}
    for fno, wt, v in _fields(buf):
        if fno == 9 and wt == 2:
            dp["attributes"].append(_key_value(v))
        elif fno == 6:
            if wt == 2:
                dp["bucketCounts"].extend(_packed_varints(v))
            elif wt == 0:
                dp["bucketCounts"].append(v)
        elif fno == 7:
            if wt == 2:
                dp["explicitBounds"].extend(_packed_doubles(v))
            elif wt == 1:
                dp["explicitBounds"].append(struct.unpack("<d", v)[0])
    return dp


def _histogram(buf: bytes) -> dict:
    h = {"dataPoints": [], "aggregationTemporality": 0}
    for fno, wt, v in _fields(buf):
        if fno == 1 and wt == 2:
            h["dataPoints"].append(_histogram_dp(v))
        elif fno == 2 and wt == 0:
            h["aggregationTemporality"] = v
    return h


def _metric(buf: bytes) -> dict:
    m = {"name": ""}
    for fno, wt, v in _fields(buf):
        if fno == 1 and wt == 2:
            m["name"] = v.decode("utf-8", "replace")
        elif fno == 9 and wt == 2:
            m["histogram"] = _histogram(v)
    return m


def _scope_metrics(buf: bytes) -> dict:
    sm = {"metrics": []}
    for fno, wt, v in _fields(buf):
        if fno == 2 and wt == 2:
            sm["metrics"].append(_metric(v))
    return sm


def _resource_metrics(buf: bytes) -> dict:
    rm = {"scopeMetrics": []}
    for fno, wt, v in _fields(buf):
        if fno == 2 and wt == 2:
            rm["scopeMetrics"].append(_scope_metrics(v))
    return rm


def decode_export_metrics(buf: bytes) -> dict:
    """ExportMetricsServiceRequest bytes -> the OTLP/JSON dict shape."""
    out = {"resourceMetrics": []}
    for fno, wt, v in _fields(buf):
        if fno == 1 and wt == 2:
            out["resourceMetrics"].append(_resource_metrics(v))
    return out
