"""OTLP metrics push ingestion (reference POST /v1/metrics,
api/metrics.go:25-99 + otel/ingest.go:22-218).

Accepts OTLP/HTTP in BOTH encodings — application/x-protobuf (the OTel
SDK default; hand-rolled wire decoder in otlp_pb.py) and
application/json — gzip-aware, 4 MiB cap. Semantics carried over:
attribute allowlist, delta-temporality-only for sums/histograms,
histogram replay via bucket midpoints capped at 10k observations per
datapoint, and source spoof protection (the `source` attribute is
overwritten from the authenticated caller when auth is enabled).
"""
from __future__ import annotations

import gzip
import json

from ..gateway.http import Request, Response

MAX_BODY = 4 * 1024 * 1024
MAX_OBSERVATIONS = 10_000

_ALLOWED_ATTRS = {"source", "team", "provider", "model", "operation"}
_INGESTIBLE = {
    "gen_ai.client.operation.duration": "client_operation_duration",
    "gen_ai.client.time_to_first_chunk": "client_time_to_first_chunk",
}


def _attrs(dp: dict) -> dict:
    out = {}
    for kv in dp.get("attributes") or []:
        k = kv.get("key", "")
        if k in _ALLOWED_ATTRS:
            v = kv.get("value") or {}
            out[k] = v.get("stringValue", "")
    return out


class MetricsIngestion:
    def __init__(self, telemetry, logger):
        self.telemetry = telemetry
        self.logger = logger

    async def handle(self, req: Request) -> Response:
        body = req.body
        if len(body) > MAX_BODY:
            return Response.error(413, "payload too large")
        if req.headers.get("content-encoding", "") == "gzip":
            try:
                body = gzip.decompress(body)
            except Exception:
                return Response.error(400, "bad gzip payload")
            if len(body) > MAX_BODY:
                return Response.error(413, "payload too large")
        ct = req.headers.get("content-type", "")
        if "protobuf" in ct:
            # OTLP/HTTP protobuf — the OTel SDK default exporter
            # encoding (hand-rolled wire decoder, otlp_pb.py; decodes
            # into the same dict shape the JSON path parses)
            from .otlp_pb import ProtoError, decode_export_metrics

            try:
                payload = decode_export_metrics(body)
            except ProtoError as e:
                return Response.error(400, f"invalid protobuf: {e}")
        elif "json" in ct:
            try:
                payload = json.loads(body)
            except Exception:
                return Response.error(400, "invalid JSON")
        else:
            return Response.error(
                415, "unsupported content type: use application/json "
                     "or application/x-protobuf OTLP")

        # source spoof protection (otel/ingest.go): authenticated callers
        # cannot claim another source
        auth_source = None
        claims = req.state.get("auth_claims")
        if claims:
            auth_source = (claims.get("azp") or claims.get("client_id")
                           or claims.get("sub"))

        accepted = rejected = 0
        for rm in payload.get("resourceMetrics") or []:
            for sm in rm.get("scopeMetrics") or []:
                for metric in sm.get("metrics") or []:
                    name = metric.get("name", "")
                    target = _INGESTIBLE.get(name)
                    if target is None:
                        rejected += 1
                        continue
                    hist = metric.get("histogram") or {}
                    # delta temporality only (2 = DELTA in OTLP enums)
                    temporality = hist.get("aggregationTemporality")
                    if temporality not in (2, "AGGREGATION_TEMPORALITY_DELTA"):
                        rejected += 1
                        continue
                    inst = getattr(self.telemetry, target)
                    for dp in hist.get("dataPoints") or []:
                        attrs = _attrs(dp)
                        if auth_source:
                            attrs["source"] = auth_source
                        labels = (attrs.get("provider", ""),
                                  attrs.get("model", ""),
                                  attrs.get("source", ""),
                                  attrs.get("team", ""))
                        try:
                            counts = [int(c) for c in
                                      dp.get("bucketCounts") or []]
                            bounds = [float(b) for b in
                                      dp.get("explicitBounds") or []]
                        except (TypeError, ValueError):
                            rejected += 1     # malformed datapoint, not
                            continue          # a 500
                        if any(c < 0 for c in counts):
                            rejected += 1
                            continue
                        n = 0
                        for i, c in enumerate(counts):
                            if n >= MAX_OBSERVATIONS:
                                break
                            # midpoint replay (otel/ingest.go)
                            if i == 0:
                                mid = bounds[0] / 2 if bounds else 0.0
                            elif i < len(bounds):
                                mid = (bounds[i - 1] + bounds[i]) / 2
                            else:
                                mid = bounds[-1] * 1.5 if bounds else 0.0
                            for _ in range(min(c, MAX_OBSERVATIONS - n)):
                                inst.labels(*labels).observe(mid)
                                n += 1
                        accepted += 1
        return Response.json({"accepted": accepted, "rejected": rejected})
