"""Adversarial-input robustness (hypothesis): the wire-facing parsers
must never crash, hang, or corrupt connection state on arbitrary bytes.
The reference gets most of this from Go's stdlib parsers; the native
server and hand-rolled decoders here carry the same obligation.

Each property is bounded (max_examples) to keep the suite fast; the
deadline is disabled because the first example pays import costs."""
import asyncio
import json
import struct

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from inference_gateway_amd.gateway.validate import validate_chat_request

_fast = settings(max_examples=120, deadline=None, derandomize=True,
                 suppress_health_check=[HealthCheck.too_slow])


# ---- wire validator: total over arbitrary JSON-ish dicts ---------------
_json_vals = st.recursive(
    st.none() | st.booleans() | st.integers() | st.floats(
        allow_nan=False) | st.text(max_size=8),
    lambda ch: st.lists(ch, max_size=4) | st.dictionaries(
        st.text(max_size=8), ch, max_size=4),
    max_leaves=12)


@_fast
@given(st.dictionaries(st.sampled_from(
    ["messages", "temperature", "top_p", "top_k", "max_tokens", "stop",
     "stream", "logprobs", "top_logprobs", "n", "seed", "logit_bias",
     "tools", "presence_penalty", "model", "whatever"]),
    _json_vals, max_size=8))
def test_validator_is_total(body):
    out = validate_chat_request(body)
    assert out is None or isinstance(out, str)


# ---- chunked request decoder: any split of a valid stream decodes ------
class _FakeServer:
    max_body = 1 << 20
    header_timeout = 30.0
    idle_timeout = 0
    is_tls = False


def _proto():
    from inference_gateway_amd.gateway.hserver import HttpProtocol

    p = HttpProtocol(None, asyncio.new_event_loop(), _FakeServer())
    p._chunked_req = True
    p._head_done = True
    return p


@_fast
@given(st.binary(min_size=0, max_size=200),
       st.lists(st.integers(min_value=1, max_value=40), min_size=1,
                max_size=8),
       st.integers(min_value=1, max_value=7))
def test_chunked_decoder_reassembles_any_split(payload, cuts, nfeeds):
    # encode payload as chunks sized by `cuts` (cycled), then feed the
    # encoded bytes to the decoder in `nfeeds` arbitrary pieces
    enc = bytearray()
    i = 0
    ci = 0
    while i < len(payload):
        n = min(cuts[ci % len(cuts)], len(payload) - i)
        ci += 1
        enc += b"%x\r\n" % n + payload[i: i + n] + b"\r\n"
        i += n
    enc += b"0\r\n\r\n"
    p = _proto()
    step = max(1, len(enc) // nfeeds)
    done = None
    for off in range(0, len(enc), step):
        p._buf += bytes(enc[off: off + step])
        done = p._consume_chunked()
        if done is True:
            break
        assert done is None, "well-formed stream must not be rejected"
    assert done is True
    assert p._body == payload


@_fast
@given(st.binary(min_size=0, max_size=120))
def test_chunked_decoder_never_crashes_on_garbage(garbage):
    p = _proto()
    p._buf = garbage
    # may need more data (None), succeed (True) or reject (False) —
    # never raise, never loop forever (find() bounds each iteration)
    assert p._consume_chunked() in (None, True, False)


# ---- request-head parser: arbitrary bytes -> ValueError or a scope -----
@_fast
@given(st.binary(min_size=0, max_size=160))
def test_head_parser_never_crashes(raw):
    p = _proto()

    class _T:
        def get_extra_info(self, k):
            return None

        def write(self, data):      # 100-continue interim writes
            pass

    p.transport = _T()
    try:
        p._parse_head(raw)
    except ValueError:
        pass                      # 400 path
    else:
        assert p._scope is not None


# ---- GGUF parser: truncations/corruptions raise, never hang ------------
def _valid_gguf() -> bytes:
    b = b"GGUF" + struct.pack("<IQQ", 3, 1, 1)
    key = b"general.architecture"
    b += struct.pack("<Q", len(key)) + key
    b += struct.pack("<I", 8)
    b += struct.pack("<Q", 5) + b"llama"
    name = b"t"
    b += struct.pack("<Q", len(name)) + name
    b += struct.pack("<I", 1) + struct.pack("<Q", 8)
    b += struct.pack("<IQ", 0, 0)
    b += b"\x00" * ((-len(b)) % 32)
    b += struct.pack("<8f", *range(8))
    return b


@_fast
@given(st.integers(min_value=0, max_value=120),
       st.binary(min_size=0, max_size=16))
def test_gguf_parser_survives_corruption(cut, junk):
    import tempfile
    from pathlib import Path

    from inference_gateway_amd.engine import gguf as G

    blob = _valid_gguf()
    mutated = blob[:cut] + junk + blob[cut + len(junk):]
    f = Path(tempfile.mkdtemp()) / "m.gguf"
    f.write_bytes(mutated)
    try:
        g = G.GGUFFile(str(f))
        g.tensor("t")
    except (ValueError, KeyError, OverflowError, MemoryError,
            struct.error, UnicodeDecodeError):
        pass                      # loud failure is the contract


# ---- OTLP protobuf decoder: arbitrary bytes never crash the server ----
@_fast
@given(st.binary(min_size=0, max_size=200))
def test_otlp_pb_decoder_total(raw):
    from inference_gateway_amd.otel.otlp_pb import (ProtoError,
                                                    decode_export_metrics)

    try:
        out = decode_export_metrics(raw)
    except ProtoError:
        return
    json.dumps(out)               # result must be JSON-serializable


# ---- Q8_0/Q4_0 dequant: arbitrary block bytes never crash --------------
@_fast
@given(st.binary(min_size=34, max_size=34 * 4))
def test_q8_dequant_total(raw):
    import numpy as np

    from inference_gateway_amd.engine.gguf import _dequant, GGML_Q8_0

    n = (len(raw) // 34) * 32
    out = _dequant(np.frombuffer(raw, dtype=np.uint8), GGML_Q8_0, n)
    assert out.shape[0] == n


# ---- config duration parser: total, non-negative ----------------------
@_fast
@given(st.text(max_size=16))
def test_parse_duration_total(sraw):
    from inference_gateway_amd.config import parse_duration

    try:
        v = parse_duration(sraw)
    except ValueError:
        return
    assert isinstance(v, float)


@_fast
@given(st.integers(min_value=0, max_value=10**6),
       st.sampled_from(["ms", "s", "m", "h"]))
def test_parse_duration_units(n, unit):
    from inference_gateway_amd.config import parse_duration

    mult = {"ms": 0.001, "s": 1, "m": 60, "h": 3600}[unit]
    assert abs(parse_duration(f"{n}{unit}") - n * mult) < 1e-6 * max(
        1, n * mult)


# ---- model routing: never raises, prefix split is consistent ----------
@_fast
@given(st.text(max_size=40))
def test_model_routing_total(model):
    from inference_gateway_amd.providers.routing import (
        determine_provider_and_model)

    pid, m = determine_provider_and_model(model)
    if pid is None:
        assert m == model
    else:
        assert model.lower().startswith(pid + "/")
        assert model.endswith(m)


# ---- MCP tool-name normalization idempotent ---------------------------
@_fast
@given(st.text(max_size=24))
def test_tool_name_normalization_idempotent(name):
    from inference_gateway_amd.mcp.client import normalize_tool_name

    once = normalize_tool_name(name)
    assert normalize_tool_name(once) == once
