"""Wire-shape validation for /v1/chat/completions.

The reference parses requests into generated types
(providers/types/common_types.go — a type mismatch is a 400 at decode
time); this is the dict-level equivalent: blatant type errors on KNOWN
fields are a 400 before any provider work, unknown fields pass through
untouched (OpenAI clients send tools/response_format/user/... that the
gateway must tolerate).  The field set mirrors
components.schemas.CreateChatCompletionRequest in the generated
openapi.yaml; test_codegen_drift pins the n-range against the engine
cap."""
from __future__ import annotations

import numbers
from typing import Optional

_ROLES = {"system", "user", "assistant", "tool", "developer"}
_NUM_FIELDS = ("temperature", "top_p", "presence_penalty",
               "frequency_penalty")
_INT_FIELDS = ("max_tokens", "max_completion_tokens", "top_k", "n",
               "seed", "top_logprobs")
_BOOL_FIELDS = ("stream", "logprobs")


def _is_num(v) -> bool:
    return isinstance(v, numbers.Number) and not isinstance(v, bool)


def _is_int(v) -> bool:
    return isinstance(v, int) and not isinstance(v, bool)


def validate_chat_request(body: dict) -> Optional[str]:
    """Returns an error message (caller answers 400) or None."""
    msgs = body.get("messages")
    if not isinstance(msgs, list) or not msgs:
        return "messages must be a non-empty array"
    for i, m in enumerate(msgs):
        if not isinstance(m, dict):
            return f"messages[{i}] must be an object"
        role = m.get("role")
        if not isinstance(role, str) or role not in _ROLES:
            return (f"messages[{i}].role must be one of "
                    f"{sorted(_ROLES)}")
        c = m.get("content")
        if c is not None and not isinstance(c, (str, list)):
            return (f"messages[{i}].content must be a string, an array "
                    "of parts, or null")
        if isinstance(c, list):
            for j, part in enumerate(c):
                if not isinstance(part, dict):
                    return (f"messages[{i}].content[{j}] must be an "
                            "object")
    for f in _NUM_FIELDS:
        v = body.get(f)
        if v is not None and not _is_num(v):
            return f"{f} must be a number"
    for f in _INT_FIELDS:
        v = body.get(f)
        if v is not None and not _is_int(v):
            return f"{f} must be an integer"
    for f in _BOOL_FIELDS:
        v = body.get(f)
        if v is not None and not isinstance(v, bool):
            return f"{f} must be a boolean"
    stop = body.get("stop")
    if stop is not None and not isinstance(stop, (str, list)):
        return "stop must be a string or an array of strings"
    if isinstance(stop, list):
        if len(stop) > 4:
            return "stop supports at most 4 sequences"
        for s in stop:
            if not isinstance(s, str):
                return "stop must contain only strings"
    tools = body.get("tools")
    if tools is not None and not isinstance(tools, list):
        return "tools must be an array"
    lb = body.get("logit_bias")
    if lb is not None:
        if not isinstance(lb, dict):
            return "logit_bias must be an object"
        if len(lb) > 300:
            return "logit_bias supports at most 300 entries"
        for k, v in lb.items():
            try:
                int(k)
            except (TypeError, ValueError):
                return "logit_bias keys must be token ids"
            if not _is_num(v):
                return "logit_bias values must be numbers"
    return None
