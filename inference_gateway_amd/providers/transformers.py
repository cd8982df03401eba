"""ListModels response transformers.

Parity with the reference's generated providers/transformers/*.go (all 15
are the same shape, openai.go:14-28): parse the provider's /models response
as an OpenAI-style list, prefix every model id with '<provider>/', and stamp
served_by.
"""
from __future__ import annotations

from typing import List


def transform_list_models(provider_id: str, payload: dict) -> dict:
    data = payload.get("data") or []
    if not data and isinstance(payload.get("models"), list):
        # Cohere's native /v1/models shape ({"models": [{"name": ...}]})
        # — the reference's generated transformer reads only data/id and
        # returns an empty list there; tolerate the native shape instead
        data = [{**m, "id": m.get("id") or m.get("name", "")}
                for m in payload["models"] if isinstance(m, dict)]
    models: List[dict] = []
    for m in data:
        m = dict(m)
        m["id"] = f"{provider_id}/{m.get('id', '')}"
        m["served_by"] = provider_id
        m.setdefault("object", "model")
        m.setdefault("created", 0)
        m.setdefault("owned_by", provider_id)
        models.append(m)
    return {
        "provider": provider_id,
        "object": payload.get("object", "list"),
        "data": models,
    }
