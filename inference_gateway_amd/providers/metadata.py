"""Model metadata enrichment: context windows, pricing, modalities.

Re-implements the reference's three-source resolution for
`GET /v1/models?include=...` (api/context_window.go:28-137,
providers/core/context_window.go:40-57, community_context_window.go,
pricing.go, community_modalities.go):

  precedence: runtime lookup > provider-published > community table

Community tables are committed JSON under inference_gateway_amd/data/
(the analog of the reference's go:embed models.dev sync,
internal/communitygen/), with a hand-maintained
community.overrides.json merged last.  tools/community_sync.py
regenerates the tables from a models.dev-style dump.

Runtime lookups hit self-hosted admin endpoints — llama.cpp `GET /props`
(default_generation_settings.n_ctx) and Ollama `POST /api/show`
(model_info "<arch>.context_length") — bounded by a 4-way semaphore like
the reference (api/context_window.go:20-44).
"""
from __future__ import annotations

import asyncio
import json
from functools import lru_cache
from pathlib import Path
from typing import Dict, List, Optional

_DATA = Path(__file__).resolve().parent.parent / "data"

INCLUDE_KEYS = ("context_window", "pricing", "modalities")
_RUNTIME_SEMAPHORE = 4


@lru_cache(maxsize=1)
def _tables() -> Dict[str, dict]:
    out = {}
    for key, fname in (("context_windows", "community_context_windows.json"),
                       ("pricing", "community_pricing.json"),
                       ("modalities", "community_modalities.json")):
        try:
            out[key] = json.loads((_DATA / fname).read_text())
        except OSError:
            out[key] = {}
    try:
        overrides = json.loads((_DATA / "community.overrides.json").read_text())
    except OSError:
        overrides = {}
    for key in out:
        out[key].update(overrides.get(key) or {})
    return out


def _lookup(table: Dict, model_id: str):
    """Exact id first, then the bare model name under any provider prefix."""
    if model_id in table:
        return table[model_id]
    bare = model_id.split("/", 1)[-1]
    for k, v in table.items():
        if k.split("/", 1)[-1] == bare:
            return v
    return None


def community_context_window(model_id: str) -> Optional[int]:
    return _lookup(_tables()["context_windows"], model_id)


def community_pricing(model_id: str) -> Optional[dict]:
    return _lookup(_tables()["pricing"], model_id)


def community_modalities(model_id: str) -> Optional[dict]:
    return _lookup(_tables()["modalities"], model_id)


def model_accepts_images(model_id: str) -> bool:
    """Vision gate helper (reference community_modalities.go:28
    ModelAcceptsImages): unknown models default to True (don't strip)."""
    m = community_modalities(model_id)
    if m is None:
        return True
    return "image" in (m.get("input") or [])


async def _llamacpp_n_ctx(provider) -> Optional[int]:
    try:
        r = await provider.forward_raw("GET", "/props", b"", {})
        if r.status_code != 200:
            return None
        props = r.json()
        return int(props.get("default_generation_settings", {}).get("n_ctx"))
    except Exception:
        return None


async def _ollama_context_length(provider, model: str) -> Optional[int]:
    try:
        r = await provider.forward_raw(
            "POST", "/api/show", json.dumps({"model": model}).encode(),
            {"content-type": "application/json"})
        if r.status_code != 200:
            return None
        info = r.json().get("model_info") or {}
        for k, v in info.items():
            if k.endswith(".context_length"):
                return int(v)
        return None
    except Exception:
        return None


async def resolve_runtime_context_windows(
        models: List[dict], registry) -> Dict[str, int]:
    """Per-model runtime lookups for self-hosted providers, semaphore-bounded
    (reference resolveContextWindows, api/context_window.go:28-137)."""
    sem = asyncio.Semaphore(_RUNTIME_SEMAPHORE)
    out: Dict[str, int] = {}

    async def one(m: dict) -> None:
        mid = m.get("id", "")
        pid = mid.split("/", 1)[0] if "/" in mid else (m.get("served_by") or "")
        if pid not in ("llamacpp", "ollama"):
            return
        async with sem:
            try:
                provider = registry.build(pid)
            except Exception:
                return
            if pid == "llamacpp":
                n = await _llamacpp_n_ctx(provider)
            else:
                n = await _ollama_context_length(provider,
                                                 mid.split("/", 1)[-1])
            if n:
                out[mid] = n

    await asyncio.gather(*(one(m) for m in models))
    return out


async def enrich_models(models: List[dict], include: List[str],
                        registry) -> List[dict]:
    """Apply the requested include= keys; requested-but-unknown values are
    rendered as explicit null (reference routes.go:373-424)."""
    include = [k for k in include if k in INCLUDE_KEYS]
    if not include:
        return models
    runtime_cw: Dict[str, int] = {}
    if "context_window" in include:
        runtime_cw = await resolve_runtime_context_windows(models, registry)
    out = []
    for m in models:
        m = dict(m)
        mid = m.get("id", "")
        if "context_window" in include:
            cw = (runtime_cw.get(mid) or m.get("context_window")
                  or community_context_window(mid))
            m["context_window"] = cw if cw else None
        if "pricing" in include:
            m["pricing"] = m.get("pricing") or community_pricing(mid)
        if "modalities" in include:
            m["modalities"] = m.get("modalities") or community_modalities(mid)
        out.append(m)
    return out
