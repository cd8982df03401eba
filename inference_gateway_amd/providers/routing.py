"""Model routing: provider-prefix resolution, allow/deny filtering, and
opt-in alias pools with per-replica round-robin.

Behavior parity with the reference's providers/routing package
(model_mapping.go:19-31, model_filter.go:10-67, pool.go:39-118): explicit
`provider/model` prefixes only (no implicit name-based routing); a non-empty
allow list wins over the deny list; pool selection is round-robin with
per-replica (per-process) cursors.
"""
from __future__ import annotations

import itertools
import threading
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import yaml

from .constants import LOCAL_PROVIDER_ID, PROVIDERS

STRATEGY_ROUND_ROBIN = "round_robin"


def determine_provider_and_model(model: str,
                                 known_providers=None) -> Tuple[Optional[str], str]:
    """'groq/llama-3' -> ('groq', 'llama-3'); unknown prefix -> (None, model)."""
    if "/" not in model:
        return None, model
    prefix, rest = model.split("/", 1)
    pid = prefix.lower()
    known = known_providers if known_providers is not None else PROVIDERS
    if pid in known or pid == LOCAL_PROVIDER_ID:
        return pid, rest
    return None, model


def parse_model_set(csv) -> set:
    if isinstance(csv, str):
        items = csv.split(",")
    else:
        items = csv or []
    return {x.strip().lower() for x in items if x.strip()}


def model_matches(model_set: set, model_id: str) -> bool:
    mid = model_id.lower()
    if mid in model_set:
        return True
    if "/" in mid and mid.split("/", 1)[1] in model_set:
        return True
    return False


def filter_models(models: List[dict], allowed, disallowed) -> List[dict]:
    """models: list of Model dicts with 'id'. Allow list wins over deny."""
    allowed_set = parse_model_set(allowed)
    if allowed_set:
        return [m for m in models if model_matches(allowed_set, m["id"])]
    disallowed_set = parse_model_set(disallowed)
    if disallowed_set:
        return [m for m in models if not model_matches(disallowed_set, m["id"])]
    return models


def model_denied(model_id: str, allowed, disallowed) -> bool:
    allowed_set = parse_model_set(allowed)
    if allowed_set:
        return not model_matches(allowed_set, model_id)
    disallowed_set = parse_model_set(disallowed)
    if disallowed_set:
        return model_matches(disallowed_set, model_id)
    return False


@dataclass
class Deployment:
    provider: str
    model: str


class _Pool:
    def __init__(self, deployments: List[Deployment]):
        self.deployments = deployments
        self._counter = itertools.count()
        self._lock = threading.Lock()

    def next(self) -> Deployment:
        with self._lock:
            i = next(self._counter)
        return self.deployments[i % len(self.deployments)]


class Selector:
    """Round-robin alias pools; state is per process (per replica)."""

    def __init__(self, pools: Dict[str, _Pool]):
        self._pools = pools

    def select(self, alias: str) -> Optional[Deployment]:
        p = self._pools.get(alias)
        return p.next() if p else None

    def aliases(self) -> List[str]:
        return sorted(self._pools)


def load_pools_config(path: str) -> dict:
    with open(path) as f:
        return yaml.safe_load(f) or {}


def new_selector(cfg: dict, known_providers=None) -> Selector:
    models = (cfg or {}).get("models") or {}
    if not models:
        raise ValueError("routing enabled but no models configured")
    known = known_providers if known_providers is not None else PROVIDERS
    pools = {}
    for alias, pc in models.items():
        strategy = (pc or {}).get("strategy") or STRATEGY_ROUND_ROBIN
        if strategy != STRATEGY_ROUND_ROBIN:
            raise ValueError(
                f"model {alias!r}: unsupported strategy {strategy!r} "
                f"(only {STRATEGY_ROUND_ROBIN!r} is supported)")
        deps = (pc or {}).get("deployments") or []
        if len(deps) < 2:
            raise ValueError(
                f"model {alias!r}: round-robin requires at least 2 "
                f"deployments, got {len(deps)}")
        built = []
        for i, d in enumerate(deps):
            prov, model = d.get("provider"), d.get("model")
            if not prov or not model:
                raise ValueError(
                    f"model {alias!r} deployment {i}: provider and model "
                    "are required")
            if prov not in known and prov != LOCAL_PROVIDER_ID:
                raise ValueError(
                    f"model {alias!r} deployment {i}: unknown provider "
                    f"{prov!r}")
            built.append(Deployment(prov, model))
        pools[alias] = _Pool(built)
    return Selector(pools)
