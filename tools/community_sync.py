#!/usr/bin/env python3
"""Sync community model-metadata tables from a models.dev-style dump.

The analog of the reference's internal/communitygen (which parses a
models.dev tarball into committed go:embed JSON, merged with a
hand-maintained .overrides.json).  This environment has no network, so
the committed tables under inference_gateway_amd/data/ are a curated
snapshot; when a dump IS available run:

    python tools/community_sync.py path/to/models_dev_api.json

Dump shape (models.dev /api.json): {provider: {models: {id: {limit:
{context}, cost: {input, output}, modalities: {input, output}}}}}.
Values land in community_{context_windows,pricing,modalities}.json;
community.overrides.json is merged LAST by the loader
(providers/metadata.py) and is never touched here.
"""
from __future__ import annotations

import json
import sys
from pathlib import Path

DATA = Path(__file__).resolve().parent.parent / "inference_gateway_amd" / \
    "data"


def sync(dump_path: str) -> None:
    dump = json.loads(Path(dump_path).read_text())
    cw, pricing, modal = {}, {}, {}
    for provider, pdata in dump.items():
        for mid, m in (pdata.get("models") or {}).items():
            key = f"{provider}/{mid}"
            ctx = (m.get("limit") or {}).get("context")
            if ctx:
                cw[key] = int(ctx)
            cost = m.get("cost") or {}
            if "input" in cost or "output" in cost:
                # models.dev costs are $ per 1M tokens; store $ per token
                pricing[key] = {
                    "input": float(cost.get("input", 0)) / 1e6,
                    "output": float(cost.get("output", 0)) / 1e6,
                }
            mods = m.get("modalities") or {}
            if mods:
                modal[key] = {"input": mods.get("input") or ["text"],
                              "output": mods.get("output") or ["text"]}
    for name, table in (("community_context_windows", cw),
                        ("community_pricing", pricing),
                        ("community_modalities", modal)):
        path = DATA / f"{name}.json"
        existing = json.loads(path.read_text()) if path.exists() else {}
        existing.update(table)
        path.write_text(json.dumps(existing, indent=1, sort_keys=True))
        print(f"{name}: {len(existing)} entries")


if __name__ == "__main__":
    if len(sys.argv) != 2:
        print(__doc__)
        sys.exit(1)
    sync(sys.argv[1])
