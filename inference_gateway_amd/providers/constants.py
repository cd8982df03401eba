"""Provider registry data: the 15 remote providers of the reference
(providers/constants/constants.go:9-82, providers/registry/registry_data.go)
plus the net-new in-process `local` MI355X engine provider.

Auth types (constants/static.go): bearer | xheader | query | none.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict


@dataclass(frozen=True)
class ProviderMeta:
    id: str
    display_name: str
    url: str
    auth_type: str                      # bearer | xheader | query | none
    models_endpoint: str = "/models"
    chat_endpoint: str = "/chat/completions"
    extra_headers: tuple = ()           # ((name, value), ...)


AUTH_BEARER = "bearer"
AUTH_XHEADER = "xheader"
AUTH_QUERY = "query"
AUTH_NONE = "none"

LOCAL_PROVIDER_ID = "local"

# registry data is GENERATED from spec/gateway.yaml (tools/generate.py),
# mirroring the reference's openapi.yaml -> registry_data.go pipeline
from .constants_gen import PROVIDER_TABLE  # noqa: E402

PROVIDERS: Dict[str, ProviderMeta] = {
    pid: ProviderMeta(pid, name, url, auth, models_ep, chat_ep, extra)
    for pid, name, url, auth, models_ep, chat_ep, extra in PROVIDER_TABLE
}

# endpoints only openai supports (reference constants.go:60-63)
OPENAI_RESPONSES_ENDPOINT = "/responses"
OPENAI_IMAGES_ENDPOINT = "/images/generations"
OPENAI_IMAGES_EDITS_ENDPOINT = "/images/edits"
OPENAI_IMAGES_VARIATIONS_ENDPOINT = "/images/variations"
# anthropic native messages endpoint (passthrough, reference routes.go:842)
ANTHROPIC_MESSAGES_ENDPOINT = "/messages"
