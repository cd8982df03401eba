"""Provider registry data: the 15 remote providers of the reference
(providers/constants/constants.go:9-82, providers/registry/registry_data.go)
plus the net-new in-process `local` MI355X engine provider.

Auth types (constants/static.go): bearer | xheader | query | none.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional


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

PROVIDERS: Dict[str, ProviderMeta] = {
    "anthropic": ProviderMeta(
        "anthropic", "Anthropic", "https://api.anthropic.com/v1",
        AUTH_XHEADER, "/models", "/chat/completions",
        (("anthropic-version", "2023-06-01"),),
    ),
    "cloudflare": ProviderMeta(
        "cloudflare", "Cloudflare",
        "https://api.cloudflare.com/client/v4/accounts/{ACCOUNT_ID}/ai",
        AUTH_BEARER, "/finetunes/public?limit=1000", "/v1/chat/completions",
    ),
    "cohere": ProviderMeta(
        "cohere", "Cohere", "https://api.cohere.ai", AUTH_BEARER,
        "/v1/models", "/compatibility/v1/chat/completions",
    ),
    "deepseek": ProviderMeta(
        "deepseek", "Deepseek", "https://api.deepseek.com", AUTH_BEARER,
    ),
    "google": ProviderMeta(
        "google", "Google",
        "https://generativelanguage.googleapis.com/v1beta/openai", AUTH_BEARER,
    ),
    "groq": ProviderMeta(
        "groq", "Groq", "https://api.groq.com/openai/v1", AUTH_BEARER,
    ),
    "llamacpp": ProviderMeta(
        "llamacpp", "Llamacpp", "http://llamacpp:8080/v1", AUTH_BEARER,
    ),
    "minimax": ProviderMeta(
        "minimax", "Minimax", "https://api.minimax.io/v1", AUTH_BEARER,
    ),
    "mistral": ProviderMeta(
        "mistral", "Mistral", "https://api.mistral.ai/v1", AUTH_BEARER,
    ),
    "moonshot": ProviderMeta(
        "moonshot", "Moonshot", "https://api.moonshot.ai/v1", AUTH_BEARER,
    ),
    "nvidia": ProviderMeta(
        "nvidia", "Nvidia", "https://integrate.api.nvidia.com/v1", AUTH_BEARER,
    ),
    "ollama": ProviderMeta(
        "ollama", "Ollama", "http://ollama:8080/v1", AUTH_NONE,
    ),
    "ollama_cloud": ProviderMeta(
        "ollama_cloud", "OllamaCloud", "https://ollama.com/v1", AUTH_BEARER,
    ),
    "openai": ProviderMeta(
        "openai", "Openai", "https://api.openai.com/v1", AUTH_BEARER,
    ),
    "zai": ProviderMeta(
        "zai", "Zai", "https://api.z.ai/api/paas/v4", AUTH_BEARER,
    ),
}

# endpoints only openai supports (reference constants.go:60-63)
OPENAI_RESPONSES_ENDPOINT = "/responses"
OPENAI_IMAGES_ENDPOINT = "/images/generations"
OPENAI_IMAGES_EDITS_ENDPOINT = "/images/edits"
OPENAI_IMAGES_VARIATIONS_ENDPOINT = "/images/variations"
# anthropic native messages endpoint (passthrough, reference routes.go:842)
ANTHROPIC_MESSAGES_ENDPOINT = "/messages"
