"""Env-var configuration system.

Mirrors the reference's env contract exactly (reference Configurations.md /
config/config.go:13-108, load.go:18-51): same variable names, same defaults,
provider bootstrap from registry defaults + <ID>_API_URL / <ID>_API_KEY
overrides. Net-new LOCAL_* block configures the in-process MI355X engine
provider. Durations accept Go-style strings ("30s", "5m") for compatibility.
"""
from __future__ import annotations

import os
import re
from dataclasses import dataclass, field
from typing import Dict, Mapping, Optional

_DUR_RE = re.compile(r"(\d+(?:\.\d+)?)(ms|s|m|h)")
_DUR_MULT = {"ms": 0.001, "s": 1.0, "m": 60.0, "h": 3600.0}


def parse_duration(s: str) -> float:
    """Go-style duration string -> seconds. Bare numbers are seconds."""
    s = s.strip()
    if not s:
        return 0.0
    try:
        return float(s)
    except ValueError:
        pass
    total, pos = 0.0, 0
    for m in _DUR_RE.finditer(s):
        if m.start() != pos:
            raise ValueError(f"invalid duration: {s!r}")
        total += float(m.group(1)) * _DUR_MULT[m.group(2)]
        pos = m.end()
    if pos != len(s):
        raise ValueError(f"invalid duration: {s!r}")
    return total


def _bool(v: str) -> bool:
    return v.strip().lower() in ("1", "true", "yes", "on")


def _csv(v: str) -> list:
    return [x.strip() for x in v.split(",") if x.strip()]


@dataclass
class GeneralConfig:
    environment: str = "production"
    allowed_models: list = field(default_factory=list)
    disallowed_models: list = field(default_factory=list)
    enable_vision: bool = False
    enable_images: bool = False
    debug_content_truncate_words: int = 10
    debug_max_messages: int = 100


@dataclass
class TelemetryConfig:
    enabled: bool = False
    metrics_push_enabled: bool = False
    metrics_port: int = 9464
    tracing_enabled: bool = False
    tracing_otlp_endpoint: str = "http://localhost:4318"


@dataclass
class MCPConfig:
    enabled: bool = False
    expose: bool = False
    servers: list = field(default_factory=list)
    tool_mode: str = "selector"       # selector | direct
    include_tools: list = field(default_factory=list)
    exclude_tools: list = field(default_factory=list)
    client_timeout: float = 5.0
    dial_timeout: float = 3.0
    tls_handshake_timeout: float = 3.0
    response_header_timeout: float = 3.0
    expect_continue_timeout: float = 1.0
    request_timeout: float = 5.0
    max_retries: int = 3
    retry_interval: float = 5.0
    initial_backoff: float = 1.0
    enable_reconnect: bool = True
    reconnect_interval: float = 30.0
    polling_enabled: bool = True
    polling_interval: float = 30.0
    polling_timeout: float = 5.0
    disable_healthcheck_logs: bool = True


@dataclass
class AuthConfig:
    enabled: bool = False
    oidc_issuer: str = "http://keycloak:8080/realms/inference-gateway-realm"
    oidc_client_id: str = "inference-gateway-client"
    oidc_client_secret: str = ""


@dataclass
class GuardrailsConfig:
    enabled: bool = False
    policy_dir: str = ""
    fail_mode: str = "closed"        # closed | open
    external_url: str = ""
    external_timeout: float = 5.0


@dataclass
class ServerConfig:
    host: str = "127.0.0.1"
    port: int = 8080
    read_timeout: float = 30.0
    write_timeout: float = 30.0
    idle_timeout: float = 120.0
    max_request_body_size: int = 10 * 1024 * 1024
    tls_cert_path: str = ""
    tls_key_path: str = ""
    # SO_REUSEPORT worker processes for proxy-only deployments (the Go
    # reference scales across cores in one process; CPython needs one
    # event loop per core).  Forced to 1 when the local engine is
    # enabled — one GPU engine per process would double-allocate HBM.
    workers: int = 1


@dataclass
class ClientConfig:
    timeout: float = 30.0
    max_idle_conns: int = 20
    max_idle_conns_per_host: int = 20
    idle_conn_timeout: float = 30.0
    tls_min_version: str = "TLS12"
    disable_compression: bool = True
    response_header_timeout: float = 10.0
    expect_continue_timeout: float = 1.0


@dataclass
class RoutingConfig:
    enabled: bool = False
    config_path: str = ""


@dataclass
class LocalConfig:
    """Net-new: the in-process MI355X engine provider."""
    enabled: bool = False
    model: str = "llama-3-8b"
    served_model_name: str = ""       # defaults to model
    tensor_parallel_size: int = 1
    max_num_seqs: int = 256
    max_model_len: int = 8192
    gpu_memory_utilization: float = 0.90
    enforce_eager: bool = False
    prefix_caching: bool = False
    chunked_prefill: bool = False
    prefill_first: bool = True        # strict admission priority
    mixed_batches: bool = False       # same-step prefill+decode rounds
    max_queue: int = 0                # waiting-queue cap (0 = unbounded)
    weights_path: str = ""            # HF safetensors dir; "" = synthetic
    tokenizer_path: str = ""          # tokenizers json; "" = byte fallback


@dataclass
class ProviderEntry:
    id: str
    url: str
    token: str = ""


@dataclass
class Config:
    general: GeneralConfig = field(default_factory=GeneralConfig)
    telemetry: TelemetryConfig = field(default_factory=TelemetryConfig)
    mcp: MCPConfig = field(default_factory=MCPConfig)
    auth: AuthConfig = field(default_factory=AuthConfig)
    guardrails: GuardrailsConfig = field(default_factory=GuardrailsConfig)
    server: ServerConfig = field(default_factory=ServerConfig)
    client: ClientConfig = field(default_factory=ClientConfig)
    routing: RoutingConfig = field(default_factory=RoutingConfig)
    local: LocalConfig = field(default_factory=LocalConfig)
    providers: Dict[str, ProviderEntry] = field(default_factory=dict)


def load(env: Optional[Mapping[str, str]] = None) -> Config:
    """Build Config from an env mapping (defaults to os.environ) — the
    analog of config.Load(lookuper) in the reference."""
    e = dict(os.environ) if env is None else dict(env)

    def g(key: str, default: str = "") -> str:
        return e.get(key, default)

    cfg = Config()
    c = cfg.general
    c.environment = g("ENVIRONMENT", "production")
    c.allowed_models = _csv(g("ALLOWED_MODELS"))
    c.disallowed_models = _csv(g("DISALLOWED_MODELS"))
    c.enable_vision = _bool(g("ENABLE_VISION", "false"))
    c.enable_images = _bool(g("ENABLE_IMAGES", "false"))
    c.debug_content_truncate_words = int(g("DEBUG_CONTENT_TRUNCATE_WORDS", "10"))
    c.debug_max_messages = int(g("DEBUG_MAX_MESSAGES", "100"))

    t = cfg.telemetry
    t.enabled = _bool(g("TELEMETRY_ENABLED", "false"))
    t.metrics_push_enabled = _bool(g("TELEMETRY_METRICS_PUSH_ENABLED", "false"))
    t.metrics_port = int(g("TELEMETRY_METRICS_PORT", "9464"))
    t.tracing_enabled = _bool(g("TELEMETRY_TRACING_ENABLED", "false"))
    t.tracing_otlp_endpoint = g("TELEMETRY_TRACING_OTLP_ENDPOINT",
                                "http://localhost:4318")

    m = cfg.mcp
    m.enabled = _bool(g("MCP_ENABLED", "false"))
    m.expose = _bool(g("MCP_EXPOSE", "false"))
    m.servers = _csv(g("MCP_SERVERS"))
    m.tool_mode = g("MCP_TOOL_MODE", "selector")
    m.include_tools = _csv(g("MCP_INCLUDE_TOOLS"))
    m.exclude_tools = _csv(g("MCP_EXCLUDE_TOOLS"))
    m.client_timeout = parse_duration(g("MCP_CLIENT_TIMEOUT", "5s"))
    m.dial_timeout = parse_duration(g("MCP_DIAL_TIMEOUT", "3s"))
    m.tls_handshake_timeout = parse_duration(g("MCP_TLS_HANDSHAKE_TIMEOUT", "3s"))
    m.response_header_timeout = parse_duration(
        g("MCP_RESPONSE_HEADER_TIMEOUT", "3s"))
    m.expect_continue_timeout = parse_duration(
        g("MCP_EXPECT_CONTINUE_TIMEOUT", "1s"))
    m.request_timeout = parse_duration(g("MCP_REQUEST_TIMEOUT", "5s"))
    m.max_retries = int(g("MCP_MAX_RETRIES", "3"))
    m.retry_interval = parse_duration(g("MCP_RETRY_INTERVAL", "5s"))
    m.initial_backoff = parse_duration(g("MCP_INITIAL_BACKOFF", "1s"))
    m.enable_reconnect = _bool(g("MCP_ENABLE_RECONNECT", "true"))
    m.reconnect_interval = parse_duration(g("MCP_RECONNECT_INTERVAL", "30s"))
    m.polling_enabled = _bool(g("MCP_POLLING_ENABLED", "true"))
    m.polling_interval = parse_duration(g("MCP_POLLING_INTERVAL", "30s"))
    m.polling_timeout = parse_duration(g("MCP_POLLING_TIMEOUT", "5s"))
    m.disable_healthcheck_logs = _bool(g("MCP_DISABLE_HEALTHCHECK_LOGS", "true"))

    a = cfg.auth
    a.enabled = _bool(g("AUTH_ENABLED", "false"))
    a.oidc_issuer = g("AUTH_OIDC_ISSUER", a.oidc_issuer)
    a.oidc_client_id = g("AUTH_OIDC_CLIENT_ID", a.oidc_client_id)
    a.oidc_client_secret = g("AUTH_OIDC_CLIENT_SECRET", "")

    gr = cfg.guardrails
    gr.enabled = _bool(g("GUARDRAILS_ENABLED", "false"))
    gr.policy_dir = g("GUARDRAILS_POLICY_DIR", "")
    gr.fail_mode = g("GUARDRAILS_FAIL_MODE", "closed")
    gr.external_url = g("GUARDRAILS_EXTERNAL_URL", "")
    gr.external_timeout = parse_duration(g("GUARDRAILS_EXTERNAL_TIMEOUT", "5s"))

    s = cfg.server
    s.host = g("SERVER_HOST", "127.0.0.1")
    s.port = int(g("SERVER_PORT", "8080"))
    s.read_timeout = parse_duration(g("SERVER_READ_TIMEOUT", "30s"))
    s.write_timeout = parse_duration(g("SERVER_WRITE_TIMEOUT", "30s"))
    s.idle_timeout = parse_duration(g("SERVER_IDLE_TIMEOUT", "120s"))
    s.max_request_body_size = int(g("SERVER_MAX_REQUEST_BODY_SIZE",
                                    str(10 * 1024 * 1024)))
    if s.max_request_body_size <= 0:
        # non-positive falls back to the default (reference
        # config/server.go ResolveMaxRequestBodySize semantics)
        s.max_request_body_size = 10 * 1024 * 1024
    s.tls_cert_path = g("SERVER_TLS_CERT_PATH", "")
    s.tls_key_path = g("SERVER_TLS_KEY_PATH", "")
    s.workers = int(g("SERVER_WORKERS", "1"))

    cl = cfg.client
    cl.timeout = parse_duration(g("CLIENT_TIMEOUT", "30s"))
    cl.max_idle_conns = int(g("CLIENT_MAX_IDLE_CONNS", "20"))
    cl.max_idle_conns_per_host = int(g("CLIENT_MAX_IDLE_CONNS_PER_HOST", "20"))
    cl.idle_conn_timeout = parse_duration(g("CLIENT_IDLE_CONN_TIMEOUT", "30s"))
    cl.tls_min_version = g("CLIENT_TLS_MIN_VERSION", "TLS12")
    cl.disable_compression = _bool(g("CLIENT_DISABLE_COMPRESSION", "true"))
    cl.response_header_timeout = parse_duration(
        g("CLIENT_RESPONSE_HEADER_TIMEOUT", "10s"))
    cl.expect_continue_timeout = parse_duration(
        g("CLIENT_EXPECT_CONTINUE_TIMEOUT", "1s"))

    r = cfg.routing
    r.enabled = _bool(g("ROUTING_ENABLED", "false"))
    r.config_path = g("ROUTING_CONFIG_PATH", "")

    lo = cfg.local
    lo.enabled = _bool(g("LOCAL_ENABLED", "false"))
    lo.model = g("LOCAL_MODEL", "llama-3-8b")
    lo.served_model_name = g("LOCAL_SERVED_MODEL_NAME", "") or lo.model
    lo.tensor_parallel_size = int(g("LOCAL_TENSOR_PARALLEL_SIZE", "1"))
    lo.max_num_seqs = int(g("LOCAL_MAX_NUM_SEQS", "256"))
    lo.max_model_len = int(g("LOCAL_MAX_MODEL_LEN", "8192"))
    lo.gpu_memory_utilization = float(g("LOCAL_GPU_MEMORY_UTILIZATION", "0.90"))
    lo.enforce_eager = _bool(g("LOCAL_ENFORCE_EAGER", "false"))
    lo.prefix_caching = _bool(g("LOCAL_PREFIX_CACHING", "false"))
    lo.chunked_prefill = _bool(g("LOCAL_CHUNKED_PREFILL", "false"))
    lo.prefill_first = _bool(g("LOCAL_PREFILL_FIRST", "true"))
    lo.mixed_batches = _bool(g("LOCAL_MIXED_BATCHES", "false"))
    lo.max_queue = int(g("LOCAL_MAX_QUEUE", "0"))
    lo.weights_path = g("LOCAL_WEIGHTS_PATH", "")
    lo.tokenizer_path = g("LOCAL_TOKENIZER_PATH", "")

    # provider bootstrap: registry defaults + env overrides
    from ..providers.constants import PROVIDERS

    for pid, meta in PROVIDERS.items():
        prefix = pid.upper()
        url = g(f"{prefix}_API_URL", meta.url)
        token = g(f"{prefix}_API_KEY", "")
        cfg.providers[pid] = ProviderEntry(id=pid, url=url, token=token)
    return cfg
