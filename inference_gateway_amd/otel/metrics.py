"""Telemetry instruments (parity with the reference's otel/otel.go:72-284).

Same instrument names and label sets, exported in Prometheus exposition
format on the dedicated metrics port (reference: :9464). GenAI semconv
conventions: gen_ai.client.token.usage histogram with power-of-4 buckets
(otel.go:101), gen_ai.server.request.duration with semconv duration
buckets (otel.go:100), tool-call and guardrail counters, plus the four
push-only histograms fed by the OTLP ingestion endpoint (otel/ingest.py).
"""
from __future__ import annotations

from typing import Optional

from prometheus_client import (CollectorRegistry, Counter, Histogram,
                               generate_latest)

TOKEN_BUCKETS = [1, 4, 16, 64, 256, 1024, 4096, 16384, 65536, 262144,
                 1048576, 4194304, 16777216, 67108864]
DURATION_BUCKETS = [0.005, 0.01, 0.025, 0.05, 0.075, 0.1, 0.25, 0.5, 0.75,
                    1.0, 2.5, 5.0, 7.5, 10.0]


class Telemetry:
    def __init__(self, registry: Optional[CollectorRegistry] = None):
        self.registry = registry or CollectorRegistry()
        r = self.registry
        self.token_usage = Histogram(
            "gen_ai_client_token_usage", "GenAI token usage",
            ["provider", "model", "token_type", "source", "team"],
            buckets=TOKEN_BUCKETS, registry=r)
        self.request_duration = Histogram(
            "gen_ai_server_request_duration_seconds",
            "GenAI request duration",
            ["provider", "model", "error_type"],
            buckets=DURATION_BUCKETS, registry=r)
        self.tool_calls = Counter(
            "inference_gateway_tool_calls_total", "Tool calls",
            ["provider", "model", "tool_type", "tool_name"], registry=r)
        self.guardrail_events = Counter(
            "inference_gateway_guardrails_total", "Guardrail decisions",
            ["phase", "decision", "rule"], registry=r)
        # push-only histograms (OTLP ingestion; reference otel.go:185-215)
        self.client_operation_duration = Histogram(
            "gen_ai_client_operation_duration_seconds",
            "Client-reported operation duration",
            ["provider", "model", "source", "team"],
            buckets=DURATION_BUCKETS, registry=r)
        self.client_time_to_first_chunk = Histogram(
            "gen_ai_client_time_to_first_chunk_seconds",
            "Client-reported time to first chunk",
            ["provider", "model", "source", "team"],
            buckets=DURATION_BUCKETS, registry=r)
        self.server_time_to_first_token = Histogram(
            "gen_ai_server_time_to_first_token_seconds",
            "Server time to first token",
            ["provider", "model"],
            buckets=DURATION_BUCKETS, registry=r)
        self.tool_duration = Histogram(
            "inference_gateway_tool_duration_seconds",
            "MCP tool execution duration",
            ["tool_name", "source", "team"],
            buckets=DURATION_BUCKETS, registry=r)

    # convenience recorders ------------------------------------------------
    def record_tokens(self, provider: str, model: str, prompt: int,
                      completion: int, source: str = "", team: str = ""):
        self.token_usage.labels(provider, model, "input", source,
                                team).observe(prompt)
        self.token_usage.labels(provider, model, "output", source,
                                team).observe(completion)

    def record_request(self, provider: str, model: str, seconds: float,
                       error_type: str = ""):
        self.request_duration.labels(provider, model, error_type).observe(
            seconds)

    def record_tool_call(self, provider: str, model: str, tool_type: str,
                         tool_name: str):
        self.tool_calls.labels(provider, model, tool_type,
                               tool_name).inc()

    def record_guardrail(self, phase: str, decision: str, rule: str = ""):
        self.guardrail_events.labels(phase, decision, rule).inc()

    def record_ttft(self, provider: str, model: str, seconds: float):
        self.server_time_to_first_token.labels(provider, model).observe(
            seconds)

    def register_engine_gauges(self, engine) -> None:
        """Scrape-time gauges over the local engine's scheduler/KV state
        (the serving-engine analog of the reference's process-level
        footprint claims): running/waiting sequences and KV-block
        utilization."""
        from prometheus_client import Gauge

        r = self.registry
        g_run = Gauge("igw_engine_running_seqs",
                      "sequences in the running batch", registry=r)
        g_wait = Gauge("igw_engine_waiting_seqs",
                       "sequences queued for prefill", registry=r)
        g_free = Gauge("igw_engine_kv_blocks_free",
                       "free paged-KV blocks", registry=r)
        g_total = Gauge("igw_engine_kv_blocks_total",
                        "total paged-KV blocks", registry=r)
        g_gen = Gauge("igw_engine_generated_tokens_total",
                      "cumulative generated tokens", registry=r)
        g_prompt = Gauge("igw_engine_prompt_tokens_total",
                         "cumulative prefilled prompt tokens", registry=r)
        g_preempt = Gauge("igw_engine_preemptions_total",
                          "cumulative KV-pressure preemptions", registry=r)
        g_forks = Gauge("igw_engine_forks_total",
                        "cumulative n>1 KV-fork branches", registry=r)
        g_pfx = Gauge("igw_engine_prefix_hit_tokens_total",
                      "prompt tokens served from the prefix cache",
                      registry=r)
        g_steps = Gauge("igw_engine_steps_total",
                        "cumulative engine steps", registry=r)
        g_mixed = Gauge("igw_engine_mixed_rounds_total",
                        "cumulative mixed prefill+decode rounds",
                        registry=r)
        sched = engine.scheduler
        bm = engine.block_manager
        g_run.set_function(lambda: len(sched.running))
        g_wait.set_function(lambda: len(sched.waiting))
        g_free.set_function(lambda: bm.num_free)
        g_total.set_function(lambda: bm.num_blocks - 1)  # block 0 reserved
        g_gen.set_function(lambda: engine.num_generated_tokens)
        g_prompt.set_function(lambda: engine.num_prompt_tokens)
        g_preempt.set_function(lambda: sched.num_preemptions)
        g_forks.set_function(lambda: engine.num_forks)
        g_pfx.set_function(lambda: bm.prefix_hit_tokens)
        g_steps.set_function(lambda: engine.num_steps)
        g_mixed.set_function(lambda: sched.num_mixed_rounds)

    def exposition(self) -> bytes:
        return generate_latest(self.registry)
