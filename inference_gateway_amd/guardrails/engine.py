"""Guardrails engine.

Functional parity with the reference's internal/guardrails/guardrails.go:
four lifecycle phases (pre_call / post_call / tool_args / tool_output,
:40-44), decisions allow/block/redact/warn (:29-34), fail open/closed
(:50-53), regex secret/PII detectors with Luhn validation for card numbers
(:198-269), [REDACTED] rewriting (:272-279), an optional external HTTP
guardrail service (:286-339), and the tool-call evaluation hook used by the
MCP agent (:354-401).

The reference compiles Rego policies; this rebuild's policy files are
YAML/JSON rule lists (a native policy engine — no OPA runtime exists for
this stack), with the same decision model:

    rules:
      - name: no-cards
        phase: pre_call            # pre_call|post_call|tool_args|tool_output
        action: redact             # block|redact|warn|allow
        detectors: [credit_card]   # builtin detector names
        patterns: ["(?i)secret project \\w+"]
        contains: ["forbidden"]
        message: "optional custom message"
"""
from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional

import yaml

PHASES = ("pre_call", "post_call", "tool_args", "tool_output")
ACTIONS = ("allow", "warn", "redact", "block")
_SEVERITY = {a: i for i, a in enumerate(ACTIONS)}


def _luhn_ok(digits: str) -> bool:
    total, alt = 0, False
    for ch in reversed(digits):
        d = ord(ch) - 48
        if alt:
            d *= 2
            if d > 9:
                d -= 9
        total += d
        alt = not alt
    return total % 10 == 0


@dataclass
class Detector:
    name: str
    pattern: re.Pattern
    validate: Optional[callable] = None

    def findall(self, text: str) -> List[re.Match]:
        out = []
        for m in self.pattern.finditer(text):
            if self.validate is None or self.validate(m.group(0)):
                out.append(m)
        return out


def default_detectors() -> Dict[str, Detector]:
    """Reference DefaultDetectors (guardrails.go:198-269)."""
    return {
        "email": Detector(
            "email",
            re.compile(r"[A-Za-z0-9._%+-]+@[A-Za-z0-9.-]+\.[A-Za-z]{2,}")),
        "phone": Detector(
            "phone",
            re.compile(r"(?<!\d)(?:\+?1[-. ]?)?\(?\d{3}\)?[-. ]?\d{3}"
                       r"[-. ]?\d{4}(?!\d)")),
        "ssn": Detector(
            "ssn", re.compile(r"(?<!\d)\d{3}-\d{2}-\d{4}(?!\d)")),
        "credit_card": Detector(
            "credit_card",
            re.compile(r"(?<!\d)(?:\d[ -]?){13,19}(?!\d)"),
            validate=lambda s: _luhn_ok(re.sub(r"[^\d]", "", s))
            and 13 <= len(re.sub(r"[^\d]", "", s)) <= 19),
        "api_key": Detector(
            "api_key",
            re.compile(r"(?:sk-[A-Za-z0-9_-]{16,}|AKIA[0-9A-Z]{16}|"
                       r"ghp_[A-Za-z0-9]{36}|xox[baprs]-[A-Za-z0-9-]{10,})")),
        "ip_address": Detector(
            "ip_address",
            re.compile(r"(?<!\d)(?:\d{1,3}\.){3}\d{1,3}(?!\d)")),
    }


def redact_sensitive(text: str, detectors: List[Detector]) -> str:
    """Replace every detector hit with [REDACTED] (guardrails.go:272-279)."""
    spans = []
    for d in detectors:
        for m in d.findall(text):
            spans.append((m.start(), m.end()))
    if not spans:
        return text
    spans.sort()
    merged = [list(spans[0])]
    for a, b in spans[1:]:
        if a <= merged[-1][1]:
            merged[-1][1] = max(merged[-1][1], b)
        else:
            merged.append([a, b])
    out, last = [], 0
    for a, b in merged:
        out.append(text[last:a])
        out.append("[REDACTED]")
        last = b
    out.append(text[last:])
    return "".join(out)


@dataclass
class Rule:
    name: str
    phase: str
    action: str
    detectors: List[str] = field(default_factory=list)
    patterns: List[re.Pattern] = field(default_factory=list)
    contains: List[str] = field(default_factory=list)
    message: str = ""

    def matches(self, text: str, registry: Dict[str, Detector]) -> bool:
        for dname in self.detectors:
            d = registry.get(dname)
            if d and d.findall(text):
                return True
        for p in self.patterns:
            if p.search(text):
                return True
        low = text.lower()
        for c in self.contains:
            if c.lower() in low:
                return True
        return False


@dataclass
class Decision:
    action: str = "allow"
    rule: str = ""
    message: str = ""
    content: Optional[str] = None     # rewritten content for redact


class ExternalClient:
    """Optional external HTTP guardrail service (guardrails.go:286-339):
    POST {phase, content} -> {"action": ..., "message": ...}."""

    def __init__(self, url: str, timeout: float, client):
        self.url = url
        self.timeout = timeout
        self.client = client

    async def check(self, phase: str, content: str) -> Decision:
        r = await self.client.post(
            self.url, json={"phase": phase, "content": content},
            timeout=self.timeout)
        r.raise_for_status()
        data = r.json()
        action = data.get("action", "allow")
        if action not in ACTIONS:
            action = "allow"
        return Decision(action=action, rule="external",
                        message=data.get("message", ""))


class Evaluator:
    def __init__(self, rules: List[Rule], fail_mode: str = "closed",
                 external: Optional[ExternalClient] = None,
                 logger=None, telemetry=None):
        self.rules = rules
        self.fail_mode = fail_mode
        self.external = external
        self.logger = logger
        self.telemetry = telemetry
        self.detectors = default_detectors()

    @staticmethod
    def from_config(cfg, logger, client=None, telemetry=None) -> "Evaluator":
        rules = []
        if cfg.policy_dir:
            rules = load_policy_dir(cfg.policy_dir)
        external = None
        if cfg.external_url and client is not None:
            external = ExternalClient(cfg.external_url, cfg.external_timeout,
                                      client)
        return Evaluator(rules, cfg.fail_mode, external, logger, telemetry)

    async def eval(self, phase: str, content: str) -> Decision:
        best = Decision()
        try:
            for rule in self.rules:
                if rule.phase != phase:
                    continue
                if rule.matches(content, self.detectors):
                    if _SEVERITY[rule.action] > _SEVERITY[best.action]:
                        best = Decision(rule.action, rule.name,
                                        rule.message or
                                        f"guardrail {rule.name}: {rule.action}")
            if self.external is not None:
                ext = await self.external.check(phase, content)
                if _SEVERITY[ext.action] > _SEVERITY[best.action]:
                    best = ext
        except Exception as e:
            if self.logger:
                self.logger.error("guardrail evaluation failed",
                                  "error", str(e), "phase", phase)
            if self.fail_mode == "closed":
                return Decision("block", "fail_closed",
                                "guardrail evaluation failed (fail closed)")
            return Decision()
        if best.action == "redact":
            active = [self.detectors[d] for r in self.rules
                      if r.phase == phase and r.action == "redact"
                      for d in r.detectors if d in self.detectors]
            new = redact_sensitive(content, active or
                                   list(self.detectors.values()))
            for r in self.rules:
                if r.phase == phase and r.action == "redact":
                    for p in r.patterns:
                        new = p.sub("[REDACTED]", new)
            best.content = new
        if self.telemetry is not None and best.action != "allow":
            self.telemetry.record_guardrail(phase, best.action, best.rule)
        return best

    async def evaluate_tool_call(self, tool_name: str, args: dict
                                 ) -> Decision:
        """Pre-execution hook used by the MCP agent (guardrails.go:354-401)."""
        payload = f"{tool_name} {json.dumps(args, sort_keys=True)}"
        return await self.eval("tool_args", payload)

    async def evaluate_tool_output(self, tool_name: str, output: str
                                   ) -> Decision:
        return await self.eval("tool_output", f"{tool_name} {output}")


def load_policy_dir(path: str) -> List[Rule]:
    rules: List[Rule] = []
    base = Path(path)
    for f in sorted(base.glob("*")):
        if f.suffix not in (".yaml", ".yml", ".json"):
            continue
        data = (json.loads(f.read_text()) if f.suffix == ".json"
                else yaml.safe_load(f.read_text()))
        for raw in (data or {}).get("rules", []):
            phase = raw.get("phase", "pre_call")
            action = raw.get("action", "allow")
            if phase not in PHASES:
                raise ValueError(f"{f}: invalid phase {phase!r}")
            if action not in ACTIONS:
                raise ValueError(f"{f}: invalid action {action!r}")
            rules.append(Rule(
                name=raw.get("name", f.stem),
                phase=phase,
                action=action,
                detectors=list(raw.get("detectors") or []),
                patterns=[re.compile(p) for p in raw.get("patterns") or []],
                contains=list(raw.get("contains") or []),
                message=raw.get("message", ""),
            ))
    return rules
