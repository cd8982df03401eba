"""Anti-drift tests for the spec/generate pipeline (the analog of the
reference's provider_drift_test.go + the CI `task generate` cleanliness
check): regenerating from spec/gateway.yaml must produce exactly the
committed artifacts, the config loader must consume exactly the spec'd
env vars, and every spec'd provider must route/build end-to-end."""
import re
import sys
from pathlib import Path

import yaml

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT / "tools"))


def test_generated_artifacts_up_to_date(tmp_path):
    import generate as gen

    files = gen.generate(tmp_path)
    for rel, content in files.items():
        committed = (ROOT / rel).read_text()
        assert committed == content, (
            f"{rel} is stale — run `python tools/generate.py`")


def test_config_consumes_exactly_the_spec_vars():
    spec = yaml.safe_load((ROOT / "spec" / "gateway.yaml").read_text())
    spec_vars = {v for group in spec["config"].values() for v in group}
    src = (ROOT / "inference_gateway_amd" / "config" / "__init__.py"
           ).read_text()
    used = set(re.findall(r'g\(\s*"([A-Z_]+)"', src))
    assert used == spec_vars, (
        f"spec/config drift: only-in-spec={sorted(spec_vars - used)} "
        f"only-in-config={sorted(used - spec_vars)}")


def test_spec_defaults_match_config_defaults():
    spec = yaml.safe_load((ROOT / "spec" / "gateway.yaml").read_text())
    src = (ROOT / "inference_gateway_amd" / "config" / "__init__.py"
           ).read_text()
    src_defaults = dict(
        re.findall(r'g\(\s*"([A-Z_]+)"\s*,\s*"((?:[^"\\]|\\.)*)"', src))
    for group in spec["config"].values():
        for var, meta in group.items():
            if var in src_defaults:
                assert str(meta.get("default", "")) == src_defaults[var], (
                    f"default drift for {var}")


def test_every_provider_routes_and_builds():
    """Adding a provider to the spec + regenerate must be sufficient
    (reference tests/provider_drift_test.go:27-62)."""
    import httpx

    from inference_gateway_amd import config as config_mod
    from inference_gateway_amd.providers import routing
    from inference_gateway_amd.providers.constants import PROVIDERS
    from inference_gateway_amd.providers.core import ProviderRegistry

    env = {f"{pid.upper()}_API_KEY": "tok" for pid in PROVIDERS}
    cfg = config_mod.load(env)
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(200)))
    reg = ProviderRegistry(cfg, client)
    for pid, meta in PROVIDERS.items():
        got_pid, model = routing.determine_provider_and_model(f"{pid}/m-x")
        assert got_pid == pid and model == "m-x"
        p = reg.build(pid)
        assert p is not None
        # auth semantics are spec-driven
        if meta.auth_type == "bearer":
            assert p.auth_headers().get(
                "Authorization", "").startswith("Bearer ")
        elif meta.auth_type == "xheader":
            assert "x-api-key" in {k.lower() for k in p.auth_headers()}


def test_example_env_vars_are_real():
    """Every IGW-looking env var referenced by the examples must exist
    in the generated Configurations.md (catches stacks drifting from the
    config contract, e.g. a renamed LOCAL_* var)."""
    import re
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    documented = set(re.findall(r"`([A-Z][A-Z0-9_]{2,})`",
                                (root / "Configurations.md").read_text()))
    # provider bootstrap vars are <ID>_API_URL/KEY patterns
    dynamic = re.compile(r".*_API_(URL|KEY)$")
    prefixes = ("LOCAL_", "MCP_", "SERVER_", "TELEMETRY_", "GUARDRAILS_",
                "AUTH_", "OIDC_", "ALLOWED_", "DISALLOWED_", "ENVIRONMENT")
    missing = []
    for f in list(root.glob("examples/**/*.yml")) + \
            list(root.glob("examples/**/*.yaml")) + [root / ".env.example"]:
        for var in re.findall(r"\b([A-Z][A-Z0-9_]{2,})=", f.read_text()):
            if not var.startswith(prefixes):
                continue
            if var in documented or dynamic.match(var):
                continue
            missing.append(f"{f.relative_to(root)}: {var}")
    assert not missing, f"undocumented env vars in examples: {missing}"


def test_openapi_document_shape():
    """openapi.yaml is the machine-readable gateway surface (reference
    openapi.yaml analog): well-formed, carries the chat schema, the
    x-config env contract and all 15 providers."""
    import yaml

    from inference_gateway_amd.providers.constants_gen import PROVIDER_TABLE
    from inference_gateway_amd.providers.local import LocalProvider

    root = Path(__file__).resolve().parent.parent
    doc = yaml.safe_load((root / "openapi.yaml").read_text())
    assert doc["openapi"].startswith("3.")
    assert "/v1/chat/completions" in doc["paths"]
    schemas = doc["components"]["schemas"]
    req = schemas["CreateChatCompletionRequest"]
    assert req["required"] == ["model", "messages"]
    # the documented n cap matches the engine's fork-group cap
    assert req["properties"]["n"]["maximum"] == LocalProvider.MAX_N
    # every registered provider is in the x-provider-configs extension
    for row in PROVIDER_TABLE:
        assert row[0] in doc["x-provider-configs"], row[0]
    # env contract carried as x-config
    assert "LOCAL_WEIGHTS_PATH" in str(doc["x-config"])
