"""Direct unit tests for small modules otherwise covered only
transitively: the byte tokenizer, the CPU sampling front-end, and the
`python -m inference_gateway_amd` entrypoint."""
from __future__ import annotations

import subprocess
import sys
from dataclasses import dataclass

import torch

from inference_gateway_amd.engine.config import SamplingParams
from inference_gateway_amd.engine.sampler import Sampler
from inference_gateway_amd.utils.tokenizer import ByteTokenizer


def test_byte_tokenizer_roundtrip():
    tok = ByteTokenizer(vocab_size=128256)
    for text in ["hello", "héllo wörld", "日本語テキスト", ""]:
        ids = tok.encode(text)
        assert ids[0] == tok.bos_token_id
        assert tok.decode(ids[1:]) == text


def test_byte_tokenizer_decode_is_total():
    """Random-init models emit arbitrary ids; decode must never raise."""
    tok = ByteTokenizer(vocab_size=128256)
    text = tok.decode([0, 1, 2, 3, 259, 1000, 128255])
    assert isinstance(text, str) and "<tok1000>" in text


@dataclass
class _Seq:
    sampling: SamplingParams


def test_sampler_greedy_and_determinism():
    torch.manual_seed(0)
    logits = torch.randn(4, 1000)
    greedy = [_Seq(SamplingParams(temperature=0.0)) for _ in range(4)]
    s = Sampler(device="cpu", seed=7)
    toks = s.sample(logits, greedy)
    assert toks == logits.argmax(dim=-1).tolist()

    random = [_Seq(SamplingParams(temperature=0.8, top_k=50, top_p=0.9))
              for _ in range(4)]
    a = Sampler(device="cpu", seed=7).sample(logits, random)
    b = Sampler(device="cpu", seed=7).sample(logits, random)
    assert a == b and len(a) == 4        # same seed, same tokens


def test_module_entrypoint_version():
    out = subprocess.run(
        [sys.executable, "-m", "inference_gateway_amd", "--version"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0
    assert "inference-gateway-amd" in (out.stdout + out.stderr)


def test_hf_tokenizer_with_real_tokenizer_file(tmp_path):
    """HFTokenizer branch with an actual tokenizers-built tokenizer.json
    (the offline wheel builds one programmatically): special tokens
    discovered, encode/decode round trip."""
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace

    vocab = {"<s>": 0, "</s>": 1, "[UNK]": 2,
             "hello": 3, "world": 4, "foo": 5}
    tok = Tokenizer(WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = Whitespace()
    path = str(tmp_path / "tokenizer.json")
    tok.save(path)

    from inference_gateway_amd.utils.tokenizer import get_tokenizer

    t = get_tokenizer(128256, path)
    assert t.bos_token_id == 0 and t.eos_token_id == 1
    ids = t.encode("hello world")
    assert ids[0] == 0                      # BOS prepended
    assert t.decode(ids[1:]) == "hello world"


def test_example_manifests_parse():
    """Every shipped example manifest must be machine-valid: compose +
    k8s YAML (multi-doc), Grafana dashboard + provisioning, Keycloak
    realm export, routing pools, generated openapi.yaml."""
    import json
    from pathlib import Path

    import yaml

    root = Path(__file__).resolve().parent.parent
    ex = root / "examples"
    n_yaml = n_json = 0
    for p in list(ex.rglob("*.yml")) + list(ex.rglob("*.yaml")):
        docs = list(yaml.safe_load_all(p.read_text()))
        assert docs, p
        n_yaml += 1
    for p in ex.rglob("*.json"):
        json.loads(p.read_text())
        n_json += 1
    assert n_yaml >= 12 and n_json >= 2, (n_yaml, n_json)
    # the dashboard targets only metrics this build actually exports
    dash = json.loads((ex / "docker-compose" / "monitoring" / "grafana"
                       / "dashboards" / "inference-gateway.json"
                       ).read_text())
    exprs = " ".join(t["expr"] for pnl in dash["panels"]
                     for t in pnl["targets"])
    from inference_gateway_amd.otel import metrics as m
    import inspect

    src = inspect.getsource(m)
    for name in ("igw_engine_running_seqs", "igw_engine_waiting_seqs",
                 "igw_engine_prefix_hit_tokens_total"):
        assert name in exprs and name in src, name
    # openapi.yaml parses and the realm export is importable JSON
    yaml.safe_load((root / "openapi.yaml").read_text().split("\n", 1)[1])
    realm = json.loads((ex / "docker-compose" / "auth" / "keycloak"
                        / "realm-export.json").read_text())
    assert realm["realm"] == "inference-gateway"
    assert realm["clients"][0]["clientId"] == "inference-gateway"


def test_nonpositive_body_cap_falls_back_to_default():
    """Reference config/server.go ResolveMaxRequestBodySize: zero or
    negative SERVER_MAX_REQUEST_BODY_SIZE means the 10 MiB default,
    not "reject every request body"."""
    from inference_gateway_amd import config as config_mod

    for bad in ("0", "-1"):
        cfg = config_mod.load({"SERVER_MAX_REQUEST_BODY_SIZE": bad})
        assert cfg.server.max_request_body_size == 10 * 1024 * 1024
    cfg = config_mod.load({"SERVER_MAX_REQUEST_BODY_SIZE": "1234"})
    assert cfg.server.max_request_body_size == 1234


def test_validate_checkpoint_cli(tmp_path, capsys):
    """tools/validate_checkpoint.py dry-runs both formats."""
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path("tools").resolve()))
    import validate_checkpoint as V

    from test_weights import _tiny_llama_checkpoint
    from test_gguf import (MODEL_PRESETS, _gguf_tensors, _meta_for,
                           _tiny_weights, write_gguf)

    hf = tmp_path / "hf"
    hf.mkdir()
    _tiny_llama_checkpoint(hf)
    assert V.main([str(hf)]) == 0
    out = capsys.readouterr().out
    assert "VALID" in out and "hf safetensors" in out

    m = MODEL_PRESETS["tiny-llama"]
    gg = tmp_path / "m.gguf"
    write_gguf(gg, _meta_for(m), _gguf_tensors(m, _tiny_weights(m)))
    assert V.main([str(gg)]) == 0
    out = capsys.readouterr().out
    assert "VALID" in out and "gguf" in out
