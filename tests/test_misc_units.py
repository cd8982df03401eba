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
