"""Tokenizers for the local engine provider (net-new engine scope,
SURVEY.md §7 — the reference proxies text and never tokenizes).

Two implementations behind one interface:

* HFTokenizer — wraps a `tokenizers` tokenizer.json (the `tokenizers`
  wheel is installed offline); used when LOCAL_TOKENIZER_PATH points at a
  real file.
* ByteTokenizer — deterministic synthetic fallback (this environment has no
  network to fetch real tokenizer files): UTF-8 bytes map to ids offset by
  the number of special tokens, and out-of-byte-range ids decode to a
  readable `<tok{id}>` placeholder. Random-init models emit arbitrary ids,
  so decode must be total.
"""
from __future__ import annotations

from typing import List, Optional


class ByteTokenizer:
    PAD, BOS, EOS = 0, 1, 2
    N_SPECIAL = 3

    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size
        self.eos_token_id = self.EOS
        self.bos_token_id = self.BOS

    def encode(self, text: str) -> List[int]:
        return [self.BOS] + [b + self.N_SPECIAL for b in text.encode("utf-8")]

    def decode(self, ids: List[int]) -> str:
        out = []
        buf = bytearray()
        for i in ids:
            if self.N_SPECIAL <= i < self.N_SPECIAL + 256:
                buf.append(i - self.N_SPECIAL)
            else:
                if buf:
                    out.append(buf.decode("utf-8", errors="replace"))
                    buf = bytearray()
                if i == self.EOS:
                    continue
                out.append(f"<tok{i}>")
        if buf:
            out.append(buf.decode("utf-8", errors="replace"))
        return "".join(out)


class HFTokenizer:
    def __init__(self, path: str = None, tok=None, bos_token_id=None,
                 eos_token_id=None):
        """From a tokenizer.json path, or an already-built `tokenizers`
        Tokenizer (GGUF metadata reconstruction — engine/gguf.py)."""
        if tok is None:
            from tokenizers import Tokenizer

            tok = Tokenizer.from_file(path)
        self._tok = tok
        self.vocab_size = self._tok.get_vocab_size()
        self.eos_token_id = eos_token_id
        if self.eos_token_id is None:
            for cand in ("</s>", "<|end_of_text|>", "<|endoftext|>",
                         "<eos>"):
                tid = self._tok.token_to_id(cand)
                if tid is not None:
                    self.eos_token_id = tid
                    break
        self.bos_token_id = bos_token_id
        if self.bos_token_id is None:
            for cand in ("<s>", "<|begin_of_text|>", "<bos>"):
                tid = self._tok.token_to_id(cand)
                if tid is not None:
                    self.bos_token_id = tid
                    break

    def encode(self, text: str) -> List[int]:
        ids = self._tok.encode(text).ids
        if self.bos_token_id is not None:
            return [self.bos_token_id] + ids
        return ids

    def decode(self, ids: List[int]) -> str:
        return self._tok.decode(ids)


def get_tokenizer(vocab_size: int, path: Optional[str] = None):
    if path:
        return HFTokenizer(path)
    return ByteTokenizer(vocab_size)


def load_chat_template(weights_path: str) -> Optional[str]:
    """The checkpoint's own jinja chat template, from
    tokenizer_config.json next to the weights (HF layout), or None."""
    import json
    from pathlib import Path

    p = Path(weights_path)
    cand = (p if p.is_dir() else p.parent) / "tokenizer_config.json"
    if not cand.is_file():
        return None
    try:
        tpl = json.loads(cand.read_text()).get("chat_template")
    except (ValueError, OSError):
        return None
    return tpl if isinstance(tpl, str) else None


def render_chat_template(template: str, messages: List[dict],
                         bos_token: str = "", eos_token: str = "") -> str:
    """Render an HF-style jinja chat template (the model's own prompt
    format — without it a real checkpoint sees out-of-distribution
    role tags and generates garbage)."""
    import jinja2

    def raise_exception(msg):
        raise jinja2.TemplateError(msg)

    env = jinja2.Environment(trim_blocks=True, lstrip_blocks=True,
                             keep_trailing_newline=True)
    env.globals["raise_exception"] = raise_exception
    # flatten OpenAI content-part unions to text for the template
    msgs = []
    for m in messages:
        c = m.get("content") or ""
        if isinstance(c, list):
            c = "".join(p.get("text", "") for p in c
                        if isinstance(p, dict) and p.get("type") == "text")
        msgs.append({**m, "content": c})
    return env.from_string(template).render(
        messages=msgs, add_generation_prompt=True,
        bos_token=bos_token, eos_token=eos_token)


def apply_chat_template(messages: List[dict]) -> str:
    """Minimal chat template (role-tagged turns). Real checkpoints would use
    the model's own template; with random-init weights the exact template is
    immaterial, but the structure keeps the token stream role-aware."""
    parts = []
    for m in messages:
        role = m.get("role", "user")
        content = m.get("content") or ""
        if isinstance(content, list):  # OpenAI content-parts union
            content = "".join(
                p.get("text", "") for p in content if isinstance(p, dict)
                and p.get("type") == "text")
        parts.append(f"<|{role}|>\n{content}\n")
    parts.append("<|assistant|>\n")
    return "".join(parts)
