"""GGUF checkpoint loading (llama.cpp ecosystem format).

The reference gateway fronts llama.cpp servers as a REMOTE provider
(providers — llamacpp entry; SURVEY §2.1 provider constants); the local
engine closes the loop by loading the same GGUF files DIRECTLY, so a
quantized community checkpoint serves from the MI355X engine without a
conversion step.  Complements engine/weights.py (safetensors):

  LOCAL_MODEL=gguf LOCAL_WEIGHTS_PATH=/path/to/model.gguf

Format (GGUF v2/v3, little-endian):
  header   : magic "GGUF", version u32, n_tensors u64, n_kv u64
  metadata : [key:string, type:u32, value] * n_kv
  infos    : [name:string, n_dims:u32, dims:u64*n, ggml_type:u32,
              offset:u64] * n_tensors  (offset is relative to the
              alignment-padded data section)
  data     : tensor bytes, each aligned to general.alignment (def 32)

ggml stores a 2-D weight's dims as [ne0=in_features, ne1=out_features]
with ne0 contiguous — byte-identical to a row-major torch tensor of
shape [out, in], so no transpose is needed after reshape.

Quantizations supported: F32, F16, BF16, Q8_0 (32-wide blocks, fp16
scale + int8), Q4_0 (fp16 scale + 16 packed nibble pairs, offset 8).
K-quants raise loudly — better a clear error at startup than silent
garbage.  Dequantization targets the engine dtype (bf16): MI355X has
288 GB of HBM3E per GPU, so serving dequantized-to-bf16 is the right
trade against adding Q4 GEMM kernels for every block layout.
"""
from __future__ import annotations

import struct
from pathlib import Path
from typing import Any, Dict, List, Tuple

import numpy as np
import torch

from ..parallel.comm import tp_rank, tp_world_size
from .config import ModelConfig

GGUF_MAGIC = b"GGUF"

# metadata value types
_T_U8, _T_I8, _T_U16, _T_I16, _T_U32, _T_I32 = 0, 1, 2, 3, 4, 5
_T_F32, _T_BOOL, _T_STR, _T_ARR, _T_U64, _T_I64, _T_F64 = (
    6, 7, 8, 9, 10, 11, 12)

_SCALAR_FMT = {
    _T_U8: "<B", _T_I8: "<b", _T_U16: "<H", _T_I16: "<h",
    _T_U32: "<I", _T_I32: "<i", _T_F32: "<f", _T_U64: "<Q",
    _T_I64: "<q", _T_F64: "<d",
}

# ggml tensor types (ggml.h)
GGML_F32, GGML_F16 = 0, 1
GGML_Q4_0, GGML_Q8_0 = 2, 8
GGML_BF16 = 30

_TYPE_NAMES = {
    0: "F32", 1: "F16", 2: "Q4_0", 3: "Q4_1", 6: "Q5_0", 7: "Q5_1",
    8: "Q8_0", 9: "Q8_1", 10: "Q2_K", 11: "Q3_K", 12: "Q4_K",
    13: "Q5_K", 14: "Q6_K", 15: "Q8_K", 30: "BF16",
}


class _Reader:
    def __init__(self, buf: memoryview):
        self.buf = buf
        self.pos = 0

    def take(self, n: int) -> memoryview:
        v = self.buf[self.pos: self.pos + n]
        if len(v) != n:
            raise ValueError("truncated GGUF file")
        self.pos += n
        return v

    def scalar(self, fmt: str):
        size = struct.calcsize(fmt)
        return struct.unpack(fmt, self.take(size))[0]

    def string(self) -> str:
        n = self.scalar("<Q")
        return bytes(self.take(n)).decode("utf-8")

    def value(self, vtype: int):
        if vtype in _SCALAR_FMT:
            return self.scalar(_SCALAR_FMT[vtype])
        if vtype == _T_BOOL:
            return bool(self.scalar("<B"))
        if vtype == _T_STR:
            return self.string()
        if vtype == _T_ARR:
            et = self.scalar("<I")
            n = self.scalar("<Q")
            return [self.value(et) for _ in range(n)]
        raise ValueError(f"unknown GGUF metadata type {vtype}")


def _dequant(raw: np.ndarray, ggml_type: int, n_elem: int) -> torch.Tensor:
    """Raw bytes -> fp32 torch tensor of n_elem values."""
    if ggml_type == GGML_F32:
        return torch.from_numpy(raw.view(np.float32)[:n_elem].copy())
    if ggml_type == GGML_F16:
        return torch.from_numpy(
            raw.view(np.float16)[:n_elem].copy()).float()
    if ggml_type == GGML_BF16:
        t = torch.from_numpy(raw[: n_elem * 2].copy())
        return t.view(torch.bfloat16).float()
    if ggml_type == GGML_Q8_0:
        # block: fp16 d + 32 * int8  (34 bytes / 32 values)
        nb = n_elem // 32
        blocks = raw[: nb * 34].reshape(nb, 34)
        d = torch.from_numpy(
            blocks[:, :2].copy().view(np.float16)).float()      # [nb,1]
        q = torch.from_numpy(
            blocks[:, 2:].copy().view(np.int8)).float()         # [nb,32]
        return (d * q).reshape(-1)
    if ggml_type == GGML_Q4_0:
        # block: fp16 d + 16 bytes of nibble pairs (18 bytes / 32 values)
        # value i in [0,16) is low nibble of byte i; value 16+i the high
        # nibble; both offset by 8
        nb = n_elem // 32
        blocks = raw[: nb * 18].reshape(nb, 18)
        d = torch.from_numpy(
            blocks[:, :2].copy().view(np.float16)).float()      # [nb,1]
        qs = torch.from_numpy(blocks[:, 2:].copy())             # [nb,16] u8
        lo = (qs & 0x0F).float() - 8.0
        hi = (qs >> 4).float() - 8.0
        return (d * torch.cat([lo, hi], dim=1)).reshape(-1)
    raise ValueError(
        f"unsupported GGUF quantization "
        f"{_TYPE_NAMES.get(ggml_type, ggml_type)} — supported: F32, "
        "F16, BF16, Q8_0, Q4_0 (K-quants need a conversion pass)")


def _type_bytes(ggml_type: int, n_elem: int) -> int:
    if ggml_type == GGML_F32:
        return n_elem * 4
    if ggml_type in (GGML_F16, GGML_BF16):
        return n_elem * 2
    if ggml_type == GGML_Q8_0:
        return (n_elem // 32) * 34
    if ggml_type == GGML_Q4_0:
        return (n_elem // 32) * 18
    raise ValueError(
        f"unsupported GGUF quantization "
        f"{_TYPE_NAMES.get(ggml_type, ggml_type)} — supported: F32, "
        "F16, BF16, Q8_0, Q4_0 (K-quants need a conversion pass)")


class GGUFFile:
    """Parsed GGUF: metadata dict + lazily dequantized tensors."""

    def __init__(self, path: str):
        p = Path(path)
        if p.is_dir():
            cands = sorted(p.glob("*.gguf"))
            if not cands:
                raise FileNotFoundError(f"no .gguf file under {path}")
            p = cands[0]
        self.path = str(p)
        self._mm = np.memmap(self.path, dtype=np.uint8, mode="r")
        r = _Reader(memoryview(self._mm))
        if bytes(r.take(4)) != GGUF_MAGIC:
            raise ValueError(f"{p}: not a GGUF file")
        self.version = r.scalar("<I")
        if self.version not in (2, 3):
            raise ValueError(f"{p}: unsupported GGUF version "
                             f"{self.version}")
        n_tensors = r.scalar("<Q")
        n_kv = r.scalar("<Q")
        self.metadata: Dict[str, Any] = {}
        for _ in range(n_kv):
            key = r.string()
            vtype = r.scalar("<I")
            self.metadata[key] = r.value(vtype)
        # name -> (torch_shape, ggml_type, offset, n_elem)
        self.infos: Dict[str, Tuple[Tuple[int, ...], int, int, int]] = {}
        for _ in range(n_tensors):
            name = r.string()
            n_dims = r.scalar("<I")
            dims = [r.scalar("<Q") for _ in range(n_dims)]
            ggml_type = r.scalar("<I")
            offset = r.scalar("<Q")
            n_elem = 1
            for d in dims:
                n_elem *= d
            # ggml dims are fastest-varying first: torch shape reversed
            self.infos[name] = (tuple(reversed(dims)), ggml_type,
                                offset, n_elem)
        align = int(self.metadata.get("general.alignment", 32))
        self._data_start = (r.pos + align - 1) // align * align

    def keys(self):
        return self.infos.keys()

    def tensor(self, name: str) -> torch.Tensor:
        """Dequantized fp32 tensor in torch layout ([out, in] for 2-D)."""
        if name not in self.infos:
            raise KeyError(
                f"{self.path}: tensor {name!r} not in checkpoint")
        shape, ggml_type, offset, n_elem = self.infos[name]
        start = self._data_start + offset
        nbytes = _type_bytes(ggml_type, n_elem)
        if start + nbytes > self._mm.size or start < 0:
            raise ValueError(
                f"{self.path}: tensor {name!r} data [{start}, "
                f"{start + nbytes}) outside the file — corrupt GGUF")
        raw = np.asarray(self._mm[start: start + nbytes])
        return _dequant(raw, ggml_type, n_elem).reshape(shape)


def _arch(g: GGUFFile) -> str:
    a = g.metadata.get("general.architecture")
    if not a:
        raise ValueError("GGUF missing general.architecture")
    return a


def model_config_from_gguf(path: str) -> ModelConfig:
    g = GGUFFile(path)
    a = _arch(g)
    md = g.metadata

    def m(key, default=None):
        v = md.get(f"{a}.{key}", default)
        if v is None:
            raise ValueError(f"GGUF missing {a}.{key}")
        return v

    n_heads = int(m("attention.head_count"))
    hidden = int(m("embedding_length"))
    tied = "output.weight" not in g.infos
    # llama.cpp linear rope scaling metadata; llama-3.1-style GGUFs
    # instead ship a precomputed rope_freqs tensor (not mapped here —
    # use the HF safetensors form for 3.1 checkpoints)
    rs = None
    st = md.get(f"{a}.rope.scaling.type")
    if st == "linear":
        rs = {"type": "linear",
              "factor": float(md.get(f"{a}.rope.scaling.factor", 1.0))}
    return ModelConfig(
        name=str(md.get("general.name") or a),
        hidden_size=hidden,
        num_layers=int(m("block_count")),
        num_heads=n_heads,
        num_kv_heads=int(m("attention.head_count_kv", n_heads)),
        head_dim=int(md.get(f"{a}.attention.key_length",
                            hidden // n_heads)),
        intermediate_size=int(m("feed_forward_length")),
        vocab_size=int(g.infos["token_embd.weight"][0][0]),
        rope_base=float(md.get(f"{a}.rope.freq_base", 10000.0)),
        max_position=int(m("context_length", 8192)),
        norm_eps=float(md.get(
            f"{a}.attention.layer_norm_rms_epsilon", 1e-5)),
        tie_embeddings=tied,
        rope_scaling=rs,
        qkv_bias=("blk.0.attn_q.bias" in g.infos),
        num_experts=int(md.get(f"{a}.expert_count", 0) or 0),
        num_experts_per_tok=int(md.get(f"{a}.expert_used_count", 2) or 2),
    )


def tokenizer_from_gguf(path: str):
    """Rebuild the embedded tokenizer from GGUF metadata.

    Supports tokenizer.ggml.model == "gpt2" (byte-level BPE — what
    Llama-3 and Qwen GGUFs carry): tokens + merges arrays become a
    `tokenizers` BPE with ByteLevel pre-tokenizer/decoder.  Returns an
    HFTokenizer or None when the metadata has no reconstructable
    tokenizer (caller falls back to the synthetic ByteTokenizer)."""
    g = path if isinstance(path, GGUFFile) else GGUFFile(path)
    md = g.metadata
    tokens: List[str] = md.get("tokenizer.ggml.tokens") or []
    if not tokens or md.get("tokenizer.ggml.model") != "gpt2":
        return None
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers

    from ..utils.tokenizer import HFTokenizer

    vocab = {t: i for i, t in enumerate(tokens)}
    merges = []
    for mstr in md.get("tokenizer.ggml.merges") or []:
        a, _, b = mstr.partition(" ")
        if b:
            merges.append((a, b))
    tok = Tokenizer(models.BPE(vocab, merges, fuse_unk=False))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    bos = md.get("tokenizer.ggml.bos_token_id")
    eos = md.get("tokenizer.ggml.eos_token_id")
    return HFTokenizer(tok=tok,
                       bos_token_id=int(bos) if bos is not None else None,
                       eos_token_id=int(eos) if eos is not None else None)


def load_gguf_weights(model, path: str) -> int:
    """Load a GGUF llama/qwen2-family checkpoint into LlamaForCausalLM.

    TP sharding dequantizes the full tensor then slices this rank's
    rows/cols (GGUF block quantization has no safetensors-style lazy
    slice; peak extra memory is one fp32 tensor — fine against 288 GB).
    MoE GGUFs are rejected: Mixtral GGUF expert layouts (3-D
    ffn_*_exps) need their own mapping, and safetensors is the
    supported Mixtral path (engine/weights.py)."""
    from .weights import _copy  # shared loud shape check

    g = GGUFFile(path)
    cfg = model.cfg
    if cfg.is_moe:
        raise ValueError("GGUF loading supports dense llama/qwen2 "
                         "models; use safetensors for MoE")
    tp, rank = tp_world_size(), tp_rank()
    hd = cfg.head_dim
    nq, nkv = cfg.num_heads // tp, cfg.num_kv_heads // tp
    inter = cfg.intermediate_size // tp
    dtype = model.embed.dtype
    loaded = 0

    def t(name):
        return g.tensor(name).to(dtype)

    _copy(model.embed, t("token_embd.weight"), "embed")
    loaded += 1
    for i, layer in enumerate(model.layers):
        p = f"blk.{i}."
        q = t(p + "attn_q.weight")[rank * nq * hd: (rank + 1) * nq * hd]
        k = t(p + "attn_k.weight")[rank * nkv * hd:
                                   (rank + 1) * nkv * hd]
        v = t(p + "attn_v.weight")[rank * nkv * hd:
                                   (rank + 1) * nkv * hd]
        _copy(layer.attn.qkv, torch.cat([q, k, v], 0), f"blk{i}.qkv")
        if layer.attn.qkv_b is not None:
            qb = t(p + "attn_q.bias")[rank * nq * hd:
                                      (rank + 1) * nq * hd]
            kb = t(p + "attn_k.bias")[rank * nkv * hd:
                                      (rank + 1) * nkv * hd]
            vb = t(p + "attn_v.bias")[rank * nkv * hd:
                                      (rank + 1) * nkv * hd]
            _copy(layer.attn.qkv_b, torch.cat([qb, kb, vb], 0),
                  f"blk{i}.qkv_bias")
        elif (p + "attn_q.bias") in g.infos:
            raise ValueError(
                f"blk{i}: GGUF carries attn bias but the model config "
                "has qkv_bias=False — architecture mismatch")
        o = t(p + "attn_output.weight")[:, rank * nq * hd:
                                        (rank + 1) * nq * hd]
        _copy(layer.attn.o_proj.weight, o, f"blk{i}.attn_output")
        gt = t(p + "ffn_gate.weight")[rank * inter: (rank + 1) * inter]
        up = t(p + "ffn_up.weight")[rank * inter: (rank + 1) * inter]
        _copy(layer.mlp.gate_up, torch.cat([gt, up], 0),
              f"blk{i}.gate_up")
        dn = t(p + "ffn_down.weight")[:, rank * inter:
                                      (rank + 1) * inter]
        _copy(layer.mlp.down_proj.weight, dn, f"blk{i}.ffn_down")
        _copy(layer.input_norm_w, t(p + "attn_norm.weight"),
              f"blk{i}.attn_norm")
        _copy(layer.post_norm_w, t(p + "ffn_norm.weight"),
              f"blk{i}.ffn_norm")
        loaded += 6
    _copy(model.final_norm_w, t("output_norm.weight"), "output_norm")
    loaded += 1
    if model.lm_head is not None:
        name = ("output.weight" if "output.weight" in g.infos
                else "token_embd.weight")
        _copy(model.lm_head.weight, t(name), "output")
        loaded += 1
    return loaded
