"""Real checkpoint loading: safetensors + HF config ingestion.

The reference gateway is stateless and has no checkpoints; SURVEY.md §5
names model-weight loading at engine start as the rebuild's
checkpoint/resume analog.  This loader maps a HuggingFace-format
directory (config.json + *.safetensors) onto the engine's fused,
TP-sharded parameter layout:

  model.embed_tokens.weight                  -> embed
  layers.N.self_attn.{q,k,v}_proj.weight     -> layer.attn.qkv   (fused,
                                                column-sharded per rank)
  layers.N.self_attn.qkv_proj.weight         -> same (HF Phi-3 ships the
                                                fusion pre-done; sections
                                                are sliced per rank)
  layers.N.mlp.gate_up_proj.weight           -> layer.mlp.gate_up (Phi-3)
  layers.N.self_attn.o_proj.weight           -> layer.attn.o_proj (row)
  layers.N.mlp.{gate,up}_proj.weight         -> layer.mlp.gate_up (fused)
  layers.N.mlp.down_proj.weight              -> layer.mlp.down_proj (row)
  layers.N.{input,post_attention}_layernorm  -> norms
  model.norm.weight / lm_head.weight         -> final_norm / lm_head
  Mixtral: block_sparse_moe.gate             -> router
           experts.E.{w1,w3}                 -> gate_up[E] (fused)
           experts.E.w2                      -> down[E]

Reads are per-rank sliced BEFORE materialization (safetensors
get_slice => only this rank's shard is read from disk/mmap), so a 70B
TP=8 rank reads ~17.5 GB, not 140 GB.  Synthetic seed-derived init
remains the no-network default (parallel/layers.py)."""
from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, List

import torch

from ..parallel.comm import tp_rank, tp_world_size
from .config import ModelConfig


def model_config_from_hf(path: str, name: str = "hf-model"
                         ) -> ModelConfig:
    """Build a ModelConfig from a HuggingFace config.json."""
    cfg = json.loads((Path(path) / "config.json").read_text())
    num_experts = int(cfg.get("num_local_experts") or 0)
    mtype = cfg.get("model_type", name)
    # HF Qwen2 hardcodes qkv bias True and carries no attention_bias
    # field; Llama-family configs carry attention_bias (default False)
    qkv_bias = bool(cfg.get("attention_bias", mtype in ("qwen2",
                                                        "qwen2_moe")))
    rs = cfg.get("rope_scaling")
    if isinstance(rs, dict) and (rs.get("rope_type") or rs.get("type")) \
            in ("default", None):
        rs = None
    return ModelConfig(
        name=mtype,
        qkv_bias=qkv_bias,
        rope_scaling=rs if isinstance(rs, dict) else None,
        hidden_size=int(cfg["hidden_size"]),
        num_layers=int(cfg["num_hidden_layers"]),
        num_heads=int(cfg["num_attention_heads"]),
        num_kv_heads=int(cfg.get("num_key_value_heads",
                                 cfg["num_attention_heads"])),
        head_dim=int(cfg.get("head_dim") or (
            int(cfg["hidden_size"]) // int(cfg["num_attention_heads"]))),
        intermediate_size=int(cfg["intermediate_size"]),
        vocab_size=int(cfg["vocab_size"]),
        rope_base=float(cfg.get("rope_theta", 10000.0)),
        max_position=int(cfg.get("max_position_embeddings", 8192)),
        norm_eps=float(cfg.get("rms_norm_eps", 1e-5)),
        tie_embeddings=bool(cfg.get("tie_word_embeddings", False)),
        num_experts=num_experts,
        num_experts_per_tok=int(cfg.get("num_experts_per_tok", 2)),
    )


class _Shards:
    """Tensor-name -> (file, slice api) index over a checkpoint dir."""

    def __init__(self, path: str):
        from safetensors import safe_open

        self.files: List = []
        self.index: Dict[str, int] = {}
        for f in sorted(Path(path).glob("*.safetensors")):
            h = safe_open(str(f), framework="pt")
            i = len(self.files)
            self.files.append(h)
            for name in h.keys():
                self.index[name] = i

    def keys(self):
        return self.index.keys()

    def full(self, name: str) -> torch.Tensor:
        return self.files[self.index[name]].get_tensor(name)

    def rows(self, name: str, lo: int, hi: int) -> torch.Tensor:
        """Row slice [lo:hi] without materializing the full tensor."""
        return self.files[self.index[name]].get_slice(name)[lo:hi]

    def cols(self, name: str, lo: int, hi: int) -> torch.Tensor:
        return self.files[self.index[name]].get_slice(name)[:, lo:hi]


def _copy(param: torch.nn.Parameter, src: torch.Tensor, what: str) -> None:
    if param.shape != src.shape:
        raise ValueError(
            f"{what}: checkpoint shape {tuple(src.shape)} != engine "
            f"shape {tuple(param.shape)}")
    param.data.copy_(src.to(param.dtype))


def load_llama_weights(model, path: str) -> int:
    """Load HF Llama-family weights into LlamaForCausalLM (TP-sharded).
    Returns the number of engine parameters written."""
    sh = _Shards(path)
    cfg = model.cfg
    tp, rank = tp_world_size(), tp_rank()
    hd = cfg.head_dim
    nq, nkv = cfg.num_heads // tp, cfg.num_kv_heads // tp
    inter = cfg.intermediate_size // tp
    loaded = 0

    _copy(model.embed, sh.full("model.embed_tokens.weight"), "embed")
    loaded += 1
    # HF Phi-3 ships PRE-FUSED tensors (qkv_proj / gate_up_proj); the
    # sections inside still need per-rank slicing
    fused_ckpt = "model.layers.0.self_attn.qkv_proj.weight" in sh.index
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        if fused_ckpt:
            qkv_name = p + "self_attn.qkv_proj.weight"
            Q = cfg.num_heads * hd          # full q section rows
            K = cfg.num_kv_heads * hd
            q = sh.rows(qkv_name, rank * nq * hd, (rank + 1) * nq * hd)
            kk = sh.rows(qkv_name, Q + rank * nkv * hd,
                         Q + (rank + 1) * nkv * hd)
            v = sh.rows(qkv_name, Q + K + rank * nkv * hd,
                        Q + K + (rank + 1) * nkv * hd)
        else:
            q = sh.rows(p + "self_attn.q_proj.weight",
                        rank * nq * hd, (rank + 1) * nq * hd)
            kk = sh.rows(p + "self_attn.k_proj.weight",
                         rank * nkv * hd, (rank + 1) * nkv * hd)
            v = sh.rows(p + "self_attn.v_proj.weight",
                        rank * nkv * hd, (rank + 1) * nkv * hd)
        _copy(layer.attn.qkv, torch.cat([q, kk, v], 0), f"l{i}.qkv")
        has_bias = (p + "self_attn.q_proj.bias") in sh.index
        if layer.attn.qkv_b is not None:
            if not has_bias:
                raise ValueError(
                    f"l{i}: model expects qkv bias (qkv_bias=True) but "
                    "the checkpoint has no self_attn.*.bias tensors")
            qb = sh.rows(p + "self_attn.q_proj.bias",
                         rank * nq * hd, (rank + 1) * nq * hd)
            kb = sh.rows(p + "self_attn.k_proj.bias",
                         rank * nkv * hd, (rank + 1) * nkv * hd)
            vb = sh.rows(p + "self_attn.v_proj.bias",
                         rank * nkv * hd, (rank + 1) * nkv * hd)
            _copy(layer.attn.qkv_b, torch.cat([qb, kb, vb], 0),
                  f"l{i}.qkv_bias")
        elif has_bias:
            raise ValueError(
                f"l{i}: checkpoint carries self_attn.*.bias but the "
                "model config has qkv_bias=False — architecture mismatch")
        o = sh.cols(p + "self_attn.o_proj.weight",
                    rank * nq * hd, (rank + 1) * nq * hd)
        _copy(layer.attn.o_proj.weight, o, f"l{i}.o")
        if fused_ckpt:
            gu_name = p + "mlp.gate_up_proj.weight"
            I = cfg.intermediate_size       # full gate section rows
            g = sh.rows(gu_name, rank * inter, (rank + 1) * inter)
            u = sh.rows(gu_name, I + rank * inter, I + (rank + 1) * inter)
        else:
            g = sh.rows(p + "mlp.gate_proj.weight",
                        rank * inter, (rank + 1) * inter)
            u = sh.rows(p + "mlp.up_proj.weight",
                        rank * inter, (rank + 1) * inter)
        _copy(layer.mlp.gate_up, torch.cat([g, u], 0), f"l{i}.gate_up")
        d = sh.cols(p + "mlp.down_proj.weight",
                    rank * inter, (rank + 1) * inter)
        _copy(layer.mlp.down_proj.weight, d, f"l{i}.down")
        _copy(layer.input_norm_w, sh.full(p + "input_layernorm.weight"),
              f"l{i}.in_norm")
        _copy(layer.post_norm_w,
              sh.full(p + "post_attention_layernorm.weight"),
              f"l{i}.post_norm")
        loaded += 6
    _copy(model.final_norm_w, sh.full("model.norm.weight"), "final_norm")
    loaded += 1
    if model.lm_head is not None:
        name = ("lm_head.weight" if "lm_head.weight" in sh.index
                else "model.embed_tokens.weight")
        _copy(model.lm_head.weight, sh.full(name), "lm_head")
        loaded += 1
    return loaded


def load_mixtral_weights(model, path: str) -> int:
    """Load HF Mixtral weights (experts TP-sharded like the dense MLP)."""
    sh = _Shards(path)
    cfg = model.cfg
    tp, rank = tp_world_size(), tp_rank()
    hd = cfg.head_dim
    nq, nkv = cfg.num_heads // tp, cfg.num_kv_heads // tp
    shard = cfg.intermediate_size // tp
    loaded = 0

    _copy(model.embed, sh.full("model.embed_tokens.weight"), "embed")
    loaded += 1
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        q = sh.rows(p + "self_attn.q_proj.weight",
                    rank * nq * hd, (rank + 1) * nq * hd)
        kk = sh.rows(p + "self_attn.k_proj.weight",
                     rank * nkv * hd, (rank + 1) * nkv * hd)
        v = sh.rows(p + "self_attn.v_proj.weight",
                    rank * nkv * hd, (rank + 1) * nkv * hd)
        _copy(layer.attn.qkv, torch.cat([q, kk, v], 0), f"l{i}.qkv")
        o = sh.cols(p + "self_attn.o_proj.weight",
                    rank * nq * hd, (rank + 1) * nq * hd)
        _copy(layer.attn.o_proj.weight, o, f"l{i}.o")
        moe = layer.moe
        _copy(moe.router.weight,
              sh.full(p + "block_sparse_moe.gate.weight"), f"l{i}.gate")
        gus, downs = [], []
        for e in range(cfg.num_experts):
            ep = p + f"block_sparse_moe.experts.{e}."
            w1 = sh.rows(ep + "w1.weight", rank * shard,
                         (rank + 1) * shard)          # gate
            w3 = sh.rows(ep + "w3.weight", rank * shard,
                         (rank + 1) * shard)          # up
            w2 = sh.cols(ep + "w2.weight", rank * shard,
                         (rank + 1) * shard)          # down
            gus.append(torch.cat([w1, w3], 0))
            downs.append(w2)
        _copy(moe.gate_up, torch.stack(gus), f"l{i}.experts.gate_up")
        _copy(moe.down, torch.stack(downs), f"l{i}.experts.down")
        # keep the K-major bmm copies in sync (mixtral.py: contiguous-B
        # is both the hipBLASLt fault fix and faster)
        moe.gate_up_t.data.copy_(
            moe.gate_up.data.transpose(1, 2).contiguous())
        moe.down_t.data.copy_(moe.down.data.transpose(1, 2).contiguous())
        _copy(layer.input_norm_w, sh.full(p + "input_layernorm.weight"),
              f"l{i}.in_norm")
        _copy(layer.post_norm_w,
              sh.full(p + "post_attention_layernorm.weight"),
              f"l{i}.post_norm")
        loaded += 7
    _copy(model.final_norm_w, sh.full("model.norm.weight"), "final_norm")
    loaded += 1
    if model.lm_head is not None:
        _copy(model.lm_head.weight, sh.full("lm_head.weight"), "lm_head")
        loaded += 1
    return loaded


def _is_gguf(path: str) -> bool:
    p = Path(path)
    if p.is_file():
        return p.suffix == ".gguf"
    # a directory with .gguf files and no safetensors is a GGUF
    # checkpoint (llama.cpp download layouts)
    return (not any(p.glob("*.safetensors"))
            and any(p.glob("*.gguf")))


def load_weights(model, path: str) -> int:
    """Dispatch on format (safetensors dir vs .gguf) and architecture;
    raises on shape/name mismatches (a checkpoint that does not match
    the configured architecture must fail loudly at startup, not
    generate garbage)."""
    if _is_gguf(path):
        from .gguf import load_gguf_weights

        return load_gguf_weights(model, path)
    if getattr(model.cfg, "is_moe", False):
        return load_mixtral_weights(model, path)
    return load_llama_weights(model, path)
