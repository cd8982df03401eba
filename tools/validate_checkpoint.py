#!/usr/bin/env python3
"""Dry-run a checkpoint against the engine's loader WITHOUT a GPU.

    python tools/validate_checkpoint.py /path/to/hf-dir-or-model.gguf

Builds the architecture from the checkpoint's own metadata
(config.json or GGUF header), loads every tensor into a CPU model at
full fidelity, and reports the resolved config, tokenizer and chat
template — the same code paths `LOCAL_MODEL=hf/gguf` runs at engine
start, so a checkpoint that validates here serves on the MI355X box
(modulo the head_dim=128 GPU attention-tile requirement, which is
checked and reported)."""
from __future__ import annotations

import sys
from pathlib import Path


def main(argv=None) -> int:
    args = argv if argv is not None else sys.argv[1:]
    if len(args) != 1:
        print(__doc__)
        return 2
    path = args[0]
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

    from inference_gateway_amd.engine import gguf as G
    from inference_gateway_amd.engine import weights as W
    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM
    from inference_gateway_amd.utils.tokenizer import load_chat_template

    is_gguf = W._is_gguf(path)
    if is_gguf:
        cfg = G.model_config_from_gguf(path)
    else:
        cfg = W.model_config_from_hf(path)
    print(f"format      : {'gguf' if is_gguf else 'hf safetensors'}")
    print(f"architecture: {cfg.name}")
    print(f"hidden/layers/heads/kv: {cfg.hidden_size}/{cfg.num_layers}/"
          f"{cfg.num_heads}/{cfg.num_kv_heads}  head_dim={cfg.head_dim}")
    print(f"vocab       : {cfg.vocab_size}  ctx: {cfg.max_position}  "
          f"rope_base: {cfg.rope_base}  scaling: {cfg.rope_scaling}")
    print(f"qkv_bias    : {cfg.qkv_bias}  tied: {cfg.tie_embeddings}  "
          f"moe: {cfg.num_experts or 'no'}")
    if cfg.head_dim != 128:
        print("WARNING: head_dim != 128 — loads on CPU only; the MFMA "
              "attention kernels require head_dim=128 on GPU")
    import torch

    if cfg.is_moe:
        from inference_gateway_amd.engine.models.mixtral import (
            MixtralForCausalLM)

        model = MixtralForCausalLM(cfg, "cpu", torch.bfloat16)
    else:
        model = LlamaForCausalLM(cfg, "cpu", torch.bfloat16)
    n = W.load_weights(model, path)
    print(f"loaded      : {n} engine parameters OK")
    tok = None
    if is_gguf:
        tok = G.tokenizer_from_gguf(path)
        print(f"tokenizer   : "
              f"{'embedded BPE' if tok else 'none in metadata'}")
    else:
        tj = Path(path) / "tokenizer.json"
        print(f"tokenizer   : {'tokenizer.json' if tj.is_file() else 'none'}")
        tpl = load_chat_template(path)
        print(f"chat tmpl   : {'yes' if tpl else 'no'}")
    print("VALID")
    return 0


if __name__ == "__main__":
    sys.exit(main())
