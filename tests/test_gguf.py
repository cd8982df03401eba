"""GGUF checkpoint loading (engine/gguf.py): format parsing, Q8_0/Q4_0
dequantization, metadata -> ModelConfig, end-to-end generation, and
token-exact equivalence with the safetensors loader on identical
weights.  The writer here is built from the GGUF spec, independent of
the reader's code paths."""
import json
import struct

import numpy as np
import pytest
import torch
from safetensors.torch import save_file

from inference_gateway_amd.engine import gguf as G
from inference_gateway_amd.engine.config import (MODEL_PRESETS,
                                                 EngineConfig,
                                                 SamplingParams)
from inference_gateway_amd.engine.engine import LLMEngine

ALIGN = 32


def _s(b: bytes) -> bytes:
    return struct.pack("<Q", len(b)) + b


def _kv(key: str, vtype: int, raw: bytes) -> bytes:
    return _s(key.encode()) + struct.pack("<I", vtype) + raw


def _kv_u32(key, v):
    return _kv(key, 4, struct.pack("<I", v))


def _kv_f32(key, v):
    return _kv(key, 6, struct.pack("<f", v))


def _kv_str(key, v):
    return _kv(key, 8, _s(v.encode()))


def _q8_0(w: torch.Tensor) -> bytes:
    """llama.cpp Q8_0: per-32 block fp16 scale + int8 quants."""
    flat = w.float().reshape(-1, 32)
    d = flat.abs().amax(dim=1, keepdim=True) / 127.0
    d = torch.where(d == 0, torch.ones_like(d), d).to(torch.float16)
    q = torch.clamp(torch.round(flat / d.float()), -127, 127).to(torch.int8)
    out = bytearray()
    dn = d.numpy()
    qn = q.numpy()
    for i in range(flat.shape[0]):
        out += dn[i].tobytes() + qn[i].tobytes()
    return bytes(out)


def _q8_0_expected(raw: bytes, n: int) -> torch.Tensor:
    nb = n // 32
    a = np.frombuffer(raw, dtype=np.uint8)[: nb * 34].reshape(nb, 34)
    d = torch.from_numpy(a[:, :2].copy().view(np.float16)).float()
    q = torch.from_numpy(a[:, 2:].copy().view(np.int8)).float()
    return (d * q).reshape(-1)


def _q4_0(w: torch.Tensor) -> bytes:
    flat = w.float().reshape(-1, 32)
    amax_idx = flat.abs().argmax(dim=1)
    mx = flat[torch.arange(flat.shape[0]), amax_idx]
    d = (mx / -8.0)
    d = torch.where(d == 0, torch.ones_like(d), d).to(torch.float16)
    q = torch.clamp(torch.round(flat / d.float().unsqueeze(1)) + 8,
                    0, 15).to(torch.uint8)
    out = bytearray()
    dn = d.numpy()
    qn = q.numpy()
    for i in range(flat.shape[0]):
        packed = (qn[i, :16] | (qn[i, 16:] << 4)).astype(np.uint8)
        out += dn[i].tobytes() + packed.tobytes()
    return bytes(out)


def write_gguf(path, metadata_blobs: list, tensors: dict) -> None:
    """tensors: name -> (torch_shape, ggml_type, raw_bytes)."""
    head = b"GGUF" + struct.pack("<IQQ", 3, len(tensors),
                                 len(metadata_blobs))
    meta = b"".join(metadata_blobs)
    infos = b""
    offset = 0
    for name, (shape, gt, raw) in tensors.items():
        dims = list(reversed(shape))
        infos += (_s(name.encode()) + struct.pack("<I", len(dims))
                  + b"".join(struct.pack("<Q", d) for d in dims)
                  + struct.pack("<IQ", gt, offset))
        offset += (len(raw) + ALIGN - 1) // ALIGN * ALIGN
    blob = head + meta + infos
    pad = (-len(blob)) % ALIGN
    blob += b"\x00" * pad
    for name, (shape, gt, raw) in tensors.items():
        blob += raw + b"\x00" * ((-len(raw)) % ALIGN)
    path.write_bytes(blob)


def _meta_for(m, arch="llama") -> list:
    return [
        _kv_str("general.architecture", arch),
        _kv_str("general.name", "tiny"),
        _kv_u32(f"{arch}.embedding_length", m.hidden_size),
        _kv_u32(f"{arch}.block_count", m.num_layers),
        _kv_u32(f"{arch}.attention.head_count", m.num_heads),
        _kv_u32(f"{arch}.attention.head_count_kv", m.num_kv_heads),
        _kv_u32(f"{arch}.attention.key_length", m.head_dim),
        _kv_u32(f"{arch}.feed_forward_length", m.intermediate_size),
        _kv_u32(f"{arch}.context_length", m.max_position),
        _kv_f32(f"{arch}.rope.freq_base", m.rope_base),
        _kv_f32(f"{arch}.attention.layer_norm_rms_epsilon", m.norm_eps),
    ]


def _f16(w):
    return w.to(torch.float16).numpy().tobytes()


def _f32(w):
    return w.float().numpy().tobytes()


def _tiny_weights(m, seed=0):
    """f16-representable weights: the GGUF file stores them as F16
    exactly and the safetensors file as F32 exactly, so both loaders
    see bit-identical values before the engine's bf16 conversion.
    (bf16-representable values are NOT always f16-exact: bf16's wide
    exponent range hits f16 subnormals.)"""
    g = torch.Generator().manual_seed(seed)

    def r(*shape):
        return (torch.randn(*shape, generator=g) * 0.02).to(
            torch.float16).float()

    h, hd = m.hidden_size, m.head_dim
    t = {"embed": r(m.vocab_size, h), "norm": torch.ones(h),
         "lm_head": r(m.vocab_size, h), "layers": []}
    for _ in range(m.num_layers):
        t["layers"].append({
            "q": r(m.num_heads * hd, h), "k": r(m.num_kv_heads * hd, h),
            "v": r(m.num_kv_heads * hd, h), "o": r(h, m.num_heads * hd),
            "gate": r(m.intermediate_size, h),
            "up": r(m.intermediate_size, h),
            "down": r(h, m.intermediate_size),
            "in_norm": torch.ones(h), "post_norm": torch.ones(h)})
    return t


def _gguf_tensors(m, t):
    ts = {"token_embd.weight": ((m.vocab_size, m.hidden_size), G.GGML_F16,
                                _f16(t["embed"])),
          "output_norm.weight": ((m.hidden_size,), G.GGML_F32,
                                 _f32(t["norm"])),
          "output.weight": ((m.vocab_size, m.hidden_size), G.GGML_F16,
                            _f16(t["lm_head"]))}
    for i, L in enumerate(t["layers"]):
        p = f"blk.{i}."
        ts[p + "attn_q.weight"] = (tuple(L["q"].shape), G.GGML_F16,
                                   _f16(L["q"]))
        ts[p + "attn_k.weight"] = (tuple(L["k"].shape), G.GGML_F16,
                                   _f16(L["k"]))
        ts[p + "attn_v.weight"] = (tuple(L["v"].shape), G.GGML_F16,
                                   _f16(L["v"]))
        ts[p + "attn_output.weight"] = (tuple(L["o"].shape), G.GGML_F16,
                                        _f16(L["o"]))
        ts[p + "ffn_gate.weight"] = (tuple(L["gate"].shape), G.GGML_F16,
                                     _f16(L["gate"]))
        ts[p + "ffn_up.weight"] = (tuple(L["up"].shape), G.GGML_F16,
                                   _f16(L["up"]))
        ts[p + "ffn_down.weight"] = (tuple(L["down"].shape), G.GGML_F16,
                                     _f16(L["down"]))
        ts[p + "attn_norm.weight"] = ((m.hidden_size,), G.GGML_F32,
                                      _f32(L["in_norm"]))
        ts[p + "ffn_norm.weight"] = ((m.hidden_size,), G.GGML_F32,
                                     _f32(L["post_norm"]))
    return ts


def test_parse_metadata_and_infos(tmp_path):
    m = MODEL_PRESETS["tiny-llama"]
    meta = _meta_for(m) + [
        _kv("flag", 7, b"\x01"),                       # bool
        _kv("arr", 9, struct.pack("<IQ", 4, 3)         # [u32;3]
            + struct.pack("<III", 7, 8, 9)),
    ]
    w = torch.arange(64, dtype=torch.float32).reshape(2, 32)
    write_gguf(tmp_path / "m.gguf", meta,
               {"t": ((2, 32), G.GGML_F32, _f32(w))})
    g = G.GGUFFile(str(tmp_path / "m.gguf"))
    assert g.metadata["general.architecture"] == "llama"
    assert g.metadata["flag"] is True
    assert g.metadata["arr"] == [7, 8, 9]
    assert g.infos["t"][0] == (2, 32)
    assert torch.equal(g.tensor("t"), w)


def test_q8_0_and_q4_0_dequant(tmp_path):
    torch.manual_seed(0)
    w = torch.randn(4, 64)
    raw8 = _q8_0(w)
    raw4 = _q4_0(w)
    write_gguf(tmp_path / "q.gguf",
               [_kv_str("general.architecture", "llama")],
               {"a": ((4, 64), G.GGML_Q8_0, raw8),
                "b": ((4, 64), G.GGML_Q4_0, raw4)})
    g = G.GGUFFile(str(tmp_path / "q.gguf"))
    a = g.tensor("a")
    assert torch.allclose(a.reshape(-1),
                          _q8_0_expected(raw8, 256), atol=0)
    # Q8_0 quantization error bound: d/2 per element
    assert (a - w).abs().max() < (w.abs().max() / 127) * 0.51 + 1e-6
    b = g.tensor("b")
    assert (b - w).abs().max() < (w.abs().max() / 8) * 0.51 + 1e-6
    assert b.shape == (4, 64)


def test_unsupported_quant_fails_loudly(tmp_path):
    write_gguf(tmp_path / "k.gguf",
               [_kv_str("general.architecture", "llama")],
               {"x": ((32,), 12, b"\x00" * 144)})      # Q4_K
    g = G.GGUFFile(str(tmp_path / "k.gguf"))
    with pytest.raises(ValueError, match="Q4_K"):
        g.tensor("x")


def test_config_from_gguf(tmp_path):
    m = MODEL_PRESETS["tiny-llama"]
    t = _tiny_weights(m)
    meta = _meta_for(m) + [
        _kv_str("llama.rope.scaling.type", "linear"),
        _kv_f32("llama.rope.scaling.factor", 2.0)]
    write_gguf(tmp_path / "m.gguf", meta, _gguf_tensors(m, t))
    cfg = G.model_config_from_gguf(str(tmp_path / "m.gguf"))
    assert cfg.rope_scaling == {"type": "linear", "factor": 2.0}
    assert cfg.hidden_size == m.hidden_size
    assert cfg.num_layers == m.num_layers
    assert cfg.num_kv_heads == m.num_kv_heads
    assert cfg.head_dim == m.head_dim
    assert cfg.vocab_size == m.vocab_size
    assert not cfg.tie_embeddings and not cfg.qkv_bias


def test_gguf_generation_matches_safetensors(tmp_path):
    """The SAME weights through the GGUF loader and the safetensors
    loader must generate token-identical greedy output."""
    m = MODEL_PRESETS["tiny-llama"]
    t = _tiny_weights(m, seed=11)
    # gguf form
    gdir = tmp_path / "gg"
    gdir.mkdir()
    write_gguf(gdir / "m.gguf", _meta_for(m), _gguf_tensors(m, t))
    # safetensors form (f32 of the same bf16-representable values)
    sdir = tmp_path / "st"
    sdir.mkdir()
    st = {"model.embed_tokens.weight": t["embed"],
          "model.norm.weight": t["norm"], "lm_head.weight": t["lm_head"]}
    for i, L in enumerate(t["layers"]):
        p = f"model.layers.{i}."
        st[p + "self_attn.q_proj.weight"] = L["q"]
        st[p + "self_attn.k_proj.weight"] = L["k"]
        st[p + "self_attn.v_proj.weight"] = L["v"]
        st[p + "self_attn.o_proj.weight"] = L["o"]
        st[p + "mlp.gate_proj.weight"] = L["gate"]
        st[p + "mlp.up_proj.weight"] = L["up"]
        st[p + "mlp.down_proj.weight"] = L["down"]
        st[p + "input_layernorm.weight"] = L["in_norm"]
        st[p + "post_attention_layernorm.weight"] = L["post_norm"]
    save_file(st, str(sdir / "model.safetensors"))
    (sdir / "config.json").write_text(json.dumps({
        "model_type": "llama", "hidden_size": m.hidden_size,
        "num_hidden_layers": m.num_layers,
        "num_attention_heads": m.num_heads,
        "num_key_value_heads": m.num_kv_heads, "head_dim": m.head_dim,
        "intermediate_size": m.intermediate_size,
        "vocab_size": m.vocab_size, "rope_theta": m.rope_base,
        "rms_norm_eps": m.norm_eps,
        "max_position_embeddings": m.max_position}))

    outs = []
    for d in (gdir, sdir):
        eng = LLMEngine(EngineConfig.tiny_llama(weights_path=str(d)),
                        device="cpu")
        rid = eng.add_request(list(range(1, 33)),
                              SamplingParams(max_new_tokens=8,
                                             temperature=0))
        while not eng.is_finished():
            eng.step()
        outs.append(eng.get_output(rid).token_ids)
    assert outs[0] == outs[1]


def test_gguf_tp_rank1_slices(tmp_path, monkeypatch):
    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM
    from inference_gateway_amd.parallel import comm

    m = MODEL_PRESETS["tiny-llama"]
    t = _tiny_weights(m, seed=4)
    write_gguf(tmp_path / "m.gguf", _meta_for(m), _gguf_tensors(m, t))
    monkeypatch.setattr(comm, "_TP_SIZE", 2)
    monkeypatch.setattr(comm, "_TP_RANK", 1)
    model = LlamaForCausalLM(m, "cpu", torch.bfloat16)
    G.load_gguf_weights(model, str(tmp_path / "m.gguf"))
    hd = m.head_dim
    nq, nkv = m.num_heads // 2, m.num_kv_heads // 2
    want = torch.cat([t["layers"][0]["q"][nq * hd:],
                      t["layers"][0]["k"][nkv * hd:],
                      t["layers"][0]["v"][nkv * hd:]], 0)
    assert torch.equal(model.layers[0].attn.qkv.data.float(),
                       want.to(torch.bfloat16).float())


def test_local_provider_serves_gguf(tmp_path):
    import asyncio

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider

    m = MODEL_PRESETS["tiny-llama"]
    t = _tiny_weights(m)
    write_gguf(tmp_path / "m.gguf", _meta_for(m), _gguf_tensors(m, t))
    lc = LocalConfig(enabled=True, model="gguf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path / "m.gguf")
    provider = LocalProvider(lc, device="cpu")
    try:
        async def go():
            return await provider.chat_completions({
                "model": "local/gguf",
                "messages": [{"role": "user", "content": "hello"}],
                "max_tokens": 4, "temperature": 0})

        body = asyncio.run(go())
        assert body["usage"]["completion_tokens"] == 4
    finally:
        provider.engine.shutdown()


def _kv_strarr(key, vals):
    return _kv(key, 9, struct.pack("<IQ", 8, len(vals))
               + b"".join(_s(v.encode()) for v in vals))


def test_tokenizer_from_gguf_metadata(tmp_path):
    """GPT-2-style byte-level BPE rebuilt from tokenizer.ggml.* arrays
    (what Llama-3/Qwen GGUFs embed)."""
    tokens = ["<bos>", "<eos>", "h", "i", "e", "l", "o", "hi"]
    meta = [_kv_str("general.architecture", "llama"),
            _kv_str("tokenizer.ggml.model", "gpt2"),
            _kv_strarr("tokenizer.ggml.tokens", tokens),
            _kv_strarr("tokenizer.ggml.merges", ["h i"]),
            _kv_u32("tokenizer.ggml.bos_token_id", 0),
            _kv_u32("tokenizer.ggml.eos_token_id", 1)]
    write_gguf(tmp_path / "t.gguf", meta, {})
    tok = G.tokenizer_from_gguf(str(tmp_path / "t.gguf"))
    assert tok is not None
    assert tok.bos_token_id == 0 and tok.eos_token_id == 1
    ids = tok.encode("hi")
    # bos prepended; "h"+"i" merged into the single "hi" token
    assert ids == [0, tokens.index("hi")]
    assert tok.decode([tokens.index("hi")]) == "hi"


def test_gguf_without_tokenizer_metadata_falls_back(tmp_path):
    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider
    from inference_gateway_amd.utils.tokenizer import ByteTokenizer

    m = MODEL_PRESETS["tiny-llama"]
    t = _tiny_weights(m)
    write_gguf(tmp_path / "m.gguf", _meta_for(m), _gguf_tensors(m, t))
    lc = LocalConfig(enabled=True, model="gguf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path / "m.gguf")
    provider = LocalProvider(lc, device="cpu")
    try:
        assert isinstance(provider.tokenizer, ByteTokenizer)
    finally:
        provider.engine.shutdown()


def test_gguf_qwen2_bias_and_tied_embeddings(tmp_path):
    """Qwen2-shaped GGUF: attn_q/k/v.bias tensors land in the fused
    qkv bias; a missing output.weight means tied embeddings."""
    m = MODEL_PRESETS["tiny-qwen2"]
    t = _tiny_weights(m, seed=3)
    ts = _gguf_tensors(m, t)
    del ts["output.weight"]                    # tied embeddings
    g = torch.Generator().manual_seed(99)
    hd = m.head_dim
    qb = (torch.randn(m.num_heads * hd, generator=g) * 0.02).to(
        torch.float16).float()
    kb = (torch.randn(m.num_kv_heads * hd, generator=g) * 0.02).to(
        torch.float16).float()
    vb = (torch.randn(m.num_kv_heads * hd, generator=g) * 0.02).to(
        torch.float16).float()
    for i in range(m.num_layers):
        ts[f"blk.{i}.attn_q.bias"] = ((m.num_heads * hd,), G.GGML_F32,
                                      _f32(qb))
        ts[f"blk.{i}.attn_k.bias"] = ((m.num_kv_heads * hd,),
                                      G.GGML_F32, _f32(kb))
        ts[f"blk.{i}.attn_v.bias"] = ((m.num_kv_heads * hd,),
                                      G.GGML_F32, _f32(vb))
    write_gguf(tmp_path / "m.gguf", _meta_for(m, arch="qwen2"), ts)

    cfg = G.model_config_from_gguf(str(tmp_path / "m.gguf"))
    assert cfg.qkv_bias and cfg.tie_embeddings

    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM

    model = LlamaForCausalLM(m, "cpu", torch.bfloat16)
    n = G.load_gguf_weights(model, str(tmp_path / "m.gguf"))
    assert n > 0
    want = torch.cat([qb, kb, vb], 0)
    assert torch.equal(model.layers[0].attn.qkv_b.data.float(),
                       want.to(torch.bfloat16).float())
    assert model.lm_head is None


def test_checkpoint_name_surfaces_in_models_list(tmp_path):
    """LOCAL_MODEL=gguf lists the checkpoint's general.name, not the
    literal 'gguf'."""
    import asyncio

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider

    m = MODEL_PRESETS["tiny-llama"]
    t = _tiny_weights(m)
    write_gguf(tmp_path / "m.gguf", _meta_for(m), _gguf_tensors(m, t))
    lc = LocalConfig(enabled=True, model="gguf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path / "m.gguf")
    provider = LocalProvider(lc, device="cpu")
    try:
        assert provider.model_name == "tiny"   # general.name metadata
        data = asyncio.run(provider.list_models())
        assert data["data"][0]["id"] == "local/tiny"
    finally:
        provider.engine.shutdown()
