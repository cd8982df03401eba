"""Real-checkpoint loading (engine/weights.py): HF config ingestion,
safetensors mapping onto the fused/TP-sharded engine layout, loud
failure on mismatch, end-to-end generation from a loaded checkpoint
(weight loading is the stateless gateway's
checkpoint/resume analog)."""
import json

import pytest
import torch
from safetensors.torch import save_file

from inference_gateway_amd.engine.config import (MODEL_PRESETS,
                                                 EngineConfig,
                                                 SamplingParams)
from inference_gateway_amd.engine.engine import LLMEngine
from inference_gateway_amd.engine import weights as W


def _tiny_llama_checkpoint(dirpath, seed=0):
    """Fabricate a tiny-llama HF-format checkpoint (random weights)."""
    m = MODEL_PRESETS["tiny-llama"]
    g = torch.Generator().manual_seed(seed)

    def r(*shape):
        return torch.randn(*shape, generator=g) * 0.02

    h, hd = m.hidden_size, m.head_dim
    t = {"model.embed_tokens.weight": r(m.vocab_size, h),
         "model.norm.weight": torch.ones(h),
         "lm_head.weight": r(m.vocab_size, h)}
    for i in range(m.num_layers):
        p = f"model.layers.{i}."
        t[p + "self_attn.q_proj.weight"] = r(m.num_heads * hd, h)
        t[p + "self_attn.k_proj.weight"] = r(m.num_kv_heads * hd, h)
        t[p + "self_attn.v_proj.weight"] = r(m.num_kv_heads * hd, h)
        t[p + "self_attn.o_proj.weight"] = r(h, m.num_heads * hd)
        t[p + "mlp.gate_proj.weight"] = r(m.intermediate_size, h)
        t[p + "mlp.up_proj.weight"] = r(m.intermediate_size, h)
        t[p + "mlp.down_proj.weight"] = r(h, m.intermediate_size)
        t[p + "input_layernorm.weight"] = torch.ones(h)
        t[p + "post_attention_layernorm.weight"] = torch.ones(h)
    save_file(t, str(dirpath / "model.safetensors"))
    (dirpath / "config.json").write_text(json.dumps({
        "model_type": "llama",
        "hidden_size": h,
        "num_hidden_layers": m.num_layers,
        "num_attention_heads": m.num_heads,
        "num_key_value_heads": m.num_kv_heads,
        "head_dim": hd,
        "intermediate_size": m.intermediate_size,
        "vocab_size": m.vocab_size,
        "rope_theta": m.rope_base,
        "rms_norm_eps": m.norm_eps,
        "max_position_embeddings": m.max_position,
    }))
    return t


def test_hf_config_ingestion(tmp_path):
    _tiny_llama_checkpoint(tmp_path)
    cfg = W.model_config_from_hf(str(tmp_path))
    m = MODEL_PRESETS["tiny-llama"]
    assert cfg.hidden_size == m.hidden_size
    assert cfg.num_layers == m.num_layers
    assert cfg.num_kv_heads == m.num_kv_heads
    assert cfg.head_dim == m.head_dim
    assert cfg.rope_base == m.rope_base
    assert not cfg.is_moe


def test_llama_weights_load_and_generate(tmp_path):
    t = _tiny_llama_checkpoint(tmp_path)
    eng = LLMEngine(EngineConfig.tiny_llama(weights_path=str(tmp_path)),
                    device="cpu")
    model = eng.model
    # fused qkv = [q; k; v] in row order
    want = torch.cat([t["model.layers.0.self_attn.q_proj.weight"],
                      t["model.layers.0.self_attn.k_proj.weight"],
                      t["model.layers.0.self_attn.v_proj.weight"]], 0)
    assert torch.equal(model.layers[0].attn.qkv.data.float(),
                       want.to(torch.bfloat16).float())
    # fused gate_up = [gate; up]
    want = torch.cat([t["model.layers.1.mlp.gate_proj.weight"],
                      t["model.layers.1.mlp.up_proj.weight"]], 0)
    assert torch.equal(model.layers[1].mlp.gate_up.data.float(),
                       want.to(torch.bfloat16).float())
    assert torch.equal(model.embed.data.float(),
                       t["model.embed_tokens.weight"]
                       .to(torch.bfloat16).float())
    # the engine actually generates from the loaded checkpoint
    rid = eng.add_request(list(range(1, 17)),
                          SamplingParams(max_new_tokens=4, temperature=0))
    while not eng.is_finished():
        eng.step()
    out = eng.get_output(rid)
    assert out.finished and len(out.token_ids) == 4


def test_tp_sharded_load_matches_manual_slices(tmp_path, monkeypatch):
    """Rank 1 of TP=2 must hold exactly the second half of the head/
    intermediate dimensions (sliced reads, not full reads)."""
    from inference_gateway_amd.parallel import comm
    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM

    t = _tiny_llama_checkpoint(tmp_path)
    m = MODEL_PRESETS["tiny-llama"]
    monkeypatch.setattr(comm, "_TP_SIZE", 2)
    monkeypatch.setattr(comm, "_TP_RANK", 1)
    model = LlamaForCausalLM(m, "cpu", torch.bfloat16)
    W.load_weights(model, str(tmp_path))
    hd = m.head_dim
    nq, nkv = m.num_heads // 2, m.num_kv_heads // 2
    q = t["model.layers.0.self_attn.q_proj.weight"][nq * hd:]
    k = t["model.layers.0.self_attn.k_proj.weight"][nkv * hd:]
    v = t["model.layers.0.self_attn.v_proj.weight"][nkv * hd:]
    assert torch.equal(model.layers[0].attn.qkv.data.float(),
                       torch.cat([q, k, v], 0).to(torch.bfloat16).float())
    o = t["model.layers.0.self_attn.o_proj.weight"][:, nq * hd:]
    assert torch.equal(model.layers[0].attn.o_proj.weight.data.float(),
                       o.to(torch.bfloat16).float())


def test_mismatched_checkpoint_fails_loudly(tmp_path):
    _tiny_llama_checkpoint(tmp_path)
    # break one tensor's shape
    bad = {"model.embed_tokens.weight": torch.zeros(7, 7)}
    save_file(bad, str(tmp_path / "model.safetensors"))
    with pytest.raises(Exception):
        LLMEngine(EngineConfig.tiny_llama(weights_path=str(tmp_path)),
                  device="cpu")


def test_moe_weights_load_and_generate(tmp_path):
    m = MODEL_PRESETS["tiny-moe"]
    g = torch.Generator().manual_seed(3)

    def r(*shape):
        return torch.randn(*shape, generator=g) * 0.02

    h, hd = m.hidden_size, m.head_dim
    t = {"model.embed_tokens.weight": r(m.vocab_size, h),
         "model.norm.weight": torch.ones(h),
         "lm_head.weight": r(m.vocab_size, h)}
    for i in range(m.num_layers):
        p = f"model.layers.{i}."
        t[p + "self_attn.q_proj.weight"] = r(m.num_heads * hd, h)
        t[p + "self_attn.k_proj.weight"] = r(m.num_kv_heads * hd, h)
        t[p + "self_attn.v_proj.weight"] = r(m.num_kv_heads * hd, h)
        t[p + "self_attn.o_proj.weight"] = r(h, m.num_heads * hd)
        t[p + "block_sparse_moe.gate.weight"] = r(m.num_experts, h)
        for e in range(m.num_experts):
            ep = p + f"block_sparse_moe.experts.{e}."
            t[ep + "w1.weight"] = r(m.intermediate_size, h)
            t[ep + "w3.weight"] = r(m.intermediate_size, h)
            t[ep + "w2.weight"] = r(h, m.intermediate_size)
        t[p + "input_layernorm.weight"] = torch.ones(h)
        t[p + "post_attention_layernorm.weight"] = torch.ones(h)
    save_file(t, str(tmp_path / "model.safetensors"))
    eng = LLMEngine(EngineConfig.tiny_moe(weights_path=str(tmp_path)),
                    device="cpu")
    gu = eng.model.layers[0].moe.gate_up.data
    want = torch.cat(
        [t["model.layers.0.block_sparse_moe.experts.1.w1.weight"],
         t["model.layers.0.block_sparse_moe.experts.1.w3.weight"]], 0)
    assert torch.equal(gu[1].float(), want.to(torch.bfloat16).float())
    rid = eng.add_request(list(range(1, 17)),
                          SamplingParams(max_new_tokens=3, temperature=0))
    while not eng.is_finished():
        eng.step()
    assert len(eng.get_output(rid).token_ids) == 3


def test_local_provider_serves_from_checkpoint(tmp_path):
    """LOCAL_WEIGHTS_PATH + LOCAL_MODEL=hf: the provider derives the
    architecture from config.json and serves real-checkpoint tokens."""
    import asyncio

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider

    _tiny_llama_checkpoint(tmp_path)
    lc = LocalConfig(enabled=True, model="hf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path)
    provider = LocalProvider(lc, device="cpu")
    try:
        async def go():
            return await provider.chat_completions({
                "model": "local/hf",
                "messages": [{"role": "user", "content": "hello"}],
                "max_tokens": 4, "temperature": 0})

        body = asyncio.run(go())
        assert body["usage"]["completion_tokens"] == 4
    finally:
        provider.engine.shutdown()


def _tiny_qwen2_checkpoint(dirpath, seed=5):
    """Fabricate a tiny-qwen2 HF checkpoint: qkv biases, tied
    embeddings (no lm_head tensor), model_type=qwen2 with NO
    attention_bias field (HF Qwen2 hardcodes qkv bias True)."""
    m = MODEL_PRESETS["tiny-qwen2"]
    g = torch.Generator().manual_seed(seed)

    def r(*shape):
        return torch.randn(*shape, generator=g) * 0.02

    h, hd = m.hidden_size, m.head_dim
    t = {"model.embed_tokens.weight": r(m.vocab_size, h),
         "model.norm.weight": torch.ones(h)}
    for i in range(m.num_layers):
        p = f"model.layers.{i}."
        t[p + "self_attn.q_proj.weight"] = r(m.num_heads * hd, h)
        t[p + "self_attn.q_proj.bias"] = r(m.num_heads * hd)
        t[p + "self_attn.k_proj.weight"] = r(m.num_kv_heads * hd, h)
        t[p + "self_attn.k_proj.bias"] = r(m.num_kv_heads * hd)
        t[p + "self_attn.v_proj.weight"] = r(m.num_kv_heads * hd, h)
        t[p + "self_attn.v_proj.bias"] = r(m.num_kv_heads * hd)
        t[p + "self_attn.o_proj.weight"] = r(h, m.num_heads * hd)
        t[p + "mlp.gate_proj.weight"] = r(m.intermediate_size, h)
        t[p + "mlp.up_proj.weight"] = r(m.intermediate_size, h)
        t[p + "mlp.down_proj.weight"] = r(h, m.intermediate_size)
        t[p + "input_layernorm.weight"] = torch.ones(h)
        t[p + "post_attention_layernorm.weight"] = torch.ones(h)
    save_file(t, str(dirpath / "model.safetensors"))
    (dirpath / "config.json").write_text(json.dumps({
        "model_type": "qwen2",
        "hidden_size": h,
        "num_hidden_layers": m.num_layers,
        "num_attention_heads": m.num_heads,
        "num_key_value_heads": m.num_kv_heads,
        "head_dim": hd,
        "intermediate_size": m.intermediate_size,
        "vocab_size": m.vocab_size,
        "rope_theta": m.rope_base,
        "rms_norm_eps": m.norm_eps,
        "max_position_embeddings": m.max_position,
        "tie_word_embeddings": True,
    }))
    return t


def test_qwen2_config_ingestion(tmp_path):
    _tiny_qwen2_checkpoint(tmp_path)
    cfg = W.model_config_from_hf(str(tmp_path))
    assert cfg.qkv_bias and cfg.tie_embeddings
    assert cfg.name == "qwen2"


def test_qwen2_weights_load_and_generate(tmp_path):
    t = _tiny_qwen2_checkpoint(tmp_path)
    eng = LLMEngine(
        EngineConfig(model=MODEL_PRESETS["tiny-qwen2"], max_num_seqs=4,
                     max_model_len=128, num_gpu_blocks=64,
                     weights_path=str(tmp_path)),
        device="cpu")
    attn = eng.model.layers[0].attn
    want = torch.cat([t["model.layers.0.self_attn.q_proj.bias"],
                      t["model.layers.0.self_attn.k_proj.bias"],
                      t["model.layers.0.self_attn.v_proj.bias"]], 0)
    assert torch.equal(attn.qkv_b.data.float(),
                       want.to(torch.bfloat16).float())
    # tied embeddings: no lm_head parameter, logits via embed
    assert eng.model.lm_head is None
    rid = eng.add_request(list(range(1, 17)),
                          SamplingParams(max_new_tokens=4, temperature=0))
    while not eng.is_finished():
        eng.step()
    assert len(eng.get_output(rid).token_ids) == 4
    # the loaded biases actually shape the output: zeroing them changes
    # the greedy continuation computation (guards against a silently
    # ignored bias)
    with torch.no_grad():
        before = attn.qkv_b.data.clone()
        import inference_gateway_amd.engine.models.batch as _b  # noqa
        x = torch.randn(3, eng.model.cfg.hidden_size).to(attn.qkv.dtype)
        y1 = torch.nn.functional.linear(x, attn.qkv, attn.qkv_b)
        y2 = torch.nn.functional.linear(x, attn.qkv)
        assert not torch.allclose(y1, y2)
        assert torch.equal(attn.qkv_b.data, before)


def test_qwen2_bias_architecture_mismatches_fail_loudly(tmp_path):
    # llama model config + bias-carrying checkpoint -> loud error
    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM

    _tiny_qwen2_checkpoint(tmp_path)
    cfg_nobias = MODEL_PRESETS["tiny-qwen2"].__class__(
        **{**MODEL_PRESETS["tiny-qwen2"].__dict__, "qkv_bias": False})
    model = LlamaForCausalLM(cfg_nobias, "cpu", torch.bfloat16)
    with pytest.raises(ValueError, match="qkv_bias=False"):
        W.load_weights(model, str(tmp_path))


def test_qwen2_tp_sharded_bias_slices(tmp_path, monkeypatch):
    """Rank 1 of TP=2 holds the second half of each of q/k/v bias."""
    from inference_gateway_amd.parallel import comm
    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM

    t = _tiny_qwen2_checkpoint(tmp_path)
    m = MODEL_PRESETS["tiny-qwen2"]
    monkeypatch.setattr(comm, "_TP_SIZE", 2)
    monkeypatch.setattr(comm, "_TP_RANK", 1)
    model = LlamaForCausalLM(m, "cpu", torch.bfloat16)
    W.load_weights(model, str(tmp_path))
    hd = m.head_dim
    nq, nkv = m.num_heads // 2, m.num_kv_heads // 2
    qb = t["model.layers.0.self_attn.q_proj.bias"][nq * hd:]
    kb = t["model.layers.0.self_attn.k_proj.bias"][nkv * hd:]
    vb = t["model.layers.0.self_attn.v_proj.bias"][nkv * hd:]
    assert torch.equal(model.layers[0].attn.qkv_b.data.float(),
                       torch.cat([qb, kb, vb], 0)
                       .to(torch.bfloat16).float())


def test_tokenizer_json_autodetected_next_to_checkpoint(tmp_path):
    """LOCAL_MODEL=hf with a tokenizer.json in the checkpoint dir uses
    the real tokenizer without LOCAL_TOKENIZER_PATH."""
    from tokenizers import Tokenizer, models, pre_tokenizers

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider
    from inference_gateway_amd.utils.tokenizer import HFTokenizer

    _tiny_llama_checkpoint(tmp_path)
    vocab = {"<unk>": 0, "<s>": 1, "</s>": 2, "hello": 3, "world": 4}
    tok = Tokenizer(models.WordLevel(vocab, unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    tok.save(str(tmp_path / "tokenizer.json"))

    lc = LocalConfig(enabled=True, model="hf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path)
    provider = LocalProvider(lc, device="cpu")
    try:
        assert isinstance(provider.tokenizer, HFTokenizer)
        ids = provider.tokenizer.encode("hello world")
        assert 3 in ids and 4 in ids
    finally:
        provider.engine.shutdown()


def _fused_phi3_style_checkpoint(dirpath, seed=7):
    """Same tiny-llama architecture but with HF Phi-3's PRE-FUSED
    tensor naming (self_attn.qkv_proj, mlp.gate_up_proj).  Returns the
    equivalent UNFUSED tensor dict for cross-checking."""
    m = MODEL_PRESETS["tiny-llama"]
    g = torch.Generator().manual_seed(seed)

    def r(*shape):
        return torch.randn(*shape, generator=g) * 0.02

    h, hd = m.hidden_size, m.head_dim
    unfused = {"model.embed_tokens.weight": r(m.vocab_size, h),
               "model.norm.weight": torch.ones(h),
               "lm_head.weight": r(m.vocab_size, h)}
    fused = dict(unfused)
    for i in range(m.num_layers):
        p = f"model.layers.{i}."
        q = r(m.num_heads * hd, h)
        k = r(m.num_kv_heads * hd, h)
        v = r(m.num_kv_heads * hd, h)
        gate = r(m.intermediate_size, h)
        up = r(m.intermediate_size, h)
        unfused[p + "self_attn.q_proj.weight"] = q
        unfused[p + "self_attn.k_proj.weight"] = k
        unfused[p + "self_attn.v_proj.weight"] = v
        unfused[p + "mlp.gate_proj.weight"] = gate
        unfused[p + "mlp.up_proj.weight"] = up
        fused[p + "self_attn.qkv_proj.weight"] = torch.cat([q, k, v], 0)
        fused[p + "mlp.gate_up_proj.weight"] = torch.cat([gate, up], 0)
        for name, t in ((p + "self_attn.o_proj.weight",
                         r(h, m.num_heads * hd)),
                        (p + "mlp.down_proj.weight",
                         r(h, m.intermediate_size)),
                        (p + "input_layernorm.weight", torch.ones(h)),
                        (p + "post_attention_layernorm.weight",
                         torch.ones(h))):
            unfused[name] = t
            fused[name] = t
    save_file(fused, str(dirpath / "model.safetensors"))
    (dirpath / "config.json").write_text(json.dumps({
        "model_type": "phi3",
        "hidden_size": h,
        "num_hidden_layers": m.num_layers,
        "num_attention_heads": m.num_heads,
        "num_key_value_heads": m.num_kv_heads,
        "head_dim": hd,
        "intermediate_size": m.intermediate_size,
        "vocab_size": m.vocab_size,
        "rope_theta": m.rope_base,
        "rms_norm_eps": m.norm_eps,
        "max_position_embeddings": m.max_position,
        "attention_bias": False,
    }))
    return unfused


def test_phi3_fused_checkpoint_loads(tmp_path):
    """Pre-fused qkv_proj/gate_up_proj tensors (HF Phi-3 layout) land
    in the same engine parameters as the unfused layout."""
    t = _fused_phi3_style_checkpoint(tmp_path)
    eng = LLMEngine(EngineConfig.tiny_llama(weights_path=str(tmp_path)),
                    device="cpu")
    want = torch.cat([t["model.layers.0.self_attn.q_proj.weight"],
                      t["model.layers.0.self_attn.k_proj.weight"],
                      t["model.layers.0.self_attn.v_proj.weight"]], 0)
    assert torch.equal(eng.model.layers[0].attn.qkv.data.float(),
                       want.to(torch.bfloat16).float())
    want = torch.cat([t["model.layers.1.mlp.gate_proj.weight"],
                      t["model.layers.1.mlp.up_proj.weight"]], 0)
    assert torch.equal(eng.model.layers[1].mlp.gate_up.data.float(),
                       want.to(torch.bfloat16).float())
    rid = eng.add_request(list(range(1, 17)),
                          SamplingParams(max_new_tokens=3, temperature=0))
    while not eng.is_finished():
        eng.step()
    assert len(eng.get_output(rid).token_ids) == 3


def test_phi3_fused_tp_rank1_sections(tmp_path, monkeypatch):
    """TP=2 rank 1 takes the SECOND HALF OF EACH SECTION inside the
    pre-fused tensors (not the second half of the whole tensor)."""
    from inference_gateway_amd.parallel import comm
    from inference_gateway_amd.engine.models.llama import LlamaForCausalLM

    t = _fused_phi3_style_checkpoint(tmp_path)
    m = MODEL_PRESETS["tiny-llama"]
    monkeypatch.setattr(comm, "_TP_SIZE", 2)
    monkeypatch.setattr(comm, "_TP_RANK", 1)
    model = LlamaForCausalLM(m, "cpu", torch.bfloat16)
    W.load_weights(model, str(tmp_path))
    hd = m.head_dim
    nq, nkv = m.num_heads // 2, m.num_kv_heads // 2
    want = torch.cat(
        [t["model.layers.0.self_attn.q_proj.weight"][nq * hd:],
         t["model.layers.0.self_attn.k_proj.weight"][nkv * hd:],
         t["model.layers.0.self_attn.v_proj.weight"][nkv * hd:]], 0)
    assert torch.equal(model.layers[0].attn.qkv.data.float(),
                       want.to(torch.bfloat16).float())
    inter = m.intermediate_size // 2
    want = torch.cat(
        [t["model.layers.0.mlp.gate_proj.weight"][inter:],
         t["model.layers.0.mlp.up_proj.weight"][inter:]], 0)
    assert torch.equal(model.layers[0].mlp.gate_up.data.float(),
                       want.to(torch.bfloat16).float())


def test_chat_template_from_checkpoint(tmp_path):
    """tokenizer_config.json chat_template (jinja) drives prompt
    construction — a Llama-3-style template renders role headers."""
    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider
    from inference_gateway_amd.utils.tokenizer import (
        load_chat_template, render_chat_template)

    _tiny_llama_checkpoint(tmp_path)
    tpl = ("{% for m in messages %}<|start|>{{ m['role'] }}\n"
           "{{ m['content'] }}<|end|>{% endfor %}"
           "{% if add_generation_prompt %}<|start|>assistant\n"
           "{% endif %}")
    (tmp_path / "tokenizer_config.json").write_text(
        json.dumps({"chat_template": tpl}))
    assert load_chat_template(str(tmp_path)) == tpl
    out = render_chat_template(tpl, [
        {"role": "user", "content": [{"type": "text", "text": "hi"}]}])
    assert out == "<|start|>user\nhi<|end|><|start|>assistant\n"

    lc = LocalConfig(enabled=True, model="hf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path)
    provider = LocalProvider(lc, device="cpu")
    try:
        assert provider.chat_template == tpl
        ids = provider._encode_request(
            {"messages": [{"role": "user", "content": "hi"}]})
        want = provider.tokenizer.encode(
            "<|start|>user\nhi<|end|><|start|>assistant\n")
        assert ids == want
    finally:
        provider.engine.shutdown()


def test_llama31_rope_scaling():
    """Llama-3.1 "llama3" rope scaling: short wavelengths untouched,
    wavelengths beyond original_max/low_freq_factor divided by
    `factor`, smooth ramp between — verified against the published
    piecewise formula computed independently here."""
    import math

    from inference_gateway_amd import ops

    hd, base = 128, 500000.0
    rs = {"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
          "high_freq_factor": 4.0,
          "original_max_position_embeddings": 8192}
    plain = ops.build_cos_sin_cache(hd, 64, base)
    scaled = ops.build_cos_sin_cache(hd, 64, base, rope_scaling=rs)
    half = hd // 2

    def freq_at(cache, i):
        # angle(t=1) = f_i, read via atan2(sin, cos) — fp32-relative
        # precision even for tiny frequencies (acos of cos~1 is not)
        return math.atan2(cache[1, half + i].item(),
                          cache[1, i].item())

    for i in range(half):
        f = 1.0 / (base ** (i / half))
        wl = 2 * math.pi / f
        want_f = None
        if wl < 8192 / 4.0:                  # high-freq: untouched
            want_f = f
        elif wl > 8192 / 1.0:                # low-freq: / factor
            want_f = f / 8.0
        if want_f is not None:
            got = freq_at(scaled, i)
            assert abs(got - want_f) < 1e-6 + 1e-4 * want_f, (
                i, got, want_f)
    # the ramp band lies strictly between the two regimes
    mid = [i for i in range(half)
           if 8192 / 4.0 <= 2 * math.pi / (1.0 / (base ** (i / half)))
           <= 8192]
    assert mid, "test setup: ramp band must be non-empty"
    for i in mid:
        f = 1.0 / (base ** (i / half))
        got = freq_at(scaled, i)
        assert f / 8.0 - 1e-9 <= got <= f + 1e-9
    # linear scaling divides everything
    lin = ops.build_cos_sin_cache(hd, 64, base,
                                  rope_scaling={"type": "linear",
                                                "factor": 2.0})
    assert abs(freq_at(lin, 0) - freq_at(plain, 0) / 2.0) < 1e-6
    # unknown scheme fails loudly
    with pytest.raises(ValueError, match="yarn"):
        ops.build_cos_sin_cache(hd, 64, base,
                                rope_scaling={"rope_type": "yarn",
                                              "factor": 4.0})


def test_rope_scaling_hf_ingestion_and_generation(tmp_path):
    _tiny_llama_checkpoint(tmp_path)
    cfgj = json.loads((tmp_path / "config.json").read_text())
    cfgj["rope_scaling"] = {"rope_type": "llama3", "factor": 8.0,
                            "low_freq_factor": 1.0,
                            "high_freq_factor": 4.0,
                            "original_max_position_embeddings": 1024}
    (tmp_path / "config.json").write_text(json.dumps(cfgj))
    cfg = W.model_config_from_hf(str(tmp_path))
    assert cfg.rope_scaling and cfg.rope_scaling["factor"] == 8.0
    eng = LLMEngine(EngineConfig(model=cfg, max_num_seqs=2,
                                 max_model_len=128, num_gpu_blocks=32,
                                 weights_path=str(tmp_path)),
                    device="cpu")
    rid = eng.add_request(list(range(1, 17)),
                          SamplingParams(max_new_tokens=3, temperature=0))
    while not eng.is_finished():
        eng.step()
    assert len(eng.get_output(rid).token_ids) == 3
    # "default"/absent rope_type is treated as no scaling
    cfgj["rope_scaling"] = {"rope_type": "default"}
    (tmp_path / "config.json").write_text(json.dumps(cfgj))
    assert W.model_config_from_hf(str(tmp_path)).rope_scaling is None


def test_checkpoint_eos_ids_from_generation_config(tmp_path):
    """generation_config.json eos_token_id (list form — Llama-3-
    Instruct/Qwen2 style) drives stop detection; forced via logit_bias
    so the sampled token IS the configured eos."""
    import asyncio

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider

    _tiny_llama_checkpoint(tmp_path)
    (tmp_path / "generation_config.json").write_text(
        json.dumps({"eos_token_id": [5, 9]}))
    lc = LocalConfig(enabled=True, model="hf", max_num_seqs=4,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.weights_path = str(tmp_path)
    provider = LocalProvider(lc, device="cpu")
    try:
        assert provider.eos_ids == (5, 9)

        async def go():
            return await provider.chat_completions({
                "model": "local/hf",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 30, "temperature": 0,
                "logit_bias": {"9": 100}})

        body = asyncio.run(go())
        assert body["usage"]["completion_tokens"] == 1
        assert body["choices"][0]["finish_reason"] == "stop"
    finally:
        provider.engine.shutdown()


def test_multi_shard_safetensors_checkpoint(tmp_path):
    """Real 8B/70B checkpoints ship model-0000N-of-M.safetensors; the
    shard index must resolve tensors across files and generate
    identically to the single-file layout."""
    t = _tiny_llama_checkpoint(tmp_path)
    single = LLMEngine(EngineConfig.tiny_llama(weights_path=str(tmp_path)),
                       device="cpu")
    rid = single.add_request(list(range(1, 25)),
                             SamplingParams(max_new_tokens=5,
                                            temperature=0))
    while not single.is_finished():
        single.step()
    want = single.get_output(rid).token_ids

    sharded = tmp_path / "sharded"
    sharded.mkdir()
    names = sorted(t)
    half = len(names) // 2
    save_file({k: t[k] for k in names[:half]},
              str(sharded / "model-00001-of-00002.safetensors"))
    save_file({k: t[k] for k in names[half:]},
              str(sharded / "model-00002-of-00002.safetensors"))
    (sharded / "config.json").write_text(
        (tmp_path / "config.json").read_text())
    eng = LLMEngine(EngineConfig.tiny_llama(weights_path=str(sharded)),
                    device="cpu")
    rid = eng.add_request(list(range(1, 25)),
                          SamplingParams(max_new_tokens=5, temperature=0))
    while not eng.is_finished():
        eng.step()
    assert eng.get_output(rid).token_ids == want
