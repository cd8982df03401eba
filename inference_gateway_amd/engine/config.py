"""Engine configuration: model architectures, cache sizing, scheduler knobs.

The reference gateway has no engine; these configs define the
net-new local provider models named by BASELINE.json: Llama-3 8B / 70B and
Mixtral 8x7B, plus tiny variants for CPU tests and smoke().
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class ModelConfig:
    name: str
    hidden_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    intermediate_size: int
    vocab_size: int
    rope_base: float = 500000.0
    max_position: int = 8192
    norm_eps: float = 1e-5
    tie_embeddings: bool = False
    # Qwen2 family: bias on the fused qkv projection only (HF Qwen2
    # hardcodes q/k/v bias True, o/mlp bias False)
    qkv_bias: bool = False
    # HF rope_scaling dict (Llama-3.1 "llama3" wavelength-banded
    # rescale, or "linear"); None = plain RoPE
    rope_scaling: Optional[dict] = None
    # MoE (Mixtral): 0 experts => dense MLP
    num_experts: int = 0
    num_experts_per_tok: int = 2

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    def kv_bytes_per_token(self) -> int:
        # bf16 K + V per layer
        return 2 * self.num_kv_heads * self.head_dim * 2 * self.num_layers


# Architectures from the public model cards (random-init weights — no network).
MODEL_PRESETS = {
    "llama-3-8b": ModelConfig(
        name="llama-3-8b",
        hidden_size=4096,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=128256,
        rope_base=500000.0,
    ),
    "llama-3-70b": ModelConfig(
        name="llama-3-70b",
        hidden_size=8192,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=28672,
        vocab_size=128256,
        rope_base=500000.0,
    ),
    "llama-3.1-8b": ModelConfig(
        name="llama-3.1-8b",
        hidden_size=4096,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=128256,
        rope_base=500000.0,
        max_position=131072,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
    ),
    "llama-3.1-70b": ModelConfig(
        name="llama-3.1-70b",
        hidden_size=8192,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=28672,
        vocab_size=128256,
        rope_base=500000.0,
        max_position=131072,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
    ),
    "mixtral-8x7b": ModelConfig(
        name="mixtral-8x7b",
        hidden_size=4096,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=32000,
        rope_base=1000000.0,
        num_experts=8,
        num_experts_per_tok=2,
    ),
    "qwen2.5-7b": ModelConfig(
        name="qwen2.5-7b",
        hidden_size=3584,
        num_layers=28,
        num_heads=28,
        num_kv_heads=4,
        head_dim=128,
        intermediate_size=18944,
        vocab_size=152064,
        rope_base=1000000.0,
        max_position=32768,
        norm_eps=1e-6,
        qkv_bias=True,
    ),
    # phi-3-MEDIUM, not mini: the MFMA attention kernels are tiled for
    # head_dim=128 (mini's 96 would fail the kernels' TORCH_CHECK on
    # GPU; engine init rejects non-128 head_dim on cuda devices)
    "phi-3-medium": ModelConfig(
        name="phi-3-medium",
        hidden_size=5120,
        num_layers=40,
        num_heads=40,
        num_kv_heads=10,
        head_dim=128,
        intermediate_size=17920,
        vocab_size=32064,
        rope_base=10000.0,
        max_position=4096,
    ),
    # tiny models for CPU tests / smoke
    "tiny-llama": ModelConfig(
        name="tiny-llama",
        hidden_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        intermediate_size=1024,
        vocab_size=512,
        rope_base=10000.0,
        max_position=2048,
    ),
    "tiny-qwen2": ModelConfig(
        name="tiny-qwen2",
        hidden_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        intermediate_size=1024,
        vocab_size=512,
        rope_base=10000.0,
        max_position=2048,
        qkv_bias=True,
        tie_embeddings=True,
    ),
    "tiny-moe": ModelConfig(
        name="tiny-moe",
        hidden_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=128,
        intermediate_size=512,
        vocab_size=512,
        rope_base=10000.0,
        max_position=2048,
        num_experts=4,
        num_experts_per_tok=2,
    ),
}


@dataclass
class SamplingParams:
    max_new_tokens: int = 128
    temperature: float = 1.0
    top_k: int = 0          # <=0 disables
    top_p: float = 1.0      # >=1 disables
    stop_token_ids: tuple = ()
    ignore_eos: bool = False
    # OpenAI logprobs: -1 = off; >=0 = return the sampled token's logprob
    # plus that many top alternatives per step (capped at 20, like OpenAI)
    logprobs: int = -1
    # OpenAI seed: per-request deterministic sampling.  The per-seq tick
    # keeps counting across preemption/rebuild, so a seeded request is
    # reproducible even under KV pressure (tested). None = engine RNG.
    seed: Optional[int] = None
    # OpenAI repetition penalties (0 = off). Exact on the eager path; on
    # the async graph path the host's token view lags one step, so the
    # in-flight token's penalty is applied device-side (see sampler.py).
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    # OpenAI logit_bias: {token_id: bias in [-100, 100]} added to the
    # logits every step (None = off)
    logit_bias: Optional[dict] = None


@dataclass
class EngineConfig:
    model: ModelConfig
    block_size: int = 16
    num_gpu_blocks: Optional[int] = None   # None => size from free HBM
    gpu_memory_utilization: float = 0.90
    max_num_seqs: int = 256
    max_batched_tokens: int = 8192
    max_model_len: int = 8192
    use_hip_graphs: bool = True
    enforce_eager: bool = False
    # serving mode: capture every decode bucket at engine init so no
    # client ever pays a ~1 s first-capture stall mid-stream (the bench
    # path leaves this off — it uses one fixed bucket)
    capture_at_init: bool = False
    # prefix caching: hash-indexed reuse of full prompt blocks across
    # requests (suffix-only extend-prefill).  Opt-in.
    enable_prefix_caching: bool = False
    # chunked prefill: max_batched_tokens becomes a HARD budget — a long
    # prompt prefills in successive chunks (paged-extend attention) so a
    # burst never stalls in-flight decodes for a whole long prefill.
    enable_chunked_prefill: bool = False
    # strict admission priority: always prefill when prompts wait
    # (vLLM-style, TTFT-optimal under bursts) instead of alternating
    # prefill/decode rounds (ITL-optimal).  See scheduler.py.
    scheduler_prefill_first: bool = False
    # same-step mixed batches: a round with pending prompts carries the
    # (chunked) prefill segment AND every running decode row in one
    # forward — no whole-round ITL bubble (scheduler.py; the mixed
    # round runs eagerly, pure decode keeps the hipGraph)
    enable_mixed_batches: bool = False
    # real checkpoint: HF-format directory (config.json +
    # *.safetensors); None keeps the seed-derived synthetic init
    # (engine/weights.py — the no-network default)
    weights_path: Optional[str] = None
    tensor_parallel_size: int = 1
    seed: int = 0

    @staticmethod
    def tiny_llama(**kw) -> "EngineConfig":
        defaults = dict(
            model=MODEL_PRESETS["tiny-llama"],
            num_gpu_blocks=64,
            max_num_seqs=8,
            max_batched_tokens=2048,
            max_model_len=512,
            use_hip_graphs=False,
        )
        defaults.update(kw)
        return EngineConfig(**defaults)

    @staticmethod
    def tiny_moe(**kw) -> "EngineConfig":
        defaults = dict(
            model=MODEL_PRESETS["tiny-moe"],
            num_gpu_blocks=64,
            max_num_seqs=8,
            max_batched_tokens=2048,
            max_model_len=512,
            use_hip_graphs=False,
        )
        defaults.update(kw)
        return EngineConfig(**defaults)

    @staticmethod
    def for_model(name: str, **kw) -> "EngineConfig":
        return EngineConfig(model=MODEL_PRESETS[name], **kw)
