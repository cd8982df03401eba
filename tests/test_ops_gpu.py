"""GPU numerics tests: every hand-written gfx950 HIP kernel vs the plain
PyTorch fp32 reference (SURVEY.md §4 'GPU-kernel test tier').

All tensors use asymmetric random data (guide §5.4 rule 16: symmetric inputs
miss operand/output transposes).
"""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

from inference_gateway_amd import ops
from inference_gateway_amd.ops import torch_ref


def _assert_close_bf16(got, ref, atol=2e-2, rtol=2e-2):
    torch.testing.assert_close(
        got.float().cpu(), ref.float().cpu(), atol=atol, rtol=rtol
    )


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(1234)


def test_kernels_extension_loaded():
    # On a GPU box the native extension must be present — no eager fallback.
    assert ops.HAS_KERNELS, "gfx950 _kernels extension not built/loaded"


@pytest.mark.parametrize("hidden", [4096, 8192, 6144])
def test_rmsnorm(hidden):
    x = torch.randn(33, hidden, dtype=torch.bfloat16, device="cuda") * 3
    w = torch.randn(hidden, dtype=torch.bfloat16, device="cuda")
    out = ops.rmsnorm(x, w, 1e-5)
    ref = torch_ref.rmsnorm_ref(x.cpu(), w.cpu(), 1e-5)
    _assert_close_bf16(out, ref)


def test_fused_add_rmsnorm():
    hidden = 4096
    x = torch.randn(17, hidden, dtype=torch.bfloat16, device="cuda")
    res = torch.randn(17, hidden, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(hidden, dtype=torch.bfloat16, device="cuda")
    xr, rr = torch_ref.fused_add_rmsnorm_ref(x.cpu(), res.cpu(), w.cpu(), 1e-5)
    ops.fused_add_rmsnorm(x, res, w, 1e-5)
    _assert_close_bf16(x, xr)
    _assert_close_bf16(res, rr)


def test_rope():
    T, nq, nkv, hd = 41, 8, 2, 128
    q = torch.randn(T, nq * hd, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(T, nkv * hd, dtype=torch.bfloat16, device="cuda")
    pos = torch.randint(0, 2048, (T,), dtype=torch.int32, device="cuda")
    cs = torch_ref.build_cos_sin_cache(hd, 4096, device="cuda")
    qr, kr = torch_ref.rope_ref(pos.cpu(), q.cpu(), k.cpu(), hd, cs.cpu())
    ops.rotary_embedding(pos, q, k, hd, cs)
    _assert_close_bf16(q, qr)
    _assert_close_bf16(k, kr)


def test_append_kv_cache():
    T, nkv, bs, hd, nb = 37, 4, 16, 128, 64
    k = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device="cuda")
    kc = torch.zeros(nb, nkv, bs, hd, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros_like(kc)
    slots = torch.randperm(nb * bs, device="cuda")[:T].to(torch.int32)
    kc_ref, vc_ref = kc.cpu().clone(), vc.cpu().clone()
    torch_ref.append_kv_cache_ref(k.cpu(), v.cpu(), kc_ref, vc_ref, slots.cpu())
    ops.append_kv_cache(k, v, kc, vc, slots)
    _assert_close_bf16(kc, kc_ref, atol=0, rtol=0)
    _assert_close_bf16(vc, vc_ref, atol=0, rtol=0)


@pytest.mark.parametrize("G", [1, 4, 8])
def test_paged_attn_decode(G):
    torch.manual_seed(7 + G)
    B, nkv, bs, hd = 5, 8, 16, 128
    nq = nkv * G
    ctx = torch.tensor([1, 17, 16, 300, 1024], dtype=torch.int32, device="cuda")
    max_blocks = 1024 // bs
    nb = B * max_blocks + 8
    q = torch.randn(B, nq, hd, dtype=torch.bfloat16, device="cuda")
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device="cuda")
    perm = torch.randperm(nb, device="cuda")[: B * max_blocks].to(torch.int32)
    bt = perm.view(B, max_blocks)
    scale = hd**-0.5
    out = ops.paged_attn_decode(q, kc, vc, bt, ctx, scale)
    ref = torch_ref.paged_attn_decode_ref(
        q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(), ctx.cpu(), scale
    )
    _assert_close_bf16(out, ref)


@pytest.mark.parametrize(
    "seq_lens", [[64], [1, 37, 64, 200], [128, 1024], [63, 65, 129]]
)
def test_flash_prefill(seq_lens):
    torch.manual_seed(len(seq_lens))
    nq, nkv, hd = 8, 2, 128
    T = sum(seq_lens)
    q = torch.randn(T, nq, hd, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device="cuda")
    cu = torch.tensor(
        [0] + list(torch.cumsum(torch.tensor(seq_lens), 0)),
        dtype=torch.int32,
        device="cuda",
    )
    scale = hd**-0.5
    out = ops.flash_prefill(q, k, v, cu, seq_lens, scale)
    ref = torch_ref.flash_prefill_ref(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), scale)
    _assert_close_bf16(out, ref, atol=3e-2, rtol=3e-2)


def test_silu_mul():
    gu = torch.randn(65, 2 * 14336, dtype=torch.bfloat16, device="cuda")
    out = ops.silu_mul(gu)
    ref = torch_ref.silu_mul_ref(gu.cpu())
    _assert_close_bf16(out, ref)


def test_sample_greedy():
    B, V = 16, 128256
    logits = torch.randn(B, V, dtype=torch.float32, device="cuda")
    temp = torch.zeros(B, device="cuda")
    tk = torch.zeros(B, dtype=torch.int32, device="cuda")
    tp = torch.ones(B, device="cuda")
    u = torch.rand(B, device="cuda")
    out = ops.sample(logits, temp, tk, tp, u)
    assert torch.equal(out.cpu(), logits.argmax(-1).to(torch.int32).cpu())


@pytest.mark.parametrize("k,p", [(0, 1.0), (50, 1.0), (0, 0.9), (40, 0.95)])
def test_sample_topk_topp(k, p):
    torch.manual_seed(99)
    B, V = 8, 32000
    logits = torch.randn(B, V, dtype=torch.float32, device="cuda") * 4
    temp = torch.full((B,), 0.8, device="cuda")
    tk = torch.full((B,), k, dtype=torch.int32, device="cuda")
    tp = torch.full((B,), p, device="cuda")
    u = torch.rand(B, device="cuda").clamp(0.01, 0.99)
    out = ops.sample(logits, temp, tk, tp, u)
    ref = torch_ref.sample_ref(
        logits.cpu(), temp.cpu(), tk.cpu(), tp.cpu(), u.cpu()
    )
    assert torch.equal(out.cpu(), ref)


def test_sample_bf16_greedy():
    B, V = 16, 128256
    logits = torch.randn(B, V, dtype=torch.bfloat16, device="cuda")
    temp = torch.zeros(B, device="cuda")
    tk = torch.zeros(B, dtype=torch.int32, device="cuda")
    tp = torch.ones(B, device="cuda")
    u = torch.rand(B, device="cuda")
    out = ops.sample(logits, temp, tk, tp, u)
    ref = logits.float().argmax(-1).to(torch.int32)
    assert torch.equal(out.cpu(), ref.cpu())


@pytest.mark.parametrize("k,p", [(0, 1.0), (50, 1.0), (0, 0.9), (40, 0.95),
                                 (1, 1.0)])
def test_sample_bf16_topk_topp(k, p):
    torch.manual_seed(123)
    B, V = 8, 32000
    logits = (torch.randn(B, V, device="cuda") * 4).to(torch.bfloat16)
    temp = torch.full((B,), 0.8, device="cuda")
    tk = torch.full((B,), k, dtype=torch.int32, device="cuda")
    tp = torch.full((B,), p, device="cuda")
    u = torch.rand(B, device="cuda").clamp(0.01, 0.99)
    out = ops.sample(logits, temp, tk, tp, u)
    ref = torch_ref.sample_ref(
        logits.cpu(), temp.cpu(), tk.cpu(), tp.cpu(), u.cpu()
    )
    assert torch.equal(out.cpu(), ref)


def test_sample_bf16_odd_vocab_tail():
    """V not divisible by 8 exercises the scalar tail paths."""
    torch.manual_seed(5)
    B, V = 4, 1003
    logits = (torch.randn(B, V, device="cuda") * 3).to(torch.bfloat16)
    temp = torch.full((B,), 1.0, device="cuda")
    tk = torch.full((B,), 25, dtype=torch.int32, device="cuda")
    tp = torch.full((B,), 0.9, device="cuda")
    u = torch.rand(B, device="cuda").clamp(0.01, 0.99)
    out = ops.sample(logits, temp, tk, tp, u)
    ref = torch_ref.sample_ref(
        logits.cpu(), temp.cpu(), tk.cpu(), tp.cpu(), u.cpu()
    )
    assert torch.equal(out.cpu(), ref)


def test_moe_route_matches_ref():
    torch.manual_seed(21)
    T, E, K, C = 256, 8, 2, 256
    logits = torch.randn(T, E, device="cuda")
    w, dst, idx, counts = ops.moe_route(logits, K, C)
    rw, rdst, ridx, rcounts = torch_ref.moe_route_ref(logits.cpu(), K, C)
    assert torch.equal(counts.cpu(), rcounts)
    assert torch.equal(idx.cpu(), ridx)
    assert torch.allclose(w.cpu(), rw, atol=1e-5)
    # dst validity (rank order is not part of the contract)
    d = dst.cpu()
    assert d.unique().numel() == T * K
    flat = ridx.reshape(-1).long()
    counts = torch.bincount(flat, minlength=E)
    for i in range(T * K):
        e = int(flat[i])
        r = int(d[i]) - e * C
        assert 0 <= r < int(counts[e])


def test_moe_grouped_gemm_matches_ref():
    torch.manual_seed(31)
    E, C, K, N = 4, 128, 256, 256
    a = torch.randn(E * C, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device="cuda") * 0.05
    counts = torch.tensor([128, 7, 0, 65], dtype=torch.int32, device="cuda")
    out = ops.moe_grouped_gemm(a, w, counts, C)
    # fp32 reference of the same op; RELATIVE error bound against the
    # fp32 GEMM scale (a flat atol=0.5 is wide
    # enough to hide a systematic accumulation bug at bf16 scale)
    for e in range(E):
        c = int(counts[e])
        if c == 0:
            continue      # expert with no routed rows (counts[2])
        got = out[e * C:e * C + c].float().cpu()
        want = (a[e * C:e * C + c].float().cpu()
                @ w[e].float().cpu().T)
        scale = want.abs().max().clamp_min(1.0)
        rel = (got - want).abs().max() / scale
        # bf16 inputs, fp32 accumulate over K=256: relative error vs the
        # fp32 reference is bounded by input rounding ~2^-8 * sqrt-ish
        # growth; 2e-2 relative catches any systematic 1-ulp bias
        assert rel < 2e-2, f"expert {e}: rel err {rel:.4f}"


def test_silu_mul_grouped_matches_ref():
    torch.manual_seed(32)
    E, C, inter = 4, 64, 512
    gu = torch.randn(E * C, 2 * inter, dtype=torch.bfloat16, device="cuda")
    counts = torch.tensor([64, 1, 0, 33], dtype=torch.int32, device="cuda")
    out = ops.silu_mul_grouped(gu, counts, C)
    ref = torch_ref.silu_mul_grouped_ref(gu.cpu(), counts.cpu(), C)
    for e in range(E):
        c = int(counts[e])
        assert torch.allclose(out[e * C:e * C + c].float().cpu(),
                              ref[e * C:e * C + c].float(), atol=2e-2)


def test_moe_padded_forward_gpu_matches_sorted():
    """Grouped-GEMM decode MoE vs the independent sorted-F.linear path."""
    from inference_gateway_amd.engine.config import MODEL_PRESETS
    from inference_gateway_amd.engine.models.mixtral import MixtralMoE

    torch.manual_seed(33)
    cfg = MODEL_PRESETS["tiny-moe"]
    moe = MixtralMoE(cfg, layer=0, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(37, cfg.hidden_size, dtype=torch.bfloat16,
                    device="cuda") * 0.5
    a = moe._forward_padded(x).float()
    b = moe._forward_sorted(x).float()
    assert torch.allclose(a, b, atol=5e-2, rtol=5e-2), \
        f"max diff {(a - b).abs().max()}"


def test_native_extension_is_loaded_and_in_tree():
    """On a GPU box the HIP extension must be the in-tree build — GPU
    tests passing on a silent eager fallback would be meaningless."""
    from pathlib import Path

    import inference_gateway_amd.ops as ops_pkg

    assert ops_pkg.HAS_KERNELS, "in-tree _kernels extension not loaded"
    so = Path(ops_pkg._kernels.__file__).resolve()
    pkg = Path(ops_pkg.__file__).resolve().parent
    assert so.parent == pkg, f"extension loaded from {so}, not in-tree"
    assert torch.version.hip, "expected a ROCm torch build"


@pytest.mark.parametrize("chunks", [
    [(0, 33)],                      # no prefix (plain prefill via cache)
    [(16, 20)],                     # block-aligned prefix
    [(21, 13), (0, 40), (37, 64)],  # unaligned prefixes, multi-seq
])
def test_flash_prefill_paged_vs_ref(chunks):
    """Paged-extend prefill kernel vs the CPU fp32-path reference:
    chunk queries attend over cached prefix + chunk K/V through the
    block table with absolute-position causality."""
    torch.manual_seed(0)
    NKV, NQ, HD, BS = 2, 8, 128, 16
    dev = "cuda:0"
    maxb = max((p + l + BS - 1) // BS for p, l in chunks)
    nb = 1 + sum((p + l + BS - 1) // BS for p, l in chunks)
    kcache = (torch.randn(nb, NKV, BS, HD, device=dev) / 3).bfloat16()
    vcache = (torch.randn(nb, NKV, BS, HD, device=dev) / 3).bfloat16()
    bt = torch.zeros(len(chunks), maxb, dtype=torch.int32, device=dev)
    nxt = 1
    cu, qs, plens, seq_lens = [0], [], [], []
    for i, (P, Lq) in enumerate(chunks):
        need = (P + Lq + BS - 1) // BS
        bt[i, :need] = torch.arange(nxt, nxt + need, dtype=torch.int32)
        nxt += need
        qs.append((torch.randn(Lq, NQ, HD, device=dev) / 3).bfloat16())
        cu.append(cu[-1] + Lq)
        plens.append(P)
        seq_lens.append(Lq)
    q = torch.cat(qs, 0)
    cu_t = torch.tensor(cu, dtype=torch.int32, device=dev)
    pl_t = torch.tensor(plens, dtype=torch.int32, device=dev)
    scale = HD ** -0.5
    got = ops.flash_prefill_paged(q, kcache, vcache, bt, pl_t, cu_t,
                                  seq_lens, scale)
    want = torch_ref.flash_prefill_paged_ref(
        q.cpu(), kcache.cpu(), vcache.cpu(), bt.cpu(), pl_t.cpu(),
        cu_t.cpu(), scale)
    torch.testing.assert_close(got.cpu().float(), want.float(),
                               atol=3e-2, rtol=3e-2)


def test_zz_chunked_prefill_engine_matches_unchunked():
    # placed in this (alphabetically last) file, and named to sort last,
    # so the driver's -x run covers everything else first
    from inference_gateway_amd.engine.config import (EngineConfig,
                                                     SamplingParams)
    from inference_gateway_amd.engine.engine import LLMEngine

    """Chunked prefill through the paged-extend kernel + hipGraph decode
    must equal the unchunked engine token-for-token."""
    long_prompt = list(range(3, 200))
    short = list(range(7, 30))

    def run(cfg):
        torch.manual_seed(0)
        eng = LLMEngine(cfg, device="cuda:0")
        rids = [eng.add_request(p, SamplingParams(max_new_tokens=6,
                                                  temperature=0.0))
                for p in (long_prompt, short)]
        steps = 0
        while not eng.is_finished():
            eng.step()
            steps += 1
            assert steps < 500
        return [eng.get_output(r).token_ids for r in rids]

    want = run(EngineConfig.tiny_llama(use_hip_graphs=True,
                                       max_batched_tokens=2048))
    got = run(EngineConfig.tiny_llama(use_hip_graphs=True,
                                      max_batched_tokens=48,
                                      enable_chunked_prefill=True))
    assert got == want


@pytest.mark.gpu
@pytest.mark.skipif(os.environ.get("IGW_ONESHOT_AR", "0") != "1",
                    reason="one-shot AR is opt-in (see parallel/oneshot"
                           ".py try_init: loopback flag-visibility "
                           "stalls; validate on a real 8-GPU node)")
def test_oneshot_allreduce_loopback():
    """One-shot all-reduce protocol on ONE GPU: 4 simulated ranks (4
    buffer pairs, 4 concurrent streams — ROCm's default 4 hardware
    queues dispatch at most 4 independent kernels concurrently, a
    loopback-only constraint) handshake through the real epoch-flag
    protocol and each produce the exact rank-ordered fp32 sum.  The
    8-GPU run differs only in the pointers coming from
    hipIpcOpenMemHandle (parallel/oneshot.py).  Pins the kernel math,
    per-block slice map, ragged tails, double-buffer parity and
    repeat-call epoch advance."""
    from inference_gateway_amd import ops

    k = ops._kernels
    world = 4
    max_bytes = 1 << 20
    datas = [k.ar_alloc(2 * max_bytes, False) for _ in range(world)]
    sigs = [k.ar_alloc(k.ar_sig_bytes(), True) for _ in range(world)]
    try:
        for trial, numel in enumerate([8192, 8191, 65536, 7, 524288]):
            gens = torch.Generator(device="cuda").manual_seed(100 + trial)
            ins = [torch.randn(numel, device="cuda", generator=gens,
                               dtype=torch.float32).to(torch.bfloat16)
                   for _ in range(world)]
            acc = torch.zeros(numel, device="cuda")
            for t in ins:
                acc += t.float()
            want = acc.to(torch.bfloat16)
            streams = [torch.cuda.Stream() for _ in range(world)]
            for r in range(world):
                with torch.cuda.stream(streams[r]):
                    k.oneshot_allreduce(ins[r], r, world, datas, sigs,
                                        max_bytes // 2)
            torch.cuda.synchronize()
            for r in range(world):
                assert k.ar_check_error(sigs[r]) == 0, (
                    trial, r, hex(k.ar_check_error(sigs[r])))
                assert torch.equal(ins[r], want), (
                    trial, r,
                    (ins[r].float() - want.float()).abs().max().item())
    finally:
        for p in datas + sigs:
            k.ar_free(p)


@pytest.mark.gpu
@pytest.mark.skipif(os.environ.get("IGW_ONESHOT_AR", "0") != "1",
                    reason="one-shot AR is opt-in (see above)")
def test_oneshot_allreduce_world8_subprocess():
    """Full 8-rank protocol in a subprocess with GPU_MAX_HW_QUEUES=8
    (the HIP runtime reads it at init, so it needs a fresh process).
    Exercises exactly the deployment world size."""
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "tools/debug_oneshot.py", "--world", "8"],
        cwd=root, capture_output=True, text=True, timeout=600,
        env=dict(os.environ, GPU_MAX_HW_QUEUES="8"))
    assert out.returncode == 0, out.stdout + out.stderr


def test_fused_gemm_tn_matches_fp32_ref():
    """Fused TN GEMM (plain + SwiGLU epilogue) vs the fp32 reference —
    relative bound, asymmetric operands (a symmetric B would pass a
    transposed C-write, guide §3)."""
    torch.manual_seed(41)
    M, N, K = 256, 128, 256
    x = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(2 * N, K, device="cuda") * 0.2).to(torch.bfloat16)
    w += torch.arange(2 * N, device="cuda").view(-1, 1).to(
        torch.bfloat16) * 1e-3           # asymmetry across rows
    # plain mode
    got = ops.fused_gemm_tn(x, w[:N], swiglu=False).float()
    want = x.float() @ w[:N].float().t()
    rel = (got - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel < 2e-2, f"plain rel err {rel:.4f}"
    # swiglu mode
    got = ops.fused_gemm_tn(x, w, swiglu=True).float()
    g = x.float() @ w[:N].float().t()
    u = x.float() @ w[N:].float().t()
    want = torch.nn.functional.silu(g) * u
    rel = (got - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel < 2e-2, f"swiglu rel err {rel:.4f}"
    # nt variant bitwise-matches the cached variant
    got_nt = ops.fused_gemm_tn(x, w, swiglu=True, nt_weights=True)
    assert torch.equal(got_nt.float(), got)
    # flagship decode shapes (down-proj and gate_up) at batch 256
    x2 = (torch.randn(256, 14336, device="cuda") * 0.1).to(torch.bfloat16)
    wd = (torch.randn(4096, 14336, device="cuda") * 0.05).to(
        torch.bfloat16)
    got = ops.fused_gemm_tn(x2, wd).float()
    want = x2.float() @ wd.float().t()
    rel = (got - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel < 2e-2, f"down-proj rel err {rel:.4f}"


def test_fused_gemm_tn2_matches_fp32_ref():
    """v2 (block-shared LDS W tiles, glds + XOR swizzle) vs fp32 —
    asymmetric operands; plain and SwiGLU modes; flagship shapes."""
    from inference_gateway_amd.ops import _kernels

    torch.manual_seed(43)
    M, N, K = 256, 128, 256
    x = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(2 * N, K, device="cuda") * 0.2).to(torch.bfloat16)
    w += torch.arange(2 * N, device="cuda").view(-1, 1).to(
        torch.bfloat16) * 1e-3
    out = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    _kernels.fused_gemm_tn2(out, x, w[:N], False, True)
    want = x.float() @ w[:N].float().t()
    rel = (out.float() - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel < 2e-2, f"v2 plain rel {rel:.4f}"
    _kernels.fused_gemm_tn2(out, x, w, True, True)
    g = x.float() @ w[:N].float().t()
    u = x.float() @ w[N:].float().t()
    want = torch.nn.functional.silu(g) * u
    rel = (out.float() - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel < 2e-2, f"v2 swiglu rel {rel:.4f}"
    x2 = (torch.randn(256, 14336, device="cuda") * 0.1).to(torch.bfloat16)
    wd = (torch.randn(4096, 14336, device="cuda") * 0.05).to(
        torch.bfloat16)
    out2 = torch.empty((256, 4096), dtype=torch.bfloat16, device="cuda")
    _kernels.fused_gemm_tn2(out2, x2, wd, False, True)
    want = x2.float() @ wd.float().t()
    rel = (out2.float() - want).abs().max() / want.abs().max().clamp_min(1.0)
    assert rel < 2e-2, f"v2 down rel {rel:.4f}"
