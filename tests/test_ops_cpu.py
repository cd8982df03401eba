"""CPU sanity tests for the fp32 torch reference ops (the oracle used by the
GPU numerics tier). These run in the no-GPU harness."""
import torch

from inference_gateway_amd.ops import torch_ref


def test_rope_preserves_norm_and_pos0_identity():
    torch.manual_seed(0)
    hd = 128
    cs = torch_ref.build_cos_sin_cache(hd, 64)
    q = torch.randn(4, 2 * hd, dtype=torch.bfloat16)
    k = torch.randn(4, 1 * hd, dtype=torch.bfloat16)
    pos = torch.tensor([0, 1, 5, 63], dtype=torch.int32)
    qr, kr = torch_ref.rope_ref(pos, q.clone(), k.clone(), hd, cs)
    # position 0 is identity
    torch.testing.assert_close(qr[0], q[0], atol=1e-2, rtol=1e-2)
    # rotation preserves per-pair norm => whole-vector norm
    torch.testing.assert_close(
        qr.float().norm(dim=-1), q.float().norm(dim=-1), atol=0.5, rtol=0.05
    )


def test_decode_ref_matches_prefill_ref_last_token():
    torch.manual_seed(1)
    L, nq, nkv, hd, bs = 45, 4, 2, 128, 16
    q = torch.randn(L, nq, hd, dtype=torch.bfloat16)
    k = torch.randn(L, nkv, hd, dtype=torch.bfloat16)
    v = torch.randn(L, nkv, hd, dtype=torch.bfloat16)
    cu = torch.tensor([0, L], dtype=torch.int32)
    scale = hd**-0.5
    full = torch_ref.flash_prefill_ref(q, k, v, cu, scale)

    nblocks = (L + bs - 1) // bs
    kc = torch.zeros(nblocks + 2, nkv, bs, hd, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    slots = torch.arange(L, dtype=torch.int32) + bs  # blocks 1..n
    torch_ref.append_kv_cache_ref(k, v, kc, vc, slots)
    bt = (torch.arange(nblocks, dtype=torch.int32) + 1).unsqueeze(0)
    ctx = torch.tensor([L], dtype=torch.int32)
    dec = torch_ref.paged_attn_decode_ref(q[-1:].contiguous(), kc, vc, bt, ctx, scale)
    torch.testing.assert_close(
        dec[0].float(), full[-1].float(), atol=2e-2, rtol=2e-2
    )


def test_sample_ref_semantics():
    torch.manual_seed(2)
    V = 1000
    logits = torch.randn(3, V) * 3
    # greedy
    out = torch_ref.sample_ref(
        logits,
        torch.zeros(3),
        torch.zeros(3, dtype=torch.int32),
        torch.ones(3),
        torch.full((3,), 0.5),
    )
    assert torch.equal(out, logits.argmax(-1).to(torch.int32))
    # top-k=1 equals greedy regardless of u
    out1 = torch_ref.sample_ref(
        logits,
        torch.ones(3),
        torch.ones(3, dtype=torch.int32),
        torch.ones(3),
        torch.full((3,), 0.7),
    )
    assert torch.equal(out1, logits.argmax(-1).to(torch.int32))
    # top-k keeps only the k best: sampled token must be among top-40
    tk = torch.full((3,), 40, dtype=torch.int32)
    for trial in range(5):
        u = torch.rand(3).clamp(0.01, 0.99)
        out = torch_ref.sample_ref(logits, torch.ones(3), tk, torch.ones(3), u)
        topk = logits.topk(40, dim=-1).indices
        for i in range(3):
            assert out[i] in topk[i]


def test_fused_add_rmsnorm_ref_consistency():
    torch.manual_seed(3)
    x = torch.randn(5, 256, dtype=torch.bfloat16)
    res = torch.randn(5, 256, dtype=torch.bfloat16)
    w = torch.randn(256, dtype=torch.bfloat16)
    nx, nres = torch_ref.fused_add_rmsnorm_ref(x, res, w, 1e-5)
    torch.testing.assert_close(
        nres.float(), (x.float() + res.float()), atol=2e-2, rtol=2e-2
    )
    torch.testing.assert_close(
        nx.float(),
        torch_ref.rmsnorm_ref(nres, w, 1e-5).float(),
        atol=1e-3,
        rtol=1e-3,
    )


def test_make_prefill_tiles():
    ts, to = torch_ref.make_prefill_tiles([64, 1, 130])
    assert ts.tolist() == [0, 1, 2, 2, 2]
    assert to.tolist() == [0, 0, 0, 64, 128]


def test_flash_prefill_paged_ref_matches_dense_ref():
    """The paged-extend reference must agree with the dense causal
    reference: attention of the LAST Lq rows of a sequence over the
    whole sequence == paged-extend of that chunk with prefix P."""
    torch.manual_seed(0)
    NKV, NQ, HD, BS = 2, 4, 128, 16
    for P, Lq in ((0, 33), (16, 20), (21, 13)):
        KVL = P + Lq
        nblk = (KVL + BS - 1) // BS
        kcache = torch.randn(nblk + 2, NKV, BS, HD).bfloat16()
        vcache = torch.randn(nblk + 2, NKV, BS, HD).bfloat16()
        # blocks 1..nblk hold the sequence (block 0 reserved)
        bt = torch.arange(1, nblk + 1, dtype=torch.int32).unsqueeze(0)
        # dense copy of the full sequence K/V out of the cache
        keys = kcache[bt[0].long()].permute(0, 2, 1, 3).reshape(
            nblk * BS, NKV, HD)[:KVL]
        vals = vcache[bt[0].long()].permute(0, 2, 1, 3).reshape(
            nblk * BS, NKV, HD)[:KVL]
        q_full = torch.randn(KVL, NQ, HD).bfloat16()
        scale = HD ** -0.5
        cu_full = torch.tensor([0, KVL], dtype=torch.int32)
        dense = torch_ref.flash_prefill_ref(q_full, keys, vals, cu_full,
                                            scale)
        cu_q = torch.tensor([0, Lq], dtype=torch.int32)
        paged = torch_ref.flash_prefill_paged_ref(
            q_full[P:], kcache, vcache, bt,
            torch.tensor([P], dtype=torch.int32), cu_q, scale)
        torch.testing.assert_close(paged.float(), dense[P:].float(),
                                   atol=2e-2, rtol=2e-2)
