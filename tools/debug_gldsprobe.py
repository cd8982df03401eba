import sys
sys.path.insert(0, ".")
import torch
from inference_gateway_amd.ops import _kernels as k

inp = torch.arange(64 * 64, dtype=torch.float32, device="cuda").to(torch.bfloat16).contiguous()
for swz in (0, 1):
    out = torch.zeros_like(inp)
    k.glds_probe(out, inp, swz)
    torch.cuda.synchronize()
    ok = torch.equal(out, inp)
    if not ok:
        bad = (out != inp).nonzero().flatten()
        print(f"probe swz={swz}: FAIL nbad={bad.numel()} first={bad[:8].tolist()}")
        print("  got ", out[:12].float().tolist())
        print("  want", inp[:12].float().tolist())
        i0 = int(bad[0])
        print(f"  at {i0}: got {float(out[i0])} want {float(inp[i0])}")
    else:
        print(f"probe swz={swz}: OK")

# v2 numerics with swizzle OFF and ON (tiny shape)
torch.manual_seed(43)
M, N, K = 256, 128, 256
x = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
w = (torch.randn(N, K, device="cuda") * 0.2).to(torch.bfloat16)
want = x.float() @ w.float().t()
for swz in (False, True):
    out = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    k.fused_gemm_tn2(out, x, w, False, swz)
    torch.cuda.synchronize()
    rel = (out.float() - want).abs().max() / want.abs().max()
    print(f"v2 plain swz={swz}: rel={float(rel):.5f}")
