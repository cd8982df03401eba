#!/usr/bin/env python3
"""Diagnose the one-shot all-reduce handshake on one GPU."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from inference_gateway_amd import ops  # noqa: E402

k = ops._kernels
MAXB = 1 << 20


import struct


def dump_sig(ptr, world, blocks=2):
    raw = k.ar_dump(ptr, k.ar_sig_bytes())
    words = struct.unpack(f"<{len(raw)//4}I", raw)
    # layout: flags[64][8], epochs[64], error
    out = []
    for b in range(blocks):
        out.append(list(words[b * 8: b * 8 + world]))
    epochs = list(words[64 * 8: 64 * 8 + blocks])
    err = words[64 * 8 + 64]
    return {"flags": out, "epochs": epochs, "error": hex(err)}


def run(world, numels=(4096,), repeat=2):
    datas = [k.ar_alloc(2 * MAXB, False) for _ in range(world)]
    sigs = [k.ar_alloc(k.ar_sig_bytes(), True) for _ in range(world)]
    ok = True
    try:
      for numel in numels:
        for it in range(repeat):
            ins = [torch.randn(numel, device="cuda").to(torch.bfloat16)
                   for _ in range(world)]
            acc = torch.zeros(numel, device="cuda")
            for t in ins:
                acc += t.float()
            want = acc.to(torch.bfloat16)
            if not hasattr(run, "_streams") or len(run._streams) < world:
                run._streams = [torch.cuda.Stream() for _ in range(world)]
            streams = run._streams[:world]
            for r in range(world):
                with torch.cuda.stream(streams[r]):
                    k.oneshot_allreduce(ins[r], r, world, datas, sigs,
                                        MAXB // 2)
            torch.cuda.synchronize()
            errs = [hex(k.ar_check_error(s)) for s in sigs]
            match = [bool(torch.equal(ins[r], want))
                     for r in range(world)]
            bad = [int((ins[r].float() - want.float()).abs().gt(0).sum())
                   for r in range(world)]
            first_bad = None
            for r in range(world):
                if not match[r]:
                    idx = (ins[r].float()
                           - want.float()).abs().gt(0).nonzero()
                    first_bad = (r, idx[:4].flatten().tolist(),
                                 int(idx.numel()))
                    break
            print(f"world={world} numel={numel} iter={it} errors={errs} "
                  f"match={match} nbad={bad} first={first_bad}",
                  flush=True)
            ok = ok and not any(int(e, 16) for e in errs) and all(match)
    finally:
        for p in datas + sigs:
            k.ar_free(p)
    return ok


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=0,
                    help="run ONE world size and exit 0/1 (for the "
                         "subprocess GPU test; world=8 needs "
                         "GPU_MAX_HW_QUEUES=8 so the 8 simulated ranks' "
                         "kernels dispatch concurrently — a loopback "
                         "artifact only: real TP is one process per GPU)")
    a = ap.parse_args()
    numels = (4096, 8192, 8191, 65536, 7, 524288)
    if a.world:
        sys.exit(0 if run(a.world, numels) else 1)
    print("world=1:", run(1, numels))
    print("world=2:", run(2, numels))
    print("world=4:", run(4, numels))
    print("world=8:", run(8, numels))
