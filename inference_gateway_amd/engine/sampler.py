"""Batched sampling front-end over the fused gfx950 sampling kernel."""
from __future__ import annotations

import random as _random
from typing import List

import torch

from .. import ops
from .scheduler import Sequence


def apply_seeded_uniforms(u: torch.Tensor, seqs) -> None:
    """OpenAI `seed`: overwrite the uniform draw of every row whose
    sequence carries a seed with a value derived ONLY from (seed, tick) —
    independent of batch composition, bucket size, and the other
    requests, so the same seeded request replays the same tokens.  Used
    by both the eager Sampler and the graph runner; rows without a seed
    keep the engine-global draw and pay nothing."""
    rows, vals = [], []
    for i, s in enumerate(seqs):
        sd = s.sampling.seed
        if sd is None:
            continue
        rows.append(i)
        vals.append(min(max(
            _random.Random((int(sd) << 24) ^ s.sample_tick).random(),
            1e-6), 1 - 1e-6))
        s.sample_tick += 1
    if rows:
        idx = torch.tensor(rows, dtype=torch.int64, device=u.device)
        u.scatter_(0, idx, torch.tensor(vals, dtype=u.dtype,
                                        device=u.device))


class Sampler:
    def __init__(self, device, seed: int = 0):
        self.device = torch.device(device)
        gen_device = self.device if self.device.type == "cuda" else "cpu"
        self.generator = torch.Generator(device=gen_device)
        self.generator.manual_seed(seed)

    def sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> List[int]:
        """logits: [B, vocab] fp32 on self.device -> python token ids."""
        B = logits.shape[0]
        dev = logits.device
        temps = torch.tensor(
            [s.sampling.temperature for s in seqs], dtype=torch.float32,
            device=dev)
        top_k = torch.tensor(
            [s.sampling.top_k for s in seqs], dtype=torch.int32, device=dev)
        top_p = torch.tensor(
            [s.sampling.top_p for s in seqs], dtype=torch.float32, device=dev)
        gdev = self.generator.device
        uniforms = torch.rand(
            B, generator=self.generator, device=gdev).to(dev).clamp(1e-6,
                                                                    1 - 1e-6)
        apply_seeded_uniforms(uniforms, seqs)
        out = ops.sample(logits, temps, top_k, top_p, uniforms)
        return out.cpu().tolist()
