"""Batched sampling front-end over the fused gfx950 sampling kernel."""
from __future__ import annotations

from typing import List

import torch

from .. import ops
from .scheduler import Sequence


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
        out = ops.sample(logits, temps, top_k, top_p, uniforms)
        return out.cpu().tolist()
