"""Batched sampling front-end over the fused gfx950 sampling kernel
(net-new engine scope, SURVEY.md §7 stage 3e; OpenAI sampling-parameter
semantics — seed/penalties/logit_bias — implemented for real here where
the reference only forwards them upstream)."""
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


def apply_penalties(logits: torch.Tensor, seqs,
                    last_tokens: torch.Tensor = None) -> None:
    """OpenAI presence/frequency penalties, in place on the penalized
    rows only (zero cost when nobody asks).  Host-side token counts
    drive the bulk of the penalty; on the async graph path the host's
    view lags the device by exactly one token, so the caller passes
    `last_tokens` (the device-resident last sampled token per row) and
    that token's penalty is applied device-side — immediate repeats are
    always penalized.  A token that repeats across both views gets its
    presence term counted twice (documented best-effort; the frequency
    term stays exact)."""
    rows = [i for i, s in enumerate(seqs)
            if s.sampling.presence_penalty or s.sampling.frequency_penalty
            or s.sampling.logit_bias]
    if not rows:
        return
    dev, dt = logits.device, logits.dtype
    for i in rows:
        sp = seqs[i].sampling
        if sp.logit_bias:
            # OpenAI logit_bias: plain additive, every step
            ids = torch.tensor(list(sp.logit_bias.keys()),
                               dtype=torch.int64, device=dev)
            vals = torch.tensor(list(sp.logit_bias.values()),
                                dtype=torch.float32, device=dev)
            logits[i, ids] += vals.to(dt)
        if not (sp.presence_penalty or sp.frequency_penalty):
            continue
        out = seqs[i].output_token_ids
        if out:
            counts: dict = {}
            for t in out:
                counts[t] = counts.get(t, 0) + 1
            ids = torch.tensor(list(counts), dtype=torch.int64, device=dev)
            cnt = torch.tensor(list(counts.values()), dtype=torch.float32,
                               device=dev)
            logits[i, ids] -= (sp.presence_penalty
                               + sp.frequency_penalty * cnt).to(dt)
    if last_tokens is not None:
        # only rows whose last token is still undrained (in_flight) — at
        # the first decode step after prefill the host view IS current
        # and a device-side add would double-count
        drows = [i for i in rows if seqs[i].in_flight
                 and (seqs[i].sampling.presence_penalty
                      or seqs[i].sampling.frequency_penalty)]
        if drows:
            rows_t = torch.tensor(drows, dtype=torch.int64, device=dev)
            pen = torch.tensor(
                [seqs[i].sampling.presence_penalty
                 + seqs[i].sampling.frequency_penalty for i in drows],
                dtype=torch.float32, device=dev)
            logits[rows_t, last_tokens.index_select(0, rows_t)] -= \
                pen.to(dt)


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
