"""GPU engine tests: end-to-end generation on the HIP kernel path, and
hipGraph decode vs eager decode equivalence."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from inference_gateway_amd.engine.config import EngineConfig, SamplingParams
from inference_gateway_amd.engine.engine import LLMEngine


def _run(cfg, prompts, max_new=8, seed=0):
    torch.manual_seed(seed)
    eng = LLMEngine(cfg, device="cuda:0")
    rids = [
        eng.add_request(p, SamplingParams(max_new_tokens=max_new,
                                          temperature=0.0))
        for p in prompts
    ]
    steps = 0
    while not eng.is_finished():
        eng.step()
        steps += 1
        assert steps < 200
    return [eng.get_output(r).token_ids for r in rids]


def test_graph_decode_matches_eager():
    prompts = [list(range(1, 20)), list(range(30, 47)), list(range(100, 164))]
    eager = _run(EngineConfig.tiny_llama(enforce_eager=True), prompts)
    graph = _run(EngineConfig.tiny_llama(use_hip_graphs=True), prompts)
    assert eager == graph


def test_generation_deterministic_on_gpu():
    prompts = [list(range(1, 33))]
    a = _run(EngineConfig.tiny_llama(use_hip_graphs=True), prompts, seed=7)
    b = _run(EngineConfig.tiny_llama(use_hip_graphs=True), prompts, seed=7)
    assert a == b


def test_sampled_generation_with_topk_topp():
    torch.manual_seed(0)
    eng = LLMEngine(EngineConfig.tiny_llama(), device="cuda:0")
    rid = eng.add_request(
        list(range(1, 16)),
        SamplingParams(max_new_tokens=12, temperature=0.9, top_k=40,
                       top_p=0.9),
    )
    steps = 0
    while not eng.is_finished():
        eng.step()
        steps += 1
        assert steps < 100
    out = eng.get_output(rid)
    assert len(out.token_ids) == 12
    assert all(0 <= t < 512 for t in out.token_ids)


def test_moe_graph_decode_matches_eager():
    prompts = [list(range(1, 20)), list(range(30, 47))]
    eager = _run(EngineConfig.tiny_moe(enforce_eager=True), prompts)
    graph = _run(EngineConfig.tiny_moe(use_hip_graphs=True), prompts)
    assert eager == graph


def test_graph_decode_across_membership_change():
    """Sequences of different lengths finish at different steps; the graph
    runner must refresh its active-slot buffer and keep decoding right."""
    cfg = EngineConfig.tiny_llama(use_hip_graphs=True)
    torch.manual_seed(0)
    eng = LLMEngine(cfg, device="cuda:0")
    r1 = eng.add_request(list(range(1, 20)),
                         SamplingParams(max_new_tokens=3, temperature=0.0))
    r2 = eng.add_request(list(range(5, 40)),
                         SamplingParams(max_new_tokens=24, temperature=0.0))
    while not eng.is_finished():
        eng.step()
    assert len(eng.get_output(r1).token_ids) == 3
    assert len(eng.get_output(r2).token_ids) == 24

    # same prompts, eager: long sequence must match token-for-token
    torch.manual_seed(0)
    eng2 = LLMEngine(EngineConfig.tiny_llama(enforce_eager=True),
                     device="cuda:0")
    r1b = eng2.add_request(list(range(1, 20)),
                           SamplingParams(max_new_tokens=3, temperature=0.0))
    r2b = eng2.add_request(list(range(5, 40)),
                           SamplingParams(max_new_tokens=24, temperature=0.0))
    while not eng2.is_finished():
        eng2.step()
    assert eng.get_output(r1).token_ids == eng2.get_output(r1b).token_ids
    assert eng.get_output(r2).token_ids == eng2.get_output(r2b).token_ids


def test_preemption_with_graphs_matches_unconstrained():
    """Force KV-pool preemption mid-decode under the async hipGraph
    pipeline: the preempted sequence must resume transparently and
    produce the same greedy tokens as an unconstrained run."""
    prompts = [list(range(1, 40)), list(range(50, 90)),
               list(range(100, 140))]
    torch.manual_seed(0)
    eng = LLMEngine(EngineConfig.tiny_llama(use_hip_graphs=True,
                                            num_gpu_blocks=14,
                                            max_model_len=128),
                    device="cuda:0")
    rids = [eng.add_request(p, SamplingParams(max_new_tokens=24,
                                              temperature=0.0))
            for p in prompts]
    steps = 0
    while not eng.is_finished():
        eng.step()
        steps += 1
        assert steps < 600
    constrained = [eng.get_output(r).token_ids for r in rids]
    for out in constrained:
        assert len(out) == 24

    torch.manual_seed(0)
    eng2 = LLMEngine(EngineConfig.tiny_llama(use_hip_graphs=True,
                                             num_gpu_blocks=256,
                                             max_model_len=128),
                     device="cuda:0")
    rids2 = [eng2.add_request(p, SamplingParams(max_new_tokens=24,
                                                temperature=0.0))
             for p in prompts]
    while not eng2.is_finished():
        eng2.step()
    assert constrained == [eng2.get_output(r).token_ids for r in rids2]


def test_abort_mid_decode_with_graphs():
    torch.manual_seed(0)
    eng = LLMEngine(EngineConfig.tiny_llama(use_hip_graphs=True), "cuda:0")
    r1 = eng.add_request(list(range(1, 30)),
                         SamplingParams(max_new_tokens=64, temperature=0.0))
    r2 = eng.add_request(list(range(40, 70)),
                         SamplingParams(max_new_tokens=16, temperature=0.0))
    for _ in range(6):
        eng.step()
    eng.abort_request(r1)
    steps = 0
    while not eng.is_finished():
        eng.step()
        steps += 1
        assert steps < 200
    assert eng.get_output(r1).finish_reason == "abort"
    out2 = eng.get_output(r2)
    assert out2.finished and len(out2.token_ids) == 16


def _run_lp(cfg, prompts, k=3, max_new=6, seed=0):
    torch.manual_seed(seed)
    eng = LLMEngine(cfg, device="cuda:0")
    rids = [eng.add_request(
        p, SamplingParams(max_new_tokens=max_new, temperature=0.0,
                          logprobs=k)) for p in prompts]
    outs, steps = [], 0
    while not eng.is_finished():
        outs.extend(eng.step())
        steps += 1
        assert steps < 200
    by_seq = {}
    for o in outs:
        by_seq.setdefault(o.seq_id, []).append(
            (o.token_id, o.logprob, o.top_logprobs))
    return [by_seq[r] for r in rids]


def test_logprobs_graph_matches_eager():
    """The graph runner's post-replay logprob extraction must agree with
    the eager path (same weights, same greedy tokens, same logits)."""
    prompts = [list(range(1, 20)), list(range(30, 47))]
    eager = _run_lp(EngineConfig.tiny_llama(enforce_eager=True), prompts)
    graph = _run_lp(EngineConfig.tiny_llama(use_hip_graphs=True), prompts)
    for es, gs in zip(eager, graph):
        assert len(es) == len(gs) == 6
        for (et, elp, etop), (gt, glp, gtop) in zip(es, gs):
            assert et == gt
            assert abs(elp - glp) < 2e-3
            assert [i for i, _ in etop] == [i for i, _ in gtop]
            for (_, ev), (_, gv) in zip(etop, gtop):
                assert abs(ev - gv) < 2e-3
            # greedy: sampled token is top-1
            assert etop[0][0] == et and glp <= 0.0


def test_seeded_sampling_graph_matches_eager():
    """OpenAI seed on the graph path: same seeded request must produce
    identical tokens under hipGraph decode and eager decode (stochastic
    temperature, uniforms from the per-request seed stream)."""
    def run(cfg):
        torch.manual_seed(0)
        eng = LLMEngine(cfg, device="cuda:0")
        rid = eng.add_request(
            list(range(1, 20)),
            SamplingParams(max_new_tokens=8, temperature=1.0, seed=99))
        steps = 0
        while not eng.is_finished():
            eng.step()
            steps += 1
            assert steps < 200
        return eng.get_output(rid).token_ids

    eager = run(EngineConfig.tiny_llama(enforce_eager=True))
    graph = run(EngineConfig.tiny_llama(use_hip_graphs=True))
    assert eager == graph and len(eager) == 8


def test_frequency_penalty_on_graph_path():
    """Graph path: the async pipeline's host view lags one token, so the
    device-side last-token add must close the gap — a huge frequency
    penalty still forbids ALL repeats, including immediate ones."""
    torch.manual_seed(0)
    eng = LLMEngine(EngineConfig.tiny_llama(use_hip_graphs=True),
                    device="cuda:0")
    rid = eng.add_request(
        list(range(1, 17)),
        SamplingParams(max_new_tokens=12, temperature=0.0,
                       frequency_penalty=100.0))
    steps = 0
    while not eng.is_finished():
        eng.step()
        steps += 1
        assert steps < 200
    toks = eng.get_output(rid).token_ids
    assert len(toks) == 12 and len(set(toks)) == 12


def test_fork_group_on_graph_path():
    """n>1 fork through hipGraph decode: forked branches (shared prompt
    blocks + copied tail, slot state seeded at fork) must equal an
    independent greedy request."""
    torch.manual_seed(0)
    prompt = list(range(1, 27))
    eng = LLMEngine(EngineConfig.tiny_llama(use_hip_graphs=True),
                    device="cuda:0")
    ref = eng.add_request(prompt, SamplingParams(max_new_tokens=8,
                                                 temperature=0.0))
    rids = eng.add_request_group(
        prompt, SamplingParams(max_new_tokens=8, temperature=0.0), n=3)
    steps = 0
    while not eng.is_finished():
        eng.step()
        steps += 1
        assert steps < 300
    want = eng.get_output(ref).token_ids
    assert len(want) == 8
    for r in rids:
        assert eng.get_output(r).token_ids == want
    assert eng.block_manager.num_free == eng.block_manager.num_blocks - 1


def test_prefix_caching_on_gpu_matches_uncached():
    """Prefix caching on the HIP path (paged-extend prefill kernel +
    shared blocks + graph decode): cached repeats and prefix-shared
    prompts must match a cache-less engine token-for-token."""
    torch.manual_seed(0)
    base = list(range(1, 27))
    longer = base + list(range(100, 140))

    def run(cfg, prompts):
        torch.manual_seed(0)
        eng = LLMEngine(cfg, device="cuda:0")
        rids = [eng.add_request(p, SamplingParams(max_new_tokens=6,
                                                  temperature=0.0))
                for p in prompts]
        steps = 0
        while not eng.is_finished():
            eng.step()
            steps += 1
            assert steps < 300
        return eng, [eng.get_output(r).token_ids for r in rids]

    _, want = run(EngineConfig.tiny_llama(use_hip_graphs=True),
                  [base, longer])
    eng, got1 = run(EngineConfig.tiny_llama(use_hip_graphs=True,
                                            enable_prefix_caching=True),
                    [base, longer])
    assert got1 == want
    # second wave on the SAME engine: now the pool has the blocks
    r2 = eng.add_request(base, SamplingParams(max_new_tokens=6,
                                              temperature=0.0))
    r3 = eng.add_request(longer, SamplingParams(max_new_tokens=6,
                                                temperature=0.0))
    while not eng.is_finished():
        eng.step()
    assert eng.get_output(r2).token_ids == want[0]
    assert eng.get_output(r3).token_ids == want[1]
    assert eng.block_manager.prefix_hit_tokens >= 48



def test_zz_mixed_batches_gpu_token_equivalence():
    """Same-step mixed prefill+decode rounds on the HIP kernel path
    (paged decode kernel for the decode rows + flash/paged-extend for
    the chunk, sharing one forward) must be token-exact vs the plain
    engine with graphs on."""
    def run(mixed):
        torch.manual_seed(0)
        cfg = EngineConfig.tiny_llama(
            use_hip_graphs=True, enable_chunked_prefill=True,
            enable_mixed_batches=mixed, max_batched_tokens=48,
            num_gpu_blocks=256, max_num_seqs=8)
        eng = LLMEngine(cfg, device="cuda:0")
        sp = lambda: SamplingParams(max_new_tokens=10,  # noqa
                                    temperature=0.0)
        ra = eng.add_request(list(range(1, 40)), sp())
        for _ in range(3):
            eng.step()
        rb = eng.add_request(list(range(50, 170)), sp())
        rc = eng.add_request(list(range(7, 30)), sp())
        while not eng.is_finished():
            eng.step()
        return ([eng.get_output(r).token_ids for r in (ra, rb, rc)],
                eng.scheduler.num_mixed_rounds)

    plain, m0 = run(False)
    mixed, m1 = run(True)
    assert m0 == 0 and m1 > 0
    assert mixed == plain
