"""The `local` provider: the in-process MI355X inference engine exposed
through the same provider interface as the 15 remote providers.

This is the architectural core of the rebuild: where the
reference proxies every request to a remote HTTP API through a self-proxy
hop, the local provider calls the engine directly — zero HTTP hops, tokens
stream from the hipGraph decode loop straight into the SSE writer.
"""
from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import AsyncIterator, Optional

from ..engine.async_engine import AsyncEngine
from ..engine.config import EngineConfig, MODEL_PRESETS, SamplingParams
from ..utils.tokenizer import apply_chat_template, get_tokenizer
from .constants import LOCAL_PROVIDER_ID
from .core import HTTPError


class DirectStream:
    """Direct-write SSE session for one local streaming completion
    (n=1).  The engine's delivery callback calls feed()/flush() on the
    event loop and the session writes fully framed SSE bytes straight to
    the socket — no per-token asyncio task wakeup, no generator
    resumption, no per-event ASGI message.  All of a step's ready tokens
    leave in ONE socket write.  Mirrors stream_chat_completions token
    semantics exactly (UTF-8 holdback, stop-string holdback, logprobs,
    usage, one [DONE]); equivalence is tested in tests/test_hserver.py."""

    def __init__(self, provider: "LocalProvider", req: dict, prompt_ids,
                 sampling, raw_write, raw_end, raw_backlogged=None):
        self.provider = provider
        self.engine = provider.engine
        self.tokenizer = provider.tokenizer
        self.raw_write = raw_write
        self.raw_end = raw_end
        self.raw_backlogged = raw_backlogged
        self.prompt_ids = prompt_ids
        self.sampling = sampling
        self.stops = provider._stop_strings(req)
        self.holdback = max((len(s) for s in self.stops), default=1) - 1
        self.want_lp = sampling.logprobs >= 0
        cid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        self._prefix = ("data: " + json.dumps({
            "id": cid, "object": "chat.completion.chunk",
            "created": int(time.time()),
            "model": f"{provider.id}/{provider.model_name}",
            "system_fingerprint": provider.fingerprint,
        }, separators=(",", ":"))[:-1] + ',"choices":[{"index":0,"delta":')
        self.pending: list = []
        self.pending_lp: list = []
        self.emitted_tail = ""
        self.ntok = 0
        self.finish_reason = "stop"
        self._finished = False      # engine delivered the final token
        self._closed = False        # stream over (done or socket gone)
        self._first_content = True
        self.done = asyncio.Event()
        self.rid: Optional[int] = None
        # telemetry hooks (set by the telemetry middleware)
        self.on_first_chunk = None
        self.on_end = None

    # ---- frame building -------------------------------------------------
    def _chunk(self, delta: dict, finish=None, usage=None, lp=None) -> str:
        if len(delta) == 1 and "content" in delta:
            d = '{"content":' + json.dumps(delta["content"]) + "}"
        else:
            d = json.dumps(delta, separators=(",", ":"))
        tail = (',"usage":' + json.dumps(usage, separators=(",", ":"))
                if usage is not None else "")
        fin = f'"{finish}"' if finish else "null"
        lps = (',"logprobs":' + json.dumps({"content": lp},
                                           separators=(",", ":"))
               if lp is not None else "")
        return (self._prefix + d + lps + ',"finish_reason":' + fin
                + "}]" + tail + "}")

    def _write(self, s: str) -> bool:
        if not self.raw_write((s + "\n\n").encode()):
            self._close()
            return False
        return True

    def _emit_content(self, text: str, lp) -> bool:
        if self._first_content:
            self._first_content = False
            if self.on_first_chunk is not None:
                self.on_first_chunk()
        return self._write(self._chunk({"content": text}, lp=lp))

    # ---- engine-facing sink API (called on the event loop) -------------
    def feed(self, out) -> None:
        if self._closed:
            return
        self.ntok += 1
        self.pending.append(out.token_id)
        if self.want_lp:
            self.pending_lp.append(self.provider._lp_item(out))
        if out.finished:
            self.finish_reason = ("stop" if out.finish_reason == "stop"
                                  else "length")
            self._finished = True

    def flush(self) -> None:
        if self._closed:
            return
        if (not self._finished and self.raw_backlogged is not None
                and self.raw_backlogged()):
            # socket send is behind: skip this flush and let tokens
            # coalesce — the next delivery writes one merged chunk
            return
        try:
            self._flush_inner()
        except Exception:
            self._close()

    def _flush_inner(self) -> None:
        if self.pending:
            text = self.tokenizer.decode(self.pending)
            if not text.endswith("�"):
                nflush = len(self.pending)
                if self.stops:
                    i, buf = self.provider._scan_stop_progressive(
                        self.emitted_tail, self.pending, self.stops)
                self.pending = []
                lp = self.pending_lp if self.want_lp else None
                self.pending_lp = []
                if self.stops:
                    if i >= 0:
                        if buf[:i] and not self._emit_content(buf[:i], lp):
                            return
                        self.provider.streamed_tokens += nflush
                        self.finish_reason = "stop"
                        self.emitted_tail = ""
                        self._end_stream()
                        return
                    emit = buf[:-self.holdback] if self.holdback else buf
                    self.emitted_tail = buf[len(emit):]
                    if emit and not self._emit_content(emit, lp):
                        return
                    self.provider.streamed_tokens += nflush
                else:
                    if not self._emit_content(text, lp):
                        return
                    self.provider.streamed_tokens += nflush
        if self._finished:
            self._end_stream()

    def _end_stream(self) -> None:
        # final leftover: held-back stop tail + incomplete-UTF-8 pending
        nflush = len(self.pending)
        i, leftover = self.provider._scan_stop_progressive(
            self.emitted_tail, self.pending, self.stops)
        self.pending = []
        lp = (self.pending_lp if self.want_lp and self.pending_lp
              else None)
        self.pending_lp = []
        if i >= 0:
            leftover = leftover[:i]
            self.finish_reason = "stop"
        if leftover and not self._emit_content(leftover, lp):
            return
        self.provider.streamed_tokens += nflush
        usage = {
            "prompt_tokens": len(self.prompt_ids),
            "completion_tokens": self.ntok,
            "total_tokens": len(self.prompt_ids) + self.ntok,
        }
        if not self._write(self._chunk({}, finish=self.finish_reason,
                                       usage=usage)):
            return
        if not self._write("data: [DONE]"):
            return
        self.raw_end()
        self._complete(usage)

    def fail(self, exc) -> None:
        """Engine death (async_engine._close_all_queues): emit an
        in-stream error frame so the crash is not a clean 'stop'."""
        if self._closed:
            return
        msg = f"engine thread died: {exc!r}" if exc else "engine shutdown"
        self._write("data: " + json.dumps(
            {"error": {"message": msg, "type": "server_error"}}))
        self._write("data: [DONE]")
        self.raw_end()
        self._close(error=msg)

    # ---- lifecycle ------------------------------------------------------
    def _complete(self, usage: dict) -> None:
        self._closed = True
        if self.on_end is not None:
            self.on_end(usage["prompt_tokens"],
                        usage["completion_tokens"], "")
        self.done.set()

    def _close(self, error: str = "client disconnected") -> None:
        self._closed = True
        if self.on_end is not None:
            self.on_end(len(self.prompt_ids), self.ntok, error)
            self.on_end = None
        self.done.set()

    async def run(self) -> None:
        """Awaited by send_response: registers with the engine, writes
        the role chunk, then sleeps until the session completes (the
        engine delivery callbacks do all the writing)."""
        if not self._write(self._chunk({"role": "assistant",
                                        "content": ""})):
            return
        self.rid = self.engine.add_direct(self.prompt_ids, self.sampling,
                                          self)
        try:
            await self.done.wait()
        finally:
            self.engine.close_direct(self.rid)


class LocalProvider:
    def __init__(self, local_cfg, device: str = "cuda:0",
                 engine: Optional[AsyncEngine] = None):
        self.id = LOCAL_PROVIDER_ID
        self.model_name = local_cfg.served_model_name or local_cfg.model
        weights = getattr(local_cfg, "weights_path", "") or None
        if local_cfg.model == "hf":
            # architecture straight from the checkpoint's config.json
            from ..engine.weights import model_config_from_hf

            preset = model_config_from_hf(weights)
        elif local_cfg.model == "gguf":
            # architecture from GGUF metadata (llama.cpp checkpoints)
            from ..engine.gguf import model_config_from_gguf

            preset = model_config_from_gguf(weights)
        else:
            preset = MODEL_PRESETS[local_cfg.model]
        if (not local_cfg.served_model_name
                and local_cfg.model in ("hf", "gguf")):
            # surface the checkpoint's own name in /v1/models and
            # response `model` fields instead of the literal "hf"/"gguf"
            self.model_name = preset.name
        ecfg = EngineConfig(
            model=preset,
            max_num_seqs=local_cfg.max_num_seqs,
            max_model_len=local_cfg.max_model_len,
            gpu_memory_utilization=local_cfg.gpu_memory_utilization,
            enforce_eager=local_cfg.enforce_eager,
            num_gpu_blocks=getattr(local_cfg, "num_gpu_blocks", None),
            enable_prefix_caching=getattr(local_cfg, "prefix_caching",
                                          False),
            enable_chunked_prefill=getattr(local_cfg, "chunked_prefill",
                                           False),
            scheduler_prefill_first=getattr(local_cfg, "prefill_first",
                                            True),
            enable_mixed_batches=getattr(local_cfg, "mixed_batches",
                                         False),
            weights_path=weights,
            capture_at_init=True,
        )
        self.engine = engine or AsyncEngine(ecfg, device=device)
        tok_path = getattr(local_cfg, "tokenizer_path", None) or None
        if tok_path is None and weights:
            # HF checkpoint dirs ship tokenizer.json next to the
            # safetensors — use it without extra configuration
            from pathlib import Path

            cand = Path(weights)
            cand = (cand if cand.is_dir() else cand.parent) / \
                "tokenizer.json"
            if cand.is_file():
                tok_path = str(cand)
        self.tokenizer = None
        if tok_path is None and local_cfg.model == "gguf":
            # GGUF embeds its tokenizer in metadata — fully
            # self-contained checkpoint
            from ..engine.gguf import tokenizer_from_gguf

            self.tokenizer = tokenizer_from_gguf(weights)
        if self.tokenizer is None:
            self.tokenizer = get_tokenizer(preset.vocab_size, tok_path)
        # the checkpoint's own prompt format (tokenizer_config.json)
        from ..utils.tokenizer import load_chat_template

        self.chat_template = (load_chat_template(weights) if weights
                              else None)
        # authoritative stop ids from the checkpoint: token-NAME
        # heuristics miss chat-tuned EOS tokens (Qwen2's <|im_end|>,
        # Llama-3-Instruct's <|eot_id|>) — generation_config.json /
        # config.json carry the real eos_token_id (int or list)
        self.eos_ids = self._checkpoint_eos_ids(weights)
        if not self.eos_ids and self.tokenizer.eos_token_id is not None:
            self.eos_ids = (self.tokenizer.eos_token_id,)
        # OpenAI system_fingerprint: identifies the backend configuration
        # (determinism domain for `seed`)
        self.fingerprint = f"fp_igw-amd_{local_cfg.model}"
        # tokens that have gone THROUGH the SSE write path (incremented
        # after the chunk's socket write completes) — bench.py's serving
        # mode measures the wire rate from this, not the engine rate
        self.streamed_tokens = 0
        self.max_queue = int(getattr(local_cfg, "max_queue", 0) or 0)

    @staticmethod
    def _checkpoint_eos_ids(weights) -> tuple:
        """eos_token_id(s) from generation_config.json (preferred — the
        chat EOS lives there) or config.json; () when absent."""
        if not weights:
            return ()
        import json as _json
        from pathlib import Path

        d = Path(weights)
        d = d if d.is_dir() else d.parent
        ids: list = []
        for fname in ("generation_config.json", "config.json"):
            f = d / fname
            if not f.is_file():
                continue
            try:
                v = _json.loads(f.read_text()).get("eos_token_id")
            except ValueError:
                continue
            if isinstance(v, int):
                ids = [v]
            elif isinstance(v, list):
                ids = [int(x) for x in v if isinstance(x, int)]
            if ids:
                break
        return tuple(ids)

    # ---- provider interface --------------------------------------------
    async def list_models(self) -> dict:
        return {
            "provider": self.id,
            "object": "list",
            "data": [{
                "id": f"{self.id}/{self.model_name}",
                "object": "model",
                "created": int(time.time()),
                "owned_by": self.id,
                "served_by": self.id,
                # the engine KNOWS its context limit — enrichment
                # preserves provider-supplied values (metadata.py)
                "context_window": self.engine.engine.cfg.max_model_len,
            }],
        }

    def _sampling_from_request(self, req: dict) -> SamplingParams:
        max_tokens = req.get("max_completion_tokens") or req.get(
            "max_tokens") or 256
        stop_ids = tuple(self.eos_ids)
        # OpenAI logprobs: bool `logprobs` + optional `top_logprobs` 0..20
        lp = -1
        if req.get("logprobs"):
            lp = max(0, min(20, int(req.get("top_logprobs") or 0)))
        seed = req.get("seed")
        lb = None
        if req.get("logit_bias"):
            lb = {}
            for k, val in req["logit_bias"].items():
                tid = int(k)
                if not 0 <= tid < self.engine.engine.cfg.model.vocab_size:
                    raise HTTPError(
                        400, f"logit_bias token id {tid} outside the "
                        "model vocabulary")
                lb[tid] = max(-100.0, min(100.0, float(val)))

        # explicit JSON null means "use the default" (OpenAI semantics;
        # validate.py deliberately lets null through)
        def _num(key, dflt):
            v = req.get(key)
            return dflt if v is None else v

        return SamplingParams(
            max_new_tokens=int(max_tokens),
            temperature=float(_num("temperature", 1.0)),
            top_p=float(_num("top_p", 1.0)),
            top_k=int(_num("top_k", 0)),
            stop_token_ids=stop_ids,
            logprobs=lp,
            seed=int(seed) if seed is not None else None,
            presence_penalty=float(req.get("presence_penalty") or 0.0),
            frequency_penalty=float(req.get("frequency_penalty") or 0.0),
            logit_bias=lb,
        )

    def _lp_item(self, out) -> dict:
        """One OpenAI logprobs content entry for a generated token."""
        s = self.tokenizer.decode([out.token_id])
        tops = []
        for tid, v in (out.top_logprobs or []):
            ts = self.tokenizer.decode([tid])
            tops.append({"token": ts, "logprob": v,
                         "bytes": list(ts.encode("utf-8"))})
        return {"token": s, "logprob": out.logprob,
                "bytes": list(s.encode("utf-8")), "top_logprobs": tops}

    def _encode_request(self, req: dict):
        msgs = req.get("messages") or []
        if self.chat_template:
            from ..utils.tokenizer import render_chat_template

            prompt = render_chat_template(self.chat_template, msgs)
        else:
            prompt = apply_chat_template(msgs)
        return self.tokenizer.encode(prompt)

    @staticmethod
    def _stop_strings(req: dict):
        stop = req.get("stop")
        if isinstance(stop, str):
            return [stop]
        if isinstance(stop, list):
            return [s for s in stop if isinstance(s, str)][:4]
        return []

    def _scan_stop_progressive(self, tail: str, pending: list,
                               stops) -> tuple:
        """(cut_index, buffer) — stop scan with CANONICAL per-token
        semantics for a BATCHED delivery: grow the text token by token
        (exactly as the non-streaming _collect does) and return the
        earliest-start match inside the FIRST prefix that contains any
        stop.  A single whole-batch scan could cut at a globally
        EARLIER index whose match only completes with later tokens —
        tokens the per-token path never generates (it stops at the
        first completed stop) — so batching must not change the text.
        cut_index -1 => no stop; buffer is then tail + decode(all)."""
        if stops:
            i = self._find_stop(tail, stops)
            if i >= 0:
                return i, tail
            for k in range(1, len(pending) + 1):
                cand = tail + self.tokenizer.decode(pending[:k])
                i = self._find_stop(cand, stops)
                if i >= 0:
                    return i, cand
        return -1, tail + (self.tokenizer.decode(pending) if pending
                           else "")

    @staticmethod
    def _find_stop(text: str, stops) -> int:
        """Earliest stop-string match index, or -1."""
        hit = -1
        for s in stops:
            i = text.find(s)
            if i >= 0 and (hit < 0 or i < hit):
                hit = i
        return hit

    async def _collect(self, prompt_ids, sampling, stops, gen=None):
        """One full completion: returns (token_ids, text, finish_reason,
        lp_content); stop sequences are excluded from the text (OpenAI
        semantics; logprob entries for tokens past a stop-string cut are
        kept — the text cut falls mid-token in general)."""
        token_ids = []
        lp_content = [] if sampling.logprobs >= 0 else None
        finish_reason = "stop"
        if gen is None:
            gen = self.engine.generate(prompt_ids, sampling)
        async for out in gen:
            token_ids.append(out.token_id)
            if lp_content is not None:
                lp_content.append(self._lp_item(out))
            if out.finished:
                finish_reason = ("stop" if out.finish_reason == "stop"
                                 else "length")
            if stops:
                partial = self.tokenizer.decode(token_ids)
                i = self._find_stop(partial, stops)
                if i >= 0:
                    return token_ids, partial[:i], "stop", lp_content
        return (token_ids, self.tokenizer.decode(token_ids), finish_reason,
                lp_content)

    MAX_N = 8          # engine KV-fork group cap (OpenAI's own cap is 128)

    def _resolve_n(self, req: dict) -> int:
        """OpenAI semantics: an unsupported `n` is a 400, never a silent
        clamp (a client indexing choices by its requested n would break)."""
        n = int(req.get("n") or 1)
        if n < 1 or n > self.MAX_N:
            raise HTTPError(
                400, f"n must be between 1 and {self.MAX_N}, got {n}")
        return n

    def validate_request(self, req: dict) -> None:
        """Pre-stream validation hook (gateway/routes.py calls this
        before opening the SSE response)."""
        self._resolve_n(req)
        self._check_queue()
        self._check_context(self._encode_request(req), req)

    def _check_queue(self) -> None:
        """Admission cap (LOCAL_MAX_QUEUE): shed load with 429 instead
        of queueing unboundedly — waiting-queue depth is the TTFT a new
        request will pay.  0 = unbounded (the bench's burst-hold mode
        deliberately queues thousands)."""
        cap = self.max_queue
        if cap and len(self.engine.engine.scheduler.waiting) >= cap:
            raise HTTPError(
                429, f"server overloaded: {cap} requests already "
                "queued for prefill — retry later")

    def _check_context(self, prompt_ids, req: Optional[dict] = None) -> None:
        limit = self.engine.engine.cfg.max_model_len
        if len(prompt_ids) > limit:
            # OpenAI returns 400 for context overflow, not a 5xx
            raise HTTPError(
                400, f"prompt is {len(prompt_ids)} tokens but the "
                     f"model's maximum context length is {limit}")
        if req is not None:
            # OpenAI also 400s when an EXPLICIT max_tokens cannot fit the
            # context (prompt + max_tokens > limit); a request without
            # max_tokens keeps the truncate-at-limit semantics
            mt = req.get("max_completion_tokens") or req.get("max_tokens")
            if mt is not None and len(prompt_ids) + int(mt) > limit:
                raise HTTPError(
                    400, f"prompt ({len(prompt_ids)} tokens) plus "
                         f"max_tokens ({int(mt)}) exceeds the model's "
                         f"maximum context length of {limit}")

    async def chat_completions(self, req: dict,
                               extra_headers=None) -> dict:
        prompt_ids = self._encode_request(req)
        self._check_queue()
        self._check_context(prompt_ids, req)
        sampling = self._sampling_from_request(req)
        stops = self._stop_strings(req)
        n = self._resolve_n(req)
        if n <= 1:
            token_ids, text, finish_reason, lp = await self._collect(
                prompt_ids, sampling, stops)
            return self._completion_response(prompt_ids, token_ids, text,
                                             finish_reason, lp)
        # n > 1: ONE prefill — the engine forks the prompt KV into n
        # branches (full blocks shared ref-counted, tail block copied)
        gens = self.engine.generate_group(prompt_ids, sampling, n)
        results = await asyncio.gather(
            *(self._collect(prompt_ids, sampling, stops, gen=g)
              for g in gens))
        total = sum(len(t) for t, _, _, _ in results)
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:24]}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": f"{self.id}/{self.model_name}",
            "system_fingerprint": self.fingerprint,
            "choices": [{
                "index": i,
                "message": {"role": "assistant", "content": text},
                "logprobs": ({"content": lp} if lp is not None else None),
                "finish_reason": fin,
            } for i, (_, text, fin, lp) in enumerate(results)],
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": total,
                "total_tokens": len(prompt_ids) + total,
            },
        }

    def _completion_response(self, prompt_ids, token_ids, text: str,
                             finish_reason: str, lp_content=None) -> dict:
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:24]}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": f"{self.id}/{self.model_name}",
            "system_fingerprint": self.fingerprint,
            "choices": [{
                "index": 0,
                "message": {"role": "assistant", "content": text},
                "logprobs": ({"content": lp_content}
                             if lp_content is not None else None),
                "finish_reason": finish_reason,
            }],
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": len(token_ids),
                "total_tokens": len(prompt_ids) + len(token_ids),
            },
        }

    def open_direct_stream(self, req: dict, raw_write, raw_end,
                           raw_backlogged=None) -> DirectStream:
        """SSE fast path (n=1): returns a DirectStream session that the
        server awaits; validation errors raise HTTPError BEFORE the
        response opens (plain 400, like the generator path)."""
        prompt_ids = self._encode_request(req)
        self._check_context(prompt_ids, req)
        sampling = self._sampling_from_request(req)
        return DirectStream(self, req, prompt_ids, sampling, raw_write,
                            raw_end, raw_backlogged)

    async def stream_chat_completions(self, req: dict,
                                      extra_headers=None
                                      ) -> AsyncIterator[str]:
        """Yields SSE lines ('data: {...}') matching OpenAI stream chunks;
        per-token flush (reference bar: per-line SSE flush)."""
        prompt_ids = self._encode_request(req)
        self._check_context(prompt_ids, req)
        sampling = self._sampling_from_request(req)
        cid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        created = int(time.time())
        model = f"{self.id}/{self.model_name}"

        # pre-rendered chunk frame: at 256 concurrent streams the
        # per-token json.dumps shows up in the event loop (the SSE writer
        # is the serving bottleneck long before the engine is)
        prefix = ("data: " + json.dumps({
            "id": cid, "object": "chat.completion.chunk",
            "created": created, "model": model,
            "system_fingerprint": self.fingerprint,
        }, separators=(",", ":"))[:-1] + ',"choices":[{"index":0,"delta":')

        def chunk(delta: dict, finish: Optional[str] = None,
                  usage: Optional[dict] = None, lp=None) -> str:
            tail = ',"usage":' + json.dumps(
                usage, separators=(",", ":")) if usage is not None else ""
            fin = f'"{finish}"' if finish else "null"
            lps = (',"logprobs":' + json.dumps({"content": lp},
                                               separators=(",", ":"))
                   if lp is not None else "")
            return (prefix + json.dumps(delta, separators=(",", ":"))
                    + lps + ',"finish_reason":' + fin + "}]" + tail + "}")

        n = self._resolve_n(req)
        if n > 1:
            async for line in self._stream_multi(req, prompt_ids, sampling,
                                                 cid, created, model, n):
                yield line
            return

        yield chunk({"role": "assistant", "content": ""})
        stops = self._stop_strings(req)
        holdback = max((len(s) for s in stops), default=1) - 1
        emitted_tail = ""          # un-emitted text (stop-prefix holdback)
        stopped = False
        ntok = 0
        finish_reason = "stop"
        pending: list = []
        want_lp = sampling.logprobs >= 0
        pending_lp: list = []      # lp items for not-yet-flushed tokens
        # batched drain: one chunk (and one socket write) per group of
        # already-available tokens — per-token when the engine is the
        # bottleneck, coalesced when the wire is
        async for outs in self.engine.generate_batched(prompt_ids,
                                                       sampling):
            for out in outs:
                ntok += 1
                pending.append(out.token_id)
                if want_lp:
                    pending_lp.append(self._lp_item(out))
                if out.finished:
                    # record BEFORE the holdback continue: a length-finish
                    # whose tail is an incomplete UTF-8 run must not
                    # report "stop"
                    finish_reason = ("stop" if out.finish_reason == "stop"
                                     else "length")
            text = self.tokenizer.decode(pending)
            # hold back partial UTF-8 sequences (byte tokenizer)
            if text.endswith("�"):
                continue
            nflush = len(pending)
            if stops:
                i, buf = self._scan_stop_progressive(emitted_tail,
                                                     pending, stops)
            pending = []
            lp, pending_lp = (pending_lp, []) if want_lp else (None, [])
            if stops:
                if i >= 0:
                    # emit up to (excluding) the stop sequence, then end
                    if buf[:i]:
                        yield chunk({"content": buf[:i]}, lp=lp)
                    finish_reason = "stop"
                    stopped = True
                    self.streamed_tokens += nflush
                    break
                # keep a stop-prefix-sized tail unemitted
                emit = buf[:-holdback] if holdback else buf
                emitted_tail = buf[len(emit):]
                if emit:
                    yield chunk({"content": emit}, lp=lp)
                self.streamed_tokens += nflush
                continue
            yield chunk({"content": text}, lp=lp)
            # post-yield: the chunk's write has been handed to the kernel
            self.streamed_tokens += nflush
        if not stopped:
            # flush whatever is still held back: the stop-prefix tail AND
            # any pending tokens that never completed a UTF-8 sequence
            # (emit them with replacement chars — dropping tokens would
            # leave a stream with content missing entirely)
            i, leftover = self._scan_stop_progressive(emitted_tail,
                                                       pending, stops)
            lp = pending_lp if want_lp and pending_lp else None
            if i >= 0:
                leftover = leftover[:i]
                finish_reason = "stop"
            if leftover:
                yield chunk({"content": leftover}, lp=lp)
        usage = {
            "prompt_tokens": len(prompt_ids),
            "completion_tokens": ntok,
            "total_tokens": len(prompt_ids) + ntok,
        }
        yield chunk({}, finish=finish_reason, usage=usage)
        yield "data: [DONE]"

    async def _stream_multi(self, req: dict, prompt_ids, sampling,
                            cid: str, created: int, model: str,
                            n: int) -> AsyncIterator[str]:
        """Streaming with n>1: the engine forks the prompt KV; branch
        chunks interleave with per-choice `index` (OpenAI semantics).
        One [DONE]; the final chunk carries the whole group's usage."""
        stops = self._stop_strings(req)
        holdback = max((len(s) for s in stops), default=1) - 1
        want_lp = sampling.logprobs >= 0

        def chunk(idx: int, delta: dict, finish=None, usage=None, lp=None):
            ch = {"index": idx, "delta": delta,
                  "finish_reason": finish}
            if lp is not None:
                ch["logprobs"] = {"content": lp}
            frame = {"id": cid, "object": "chat.completion.chunk",
                     "created": created, "model": model,
                     "system_fingerprint": self.fingerprint,
                     "choices": [ch]}
            if usage is not None:
                frame["usage"] = usage
            return "data: " + json.dumps(frame, separators=(",", ":"))

        queue: asyncio.Queue = asyncio.Queue()
        gens = self.engine.generate_group(prompt_ids, sampling, n,
                                          batched=True)

        async def pump(idx: int, gen) -> None:
            pending, pending_lp = [], []
            emitted_tail, ntok = "", 0
            finish_reason = "stop"
            try:
                async for outs in gen:
                    for out in outs:
                        ntok += 1
                        pending.append(out.token_id)
                        if want_lp:
                            pending_lp.append(self._lp_item(out))
                        if out.finished:
                            finish_reason = ("stop" if out.finish_reason ==
                                             "stop" else "length")
                    text = self.tokenizer.decode(pending)
                    if text.endswith("�"):
                        continue
                    if stops:
                        i, buf = self._scan_stop_progressive(
                            emitted_tail, pending, stops)
                    pending = []
                    lp = pending_lp if want_lp else None
                    pending_lp = []
                    if stops:
                        if i >= 0:
                            if buf[:i]:
                                await queue.put(
                                    chunk(idx, {"content": buf[:i]}, lp=lp))
                            finish_reason = "stop"
                            emitted_tail = ""
                            break
                        emit = buf[:-holdback] if holdback else buf
                        emitted_tail = buf[len(emit):]
                        if emit:
                            await queue.put(
                                chunk(idx, {"content": emit}, lp=lp))
                        continue
                    await queue.put(chunk(idx, {"content": text}, lp=lp))
                # flush the held-back tail + any incomplete-UTF-8 pending
                i, leftover = self._scan_stop_progressive(
                    emitted_tail, pending, stops)
                if i >= 0:
                    leftover = leftover[:i]
                    finish_reason = "stop"
                if leftover:
                    await queue.put(chunk(
                        idx, {"content": leftover},
                        lp=pending_lp if want_lp and pending_lp else None))
            finally:
                try:
                    # a stop-string break leaves the branch generating;
                    # close the stream now so its abort fires promptly
                    await gen.aclose()
                except Exception:
                    pass
                await queue.put(("done", idx, ntok, finish_reason))

        tasks = [asyncio.ensure_future(pump(i, g))
                 for i, g in enumerate(gens)]
        try:
            for i in range(n):
                yield chunk(i, {"role": "assistant", "content": ""})
            total, done = 0, 0
            while done < n:
                item = await queue.get()
                if isinstance(item, tuple):
                    _, idx, ntok, fin = item
                    total += ntok
                    done += 1
                    usage = None
                    if done == n:
                        usage = {
                            "prompt_tokens": len(prompt_ids),
                            "completion_tokens": total,
                            "total_tokens": len(prompt_ids) + total,
                        }
                    yield chunk(idx, {}, finish=fin, usage=usage)
                else:
                    yield item
            yield "data: [DONE]"
        finally:
            for t in tasks:
                t.cancel()
