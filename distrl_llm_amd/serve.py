"""Minimal OpenAI-style serving surface over the generation engine.

The reference has no serving API (Engine use is in-process only —
SURVEY.md §2.3 / docs/ROADMAP.md #10); this is a native extension so the
paged-KV engine can also be used as a standalone completion server:

    python -m distrl_llm_amd.serve --model qwen2.5-7b --port 8000

Endpoints (OpenAI completions-compatible subset):
    GET  /health              liveness
    GET  /metrics             Prometheus counters/histograms
    GET  /v1/models           the single served model
    POST /v1/completions      prompt(s) -> n sampled completions
    POST /v1/chat/completions chat-templated messages -> completions
Both POST endpoints accept ``"stream": true`` (SSE; single prompt, n=1)
and ``"seed"`` for per-request determinism (seeded/streaming requests
run in their own decode wave).

Handlers are sync ``def`` so FastAPI runs them in its threadpool;
concurrent requests are merged into shared decode waves by
``engine.batcher.DynamicBatcher`` (requests with identical sampling
params co-batch into one ``engine.generate`` call — one fused-weight
refresh and one hipGraph wave instead of one per request).
"""

# NOTE: no `from __future__ import annotations` here — FastAPI resolves
# endpoint annotations by name at request time, and CompletionRequest is
# local to create_app (postponed annotations would break body binding).
import time
from typing import List, Optional, Union

from .config import SamplingParams
from .engine.batcher import DynamicBatcher


def create_app(engine, tokenizer, model_name: str,
               batch_wait_ms: float = 2.0):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class CompletionRequest(BaseModel):
        prompt: Union[str, List[str]]
        model: Optional[str] = None
        max_tokens: int = 16
        temperature: float = 1.0
        top_p: float = 1.0
        top_k: int = 0
        n: int = 1
        echo: bool = False
        seed: Optional[int] = None
        stream: bool = False
        logprobs: Optional[int] = None
        stop: Optional[Union[str, List[str]]] = None

    class ChatMessage(BaseModel):
        role: str
        content: str

    class ChatRequest(BaseModel):
        messages: List[ChatMessage]
        model: Optional[str] = None
        max_tokens: int = 16
        temperature: float = 1.0
        top_p: float = 1.0
        top_k: int = 0
        n: int = 1
        seed: Optional[int] = None
        stream: bool = False
        stop: Optional[Union[str, List[str]]] = None

    app = FastAPI(title="distrl-mi355x", version="0.1")
    batcher = DynamicBatcher(engine, max_wait_ms=batch_wait_ms)
    app.state.batcher = batcher
    created = int(time.time())

    # Prometheus observability (per-process registry so tests and
    # multiple apps don't collide on the global default registry)
    from prometheus_client import (CollectorRegistry, Counter, Gauge,
                                   Histogram, generate_latest)
    registry = CollectorRegistry()
    m_requests = Counter("distrl_requests_total", "completion requests",
                         ["endpoint", "status"], registry=registry)
    m_latency = Histogram("distrl_request_seconds", "request latency",
                          ["endpoint"], registry=registry,
                          buckets=(.05, .1, .25, .5, 1, 2.5, 5, 10, 30, 60))
    m_prompt_toks = Counter("distrl_prompt_tokens_total",
                            "prompt tokens consumed", registry=registry)
    m_gen_toks = Counter("distrl_generated_tokens_total",
                         "tokens generated", registry=registry)
    m_engine_calls = Gauge("distrl_engine_calls_total",
                           "engine.generate invocations (batched waves)",
                           registry=registry)
    app.state.metrics_registry = registry

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.get("/metrics")
    def metrics():
        from fastapi import Response
        m_engine_calls.set(batcher.calls)
        return Response(generate_latest(registry),
                        media_type="text/plain; version=0.0.4")

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": model_name, "object": "model",
                          "created": created, "owned_by": "distrl-mi355x"}]}

    def _sp_or_400(req):
        try:
            return SamplingParams(max_tokens=req.max_tokens,
                                  temperature=req.temperature,
                                  top_p=req.top_p, top_k=req.top_k, n=req.n,
                                  seed=req.seed)
        except ValueError as e:
            raise HTTPException(status_code=400, detail=str(e))

    def _complete(prompts, sp, stop, endpoint="completions"):
        """Shared generation core: returns
        (flat [(text, finish, out_ids)], usage)."""
        stops = ([stop] if isinstance(stop, str) else stop) or []
        prompt_ids = [tokenizer.encode(p) for p in prompts]
        eos = getattr(tokenizer, "eos_token_id", None)
        t0 = time.monotonic()
        try:
            outs = batcher.submit(prompt_ids, sp, eos_token_id=eos)
        except Exception:
            m_requests.labels(endpoint=endpoint, status="error").inc()
            raise
        m_requests.labels(endpoint=endpoint, status="ok").inc()
        m_latency.labels(endpoint=endpoint).observe(time.monotonic() - t0)
        flat, completion_tokens = [], 0
        vocab = getattr(tokenizer, "vocab_size", None)
        for per_prompt in outs:
            for out_ids in per_prompt:
                completion_tokens += len(out_ids)
                finish = "stop" if (eos is not None and out_ids
                                    and out_ids[-1] == eos) else "length"
                keep = [t for t in out_ids
                        if vocab is None or t < vocab or t > 260]
                text = tokenizer.decode(keep, skip_special_tokens=True)
                for s in stops:
                    cut = text.find(s)
                    if cut >= 0:
                        text, finish = text[:cut], "stop"
                flat.append((text, finish, out_ids))
        prompt_tokens = sum(len(ids) for ids in prompt_ids)
        m_prompt_toks.inc(prompt_tokens)
        m_gen_toks.inc(completion_tokens)
        usage = {"prompt_tokens": prompt_tokens,
                 "completion_tokens": completion_tokens,
                 "total_tokens": prompt_tokens + completion_tokens}
        return flat, usage

    def _decode_delta(toks):
        vocab = getattr(tokenizer, "vocab_size", None)
        keep = [t for t in toks if vocab is None or t < vocab or t > 260]
        return tokenizer.decode(keep, skip_special_tokens=True)

    def _sse_stream(prompt_ids, sp, rid, obj, delta_fn, endpoint):
        """SSE generator over one streaming request (single prompt, n=1;
        per-token events on CPU, per-decode-chunk on GPU). Stop-string
        truncation is not applied mid-stream (tokens are emitted as
        sampled). Client disconnect aborts generation within one decode
        chunk (the cancel event reaches the engine's cancel_check) so
        dropped connections stop consuming decode slots."""
        import json as _json
        import threading as _threading
        eos = getattr(tokenizer, "eos_token_id", None)
        cancel = _threading.Event()
        q = batcher.submit_stream(prompt_ids, sp, eos_token_id=eos,
                                  cancel_event=cancel)
        try:
            yield from _sse_events(q, rid, obj, delta_fn, endpoint, eos,
                                   _json)
        finally:
            # normal completion: no-op (generation already finished);
            # disconnect (GeneratorExit): aborts the in-flight request
            cancel.set()

    def _sse_events(q, rid, obj, delta_fn, endpoint, eos, _json):
        finish = "length"
        while True:
            kind, *rest = q.get()
            if kind == "err":
                m_requests.labels(endpoint=endpoint, status="error").inc()
                yield "data: " + _json.dumps({"error": str(rest[0])}) + "\n\n"
                break
            if kind == "done":
                result = rest[0]
                out_ids = result[0][0]
                if eos is not None and out_ids and out_ids[-1] == eos:
                    finish = "stop"
                m_requests.labels(endpoint=endpoint, status="ok").inc()
                m_gen_toks.inc(len(out_ids))
                yield ("data: " + _json.dumps(
                    {"id": rid, "object": obj, "model": model_name,
                     "choices": [delta_fn("", finish)]}) + "\n\n")
                yield "data: [DONE]\n\n"
                break
            _pi, _ci, toks = rest  # kind == "tok"
            text = _decode_delta(toks)
            if text:
                yield ("data: " + _json.dumps(
                    {"id": rid, "object": obj, "model": model_name,
                     "choices": [delta_fn(text, None)]}) + "\n\n")

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        prompts = [req.prompt] if isinstance(req.prompt, str) else list(req.prompt)
        if not prompts:
            raise HTTPException(status_code=400, detail="empty prompt")
        sp = _sp_or_400(req)
        if req.stream:
            from fastapi.responses import StreamingResponse
            if len(prompts) != 1 or sp.n != 1:
                raise HTTPException(status_code=400,
                                    detail="streaming supports a single "
                                           "prompt with n=1")
            rid = f"cmpl-{created}-{int(time.time() * 1e6) & 0xFFFFFF:x}"
            gen = _sse_stream([tokenizer.encode(prompts[0])], sp, rid,
                              "text_completion",
                              lambda text, fin: {"index": 0, "text": text,
                                                 "finish_reason": fin,
                                                 "logprobs": None},
                              "completions")
            return StreamingResponse(gen, media_type="text/event-stream")
        flat, usage = _complete(prompts, sp, req.stop)
        prompt_ids = [tokenizer.encode(p) for p in prompts]
        choices = []
        for i, (text, finish, out_ids) in enumerate(flat):
            lp = None
            if req.logprobs is not None:
                lp = _rescore_logprobs(prompt_ids[i // sp.n], out_ids,
                                       min(max(req.logprobs, 0), 5))
            choices.append({"index": i,
                            "text": (prompts[i // sp.n] + text)
                                    if req.echo else text,
                            "finish_reason": finish, "logprobs": lp})
        return {"id": f"cmpl-{created}-{int(time.time() * 1e6) & 0xFFFFFF:x}",
                "object": "text_completion",
                "created": int(time.time()),
                "model": model_name,
                "choices": choices,
                "usage": usage}

    def _rescore_logprobs(prompt_ids, out_ids, top_n):
        """OpenAI-style logprobs by teacher-forced rescoring: one extra
        forward over prompt+completion, model log-probs of the chosen
        tokens (pre-temperature, as the OpenAI API defines them) plus the
        top-N alternatives per position."""
        import torch
        if not out_ids:
            return {"tokens": [], "token_logprobs": [],
                    "top_logprobs": [], "text_offset": []}
        ids = torch.tensor([list(prompt_ids) + list(out_ids)],
                           device=engine.device)
        with torch.no_grad():
            logits = engine.model(ids)[0].float()
        lp = torch.log_softmax(logits, -1)
        start = len(prompt_ids) - 1
        tokens, tlp, top = [], [], []
        for j, t in enumerate(out_ids):
            row = lp[start + j]
            tokens.append(_decode_delta([t]))
            tlp.append(float(row[t]))
            if top_n > 0:
                tv, ti = row.topk(top_n)
                top.append({_decode_delta([int(i)]): float(v)
                            for v, i in zip(tv, ti)})
            else:
                top.append(None)
        return {"tokens": tokens, "token_logprobs": tlp,
                "top_logprobs": top, "text_offset": []}

    @app.post("/v1/chat/completions")
    def chat_completions(req: ChatRequest):
        if not req.messages:
            raise HTTPException(status_code=400, detail="empty messages")
        sp = _sp_or_400(req)
        from .rl.data import apply_template
        prompt = apply_template(tokenizer,
                                [m.model_dump() for m in req.messages])
        if req.stream:
            from fastapi.responses import StreamingResponse
            if sp.n != 1:
                raise HTTPException(status_code=400,
                                    detail="streaming supports n=1")
            rid = f"chatcmpl-{created}-{int(time.time() * 1e6) & 0xFFFFFF:x}"
            gen = _sse_stream([tokenizer.encode(prompt)], sp, rid,
                              "chat.completion.chunk",
                              lambda text, fin: {"index": 0,
                                                 "delta": ({"content": text}
                                                           if fin is None
                                                           else {}),
                                                 "finish_reason": fin},
                              "chat")
            return StreamingResponse(gen, media_type="text/event-stream")
        flat, usage = _complete([prompt], sp, req.stop, endpoint="chat")
        choices = [{"index": i,
                    "message": {"role": "assistant", "content": text},
                    "finish_reason": finish}
                   for i, (text, finish, _ids) in enumerate(flat)]
        return {"id": f"chatcmpl-{created}-{int(time.time() * 1e6) & 0xFFFFFF:x}",
                "object": "chat.completion",
                "created": int(time.time()),
                "model": model_name,
                "choices": choices,
                "usage": usage}

    return app


def main():
    import argparse

    import torch

    from .config import EngineConfig
    from .engine.engine import Engine
    from .models.hf_io import (is_hf_checkpoint_dir, load_hf_checkpoint,
                               resolve_spec)
    from .models.model import CausalLM
    from .models.spec import is_4bit_model_name
    from .utils.tokenizer import load_tokenizer

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", type=str, default="qwen2.5-7b")
    ap.add_argument("--host", type=str, default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--max-seq-length", type=int, default=4096)
    ap.add_argument("--gpu-memory-utilization", type=float, default=0.9)
    ap.add_argument("--adapter", type=str, default=None,
                    help="PEFT adapter directory to apply before serving")
    ap.add_argument("--enable-prefix-caching", action="store_true",
                    help="reuse KV blocks of shared prompt prefixes "
                         "across requests (vLLM APC analogue)")
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    spec = resolve_spec(args.model)
    lora_r, lora_alpha = 0, 16.0
    if args.adapter:
        from .models.lora import adapter_hyperparams
        lora_r, lora_alpha, _ = adapter_hyperparams(args.adapter)
    model = CausalLM(spec, lora_r=lora_r, lora_alpha=lora_alpha,
                     dtype=dtype, device=device)
    model.random_init(args.seed)
    if is_hf_checkpoint_dir(args.model):
        load_hf_checkpoint(model, args.model)
    if is_4bit_model_name(args.model):
        model.quantize_nf4_()
    if args.adapter:
        from .models.lora import load_adapter
        load_adapter(model, args.adapter)
    tokenizer = load_tokenizer(args.model, spec.vocab_size)
    engine = Engine(model, EngineConfig(
        max_seq_length=args.max_seq_length,
        gpu_memory_utilization=args.gpu_memory_utilization,
        enable_prefix_caching=args.enable_prefix_caching),
        device=device, seed=args.seed)

    import uvicorn
    uvicorn.run(create_app(engine, tokenizer, args.model),
                host=args.host, port=args.port)


if __name__ == "__main__":
    main()
