"""Serving surface: OpenAI-style /v1/completions over the engine (native
extension — the reference is in-process only, docs/ROADMAP.md #10)."""

import pytest
import torch


@pytest.fixture(scope="module")
def client():
    from fastapi.testclient import TestClient

    from distrl_llm_amd.config import EngineConfig
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.serve import create_app
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    spec = get_spec("tiny-qwen2")
    model = CausalLM(spec, lora_r=0, dtype=torch.float32).random_init(5)
    engine = Engine(model, EngineConfig(max_seq_length=128, kv_block_size=8,
                                        num_kv_blocks=256, max_num_seqs=32),
                    device=torch.device("cpu"), seed=0)
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    return TestClient(create_app(engine, tok, "tiny-qwen2"))


def test_health_and_models(client):
    assert client.get("/health").json() == {"status": "ok"}
    models = client.get("/v1/models").json()
    assert models["data"][0]["id"] == "tiny-qwen2"


def test_completion_single(client):
    r = client.post("/v1/completions", json={
        "prompt": "2+2=", "max_tokens": 6, "temperature": 0.0})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "text_completion"
    assert len(body["choices"]) == 1
    c = body["choices"][0]
    assert isinstance(c["text"], str) and c["finish_reason"] in ("stop", "length")
    assert body["usage"]["prompt_tokens"] == 4  # small vocab -> byte mode
    assert body["usage"]["total_tokens"] == (body["usage"]["prompt_tokens"]
                                             + body["usage"]["completion_tokens"])
    assert body["usage"]["completion_tokens"] <= 6
    # greedy decoding is deterministic
    r2 = client.post("/v1/completions", json={
        "prompt": "2+2=", "max_tokens": 6, "temperature": 0.0})
    assert r2.json()["choices"][0]["text"] == c["text"]


def test_completion_batch_and_n(client):
    r = client.post("/v1/completions", json={
        "prompt": ["abcd", "wxyz"], "max_tokens": 4, "temperature": 1.0,
        "top_p": 0.9, "n": 3})
    body = r.json()
    assert len(body["choices"]) == 6
    assert [c["index"] for c in body["choices"]] == list(range(6))


def test_chat_completion(client):
    r = client.post("/v1/chat/completions", json={
        "messages": [{"role": "system", "content": "Be brief."},
                     {"role": "user", "content": "hi"}],
        "max_tokens": 4, "temperature": 0.0})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "chat.completion"
    c = body["choices"][0]
    assert c["message"]["role"] == "assistant"
    assert isinstance(c["message"]["content"], str)
    assert body["usage"]["prompt_tokens"] > 10  # chat template applied
    r2 = client.post("/v1/chat/completions", json={
        "messages": [], "max_tokens": 4})
    assert r2.status_code == 400


def test_seeded_request_deterministic(client):
    body = {"prompt": "roll", "max_tokens": 5, "temperature": 1.0,
            "seed": 42}
    t1 = client.post("/v1/completions", json=body).json()["choices"][0]["text"]
    # interleave an unseeded request to disturb engine RNG state
    client.post("/v1/completions", json={"prompt": "x", "max_tokens": 3,
                                         "temperature": 1.0})
    t2 = client.post("/v1/completions", json=body).json()["choices"][0]["text"]
    assert t1 == t2


def test_completion_logprobs_rescoring(client):
    """logprobs=N: greedy-chosen tokens must carry the row-max model
    log-prob and appear first in their own top_logprobs."""
    r = client.post("/v1/completions", json={
        "prompt": "abc", "max_tokens": 4, "temperature": 0.0,
        "logprobs": 2})
    body = r.json()
    lp = body["choices"][0]["logprobs"]
    n = len(lp["tokens"])
    assert n == body["usage"]["completion_tokens"] > 0
    assert len(lp["token_logprobs"]) == n and len(lp["top_logprobs"]) == n
    for j in range(n):
        assert lp["token_logprobs"][j] <= 0.0
        top = lp["top_logprobs"][j]
        assert len(top) == 2
        # greedy: the chosen token is the argmax -> the top entry's value
        # equals the chosen token's logprob
        assert max(top.values()) == pytest.approx(lp["token_logprobs"][j],
                                                  abs=1e-5)
    # default: no logprobs field computed
    r2 = client.post("/v1/completions", json={
        "prompt": "abc", "max_tokens": 2, "temperature": 0.0})
    assert r2.json()["choices"][0]["logprobs"] is None


def _collect_sse(resp):
    events = []
    for line in resp.iter_lines():
        if line.startswith("data: "):
            payload = line[len("data: "):]
            if payload == "[DONE]":
                events.append(None)
            else:
                import json
                events.append(json.loads(payload))
    return events


def test_streaming_completion_matches_nonstreaming(client):
    body = {"prompt": "2+2=", "max_tokens": 6, "temperature": 0.0}
    full = client.post("/v1/completions", json=body).json()["choices"][0]["text"]

    with client.stream("POST", "/v1/completions",
                       json={**body, "stream": True}) as r:
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("text/event-stream")
        events = _collect_sse(r)
    assert events[-1] is None  # [DONE]
    chunks = [e["choices"][0]["text"] for e in events[:-1]]
    finishes = [e["choices"][0]["finish_reason"] for e in events[:-1]]
    assert "".join(chunks) == full
    assert finishes[-1] in ("stop", "length")
    assert len(chunks) >= 2  # CPU path streams incrementally, not one blob


def test_streaming_chat_and_validation(client):
    with client.stream("POST", "/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 4, "temperature": 0.0, "stream": True}) as r:
        assert r.status_code == 200
        events = _collect_sse(r)
    assert events[-1] is None
    assert events[0]["object"] == "chat.completion.chunk"
    assert events[-2]["choices"][0]["finish_reason"] in ("stop", "length")
    # multi-prompt / n>1 streaming rejected
    r = client.post("/v1/completions", json={
        "prompt": ["a", "b"], "max_tokens": 2, "stream": True})
    assert r.status_code == 400
    r = client.post("/v1/completions", json={
        "prompt": "a", "n": 2, "max_tokens": 2, "temperature": 1.0,
        "stream": True})
    assert r.status_code == 400


def test_prometheus_metrics(client):
    client.post("/v1/completions", json={
        "prompt": "abc", "max_tokens": 3, "temperature": 0.0})
    body = client.get("/metrics").text
    assert "distrl_requests_total" in body
    assert 'endpoint="completions",status="ok"' in body
    assert "distrl_generated_tokens_total" in body
    assert "distrl_engine_calls_total" in body


def test_completion_echo_and_validation(client):
    r = client.post("/v1/completions", json={
        "prompt": "hi", "max_tokens": 2, "temperature": 0.0, "echo": True})
    assert r.json()["choices"][0]["text"].startswith("hi")
    # invalid sampling params -> 400, not a crash
    r = client.post("/v1/completions", json={"prompt": "x", "max_tokens": 0})
    assert r.status_code == 400
    r = client.post("/v1/completions", json={"prompt": [], "max_tokens": 2})
    assert r.status_code == 400


def test_streaming_disconnect_aborts_generation(client):
    """Closing the SSE stream mid-generation (client disconnect) aborts
    the request: the engine frees its decode slots and KV blocks, and
    the server keeps serving subsequent requests normally."""
    import time
    body = {"prompt": "count with me now", "max_tokens": 120,
            "temperature": 0.0, "stream": True}
    with client.stream("POST", "/v1/completions", json=body) as r:
        assert r.status_code == 200
        it = r.iter_lines()
        first = next(line for line in it if line.startswith("data:"))
        assert "[DONE]" not in first
        # exit the context without draining -> GeneratorExit -> cancel
    time.sleep(1.0)  # let the batcher thread finish the aborted wave
    # server still healthy and a fresh request completes normally
    out = client.post("/v1/completions",
                      json={"prompt": "hello", "max_tokens": 4,
                            "temperature": 0.0})
    assert out.status_code == 200
    assert out.json()["choices"][0]["text"] is not None


def test_concurrent_mixed_storm(client):
    """Production-shaped concurrency: blocking completions, chats,
    seeded requests, streams that complete, and streams that disconnect
    mid-generation — all in flight together. Every response must be
    well-formed and the server must stay healthy throughout."""
    import json as _json
    from concurrent.futures import ThreadPoolExecutor

    def blocking(i):
        r = client.post("/v1/completions",
                        json={"prompt": f"req {i}", "max_tokens": 6,
                              "temperature": 0.0})
        assert r.status_code == 200
        return r.json()["choices"][0]["text"]

    def chat(i):
        r = client.post("/v1/chat/completions", json={
            "messages": [{"role": "user", "content": f"chat {i}"}],
            "max_tokens": 5, "temperature": 0.0})
        assert r.status_code == 200
        return r.json()["choices"][0]["message"]["content"]

    def seeded(i):
        r = client.post("/v1/completions",
                        json={"prompt": "seeded", "max_tokens": 5,
                              "temperature": 0.9, "seed": 42})
        assert r.status_code == 200
        return r.json()["choices"][0]["text"]

    def stream_full(i):
        chunks = []
        with client.stream("POST", "/v1/completions",
                           json={"prompt": f"stream {i}", "max_tokens": 6,
                                 "temperature": 0.0, "stream": True}) as r:
            assert r.status_code == 200
            for line in r.iter_lines():
                if line.startswith("data:") and "[DONE]" not in line:
                    chunks.append(_json.loads(line[5:]))
        assert chunks
        return "".join(c["choices"][0].get("text", "") for c in chunks)

    def stream_abort(i):
        with client.stream("POST", "/v1/completions",
                           json={"prompt": f"abort {i}", "max_tokens": 100,
                                 "temperature": 0.0, "stream": True}) as r:
            assert r.status_code == 200
            next(line for line in r.iter_lines()
                 if line.startswith("data:"))
        return "aborted"

    jobs = []
    with ThreadPoolExecutor(max_workers=12) as ex:
        for i in range(4):
            jobs += [ex.submit(blocking, i), ex.submit(chat, i),
                     ex.submit(seeded, i), ex.submit(stream_full, i),
                     ex.submit(stream_abort, i)]
        results = [f.result(timeout=180) for f in jobs]
    assert len(results) == 20
    # seeded requests are deterministic even under the storm
    seeds = [r for f, r in zip(jobs, results)][2::5]
    assert all(s == seeds[0] for s in seeds)
    # server healthy after the storm
    assert client.get("/health").json() == {"status": "ok"}


def test_serving_with_prefix_caching():
    """A cache-enabled server handling chat traffic with a shared system
    prompt: responses equal the uncached server's, and the cache is
    actually hit on the repeated prefix."""
    from fastapi.testclient import TestClient

    from distrl_llm_amd.config import EngineConfig
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.serve import create_app
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    spec = get_spec("tiny-qwen2")
    model = CausalLM(spec, lora_r=0, dtype=torch.float32).random_init(5)
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    cfg = dict(max_seq_length=128, kv_block_size=8, num_kv_blocks=256,
               max_num_seqs=32)
    plain_eng = Engine(model, EngineConfig(**cfg),
                       device=torch.device("cpu"), seed=0)
    cached_eng = Engine(model, EngineConfig(enable_prefix_caching=True,
                                            **cfg),
                        device=torch.device("cpu"), seed=0)
    plain = TestClient(create_app(plain_eng, tok, "tiny-qwen2"))
    cached = TestClient(create_app(cached_eng, tok, "tiny-qwen2"))

    system = ("You are a terse assistant. Answer in one word whenever "
              "possible and never apologize.")
    for user in ("hi there", "what is 2+2?", "name a color"):
        body = {"messages": [{"role": "system", "content": system},
                             {"role": "user", "content": user}],
                "max_tokens": 6, "temperature": 0.0}
        a = plain.post("/v1/chat/completions", json=body).json()
        b = cached.post("/v1/chat/completions", json=body).json()
        assert (a["choices"][0]["message"]["content"]
                == b["choices"][0]["message"]["content"])
    assert cached_eng._prefix_hits > 0  # the system prompt was reused


def test_concurrent_storm_with_prefix_caching():
    """The mixed-traffic storm against a cache-enabled server: correct
    well-formed responses under concurrency while the prefix cache is
    being hit and possibly evicted; the engine survives with no block
    leak beyond the cache's own holdings."""
    from concurrent.futures import ThreadPoolExecutor

    from fastapi.testclient import TestClient

    from distrl_llm_amd.config import EngineConfig
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.serve import create_app
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    spec = get_spec("tiny-qwen2")
    model = CausalLM(spec, lora_r=0, dtype=torch.float32).random_init(5)
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    eng = Engine(model, EngineConfig(max_seq_length=128, kv_block_size=8,
                                     num_kv_blocks=128, max_num_seqs=32,
                                     enable_prefix_caching=True),
                 device=torch.device("cpu"), seed=0)
    client = TestClient(create_app(eng, tok, "tiny-qwen2"))
    system = "Shared system preamble used by every request here."

    def chat(i):
        r = client.post("/v1/chat/completions", json={
            "messages": [{"role": "system", "content": system},
                         {"role": "user", "content": f"q{i}"}],
            "max_tokens": 5, "temperature": 0.0})
        assert r.status_code == 200
        return r.json()["choices"][0]["message"]["content"]

    with ThreadPoolExecutor(max_workers=8) as ex:
        results = [f.result(timeout=120)
                   for f in [ex.submit(chat, i) for i in range(16)]]
    assert len(results) == 16
    assert eng._prefix_hits > 0
    # deterministic: identical user turns give identical answers
    again = chat(3)
    assert again == results[3]
    # block accounting: everything free except the cache's holdings
    assert (eng.pool.allocator.num_free
            == eng.pool.num_blocks - len(eng._prefix_cache))
