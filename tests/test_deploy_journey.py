"""Deploy journey: pretrained checkpoint + trained adapter -> merge
script -> serve the merged model. Chains hf_io, the PEFT adapter format,
scripts/merge_adapter.py and the serving surface the way a user would."""

import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_train_merge_serve_journey(tmp_path):
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.models.hf_io import (load_hf_checkpoint,
                                             resolve_spec,
                                             save_hf_checkpoint)
    from distrl_llm_amd.models.lora import save_adapter
    from distrl_llm_amd.serve import create_app
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    # 1. a "pretrained" base checkpoint + a "trained" adapter
    base_dir = str(tmp_path / "base")
    spec = get_spec("tiny-qwen2")
    m = CausalLM(spec, lora_r=4, lora_alpha=8,
                 dtype=torch.float32).random_init(41)
    save_hf_checkpoint(m, base_dir)
    with torch.no_grad():
        for name, p in m.named_parameters():
            if "lora_" in name:
                p.add_(torch.randn_like(p) * 0.05)
    adapter_dir = str(tmp_path / "adapter")
    save_adapter(m, adapter_dir, "tiny-qwen2", r=4, alpha=8)

    # 2. merge via the CLI script
    merged_dir = str(tmp_path / "merged")
    # rank/alpha auto-read from the adapter's adapter_config.json
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts/merge_adapter.py"),
         "--model", base_dir, "--adapter", adapter_dir,
         "--out", merged_dir],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "merged 14 projection sites" in out.stdout  # 7 sites x 2 layers

    # 3. serve the merged checkpoint and check it behaves like the
    #    adapter-bearing model
    served_spec = resolve_spec(merged_dir)
    served = CausalLM(served_spec, lora_r=0,
                      dtype=torch.float32).random_init(0)
    load_hf_checkpoint(served, merged_dir)
    engine = Engine(served, EngineConfig(max_seq_length=128, kv_block_size=8,
                                         num_kv_blocks=128, max_num_seqs=16),
                    device=torch.device("cpu"), seed=0)
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    from fastapi.testclient import TestClient
    client = TestClient(create_app(engine, tok, "merged"))
    r = client.post("/v1/completions", json={
        "prompt": "2+2=", "max_tokens": 5, "temperature": 0.0})
    assert r.status_code == 200
    served_text = r.json()["choices"][0]["text"]

    ref_engine = Engine(m, EngineConfig(max_seq_length=128, kv_block_size=8,
                                        num_kv_blocks=128, max_num_seqs=16),
                        device=torch.device("cpu"), seed=0)
    ids = tok.encode("2+2=")
    ref_out = ref_engine.generate(
        [ids], SamplingParams(max_tokens=5, temperature=0.0, n=1),
        eos_token_id=tok.eos_token_id)[0][0]
    keep = [t for t in ref_out if t < tok.vocab_size or t > 260]
    assert served_text == tok.decode(keep, skip_special_tokens=True)
