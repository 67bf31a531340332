"""serve.main() end-to-end: boot the real uvicorn server as a
subprocess (tiny model, random init), hit /health and /v1/completions
over HTTP, then terminate that exact process."""

import os
import signal
import subprocess
import sys
import time

import httpx
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_serve_main_boots_and_serves(tmp_path):
    port = 18100 + os.getpid() % 500
    proc = subprocess.Popen(
        [sys.executable, "-m", "distrl_llm_amd.serve",
         "--model", "tiny-qwen2", "--host", "127.0.0.1",
         "--port", str(port), "--max-seq-length", "128",
         "--enable-prefix-caching"],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    base = f"http://127.0.0.1:{port}"
    try:
        deadline = time.time() + 120
        last_err = None
        while time.time() < deadline:
            if proc.poll() is not None:
                out = proc.stdout.read()
                raise AssertionError(f"server exited early:\n{out[-3000:]}")
            try:
                r = httpx.get(base + "/health", timeout=2.0)
                if r.status_code == 200:
                    break
            except Exception as e:
                last_err = e
            time.sleep(0.5)
        else:
            raise AssertionError(f"server never came up: {last_err}")

        r = httpx.post(base + "/v1/completions",
                       json={"prompt": "2+2=", "max_tokens": 4,
                             "temperature": 0.0}, timeout=60.0)
        assert r.status_code == 200
        body = r.json()
        assert body["choices"][0]["finish_reason"] in ("stop", "length")
        assert httpx.get(base + "/metrics", timeout=5.0).status_code == 200
    finally:
        # kill the exact PID we started (never by pattern)
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=20)
