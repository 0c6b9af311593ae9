"""/v1/responses end-to-end: gateway -> real (tiny) backend serving the
Open Responses API shape, stream + non-stream, with token accounting."""
import json
import socket
import subprocess
import sys
import tempfile
import time
import urllib.request
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parents[1]
BIN = REPO / "gateway" / "bin"


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def stack():
    backend_port, gw_port = _free_port(), _free_port()
    backend = subprocess.Popen(
        [sys.executable, "-m", "llmlb_amd.backend.launcher", "--model",
         "llama-tiny", "--device", "cpu", "--port", str(backend_port)],
        cwd=str(REPO), stdout=subprocess.DEVNULL,
        stderr=subprocess.DEVNULL)
    import os
    db = tempfile.mktemp(suffix=".sqlite3")
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "2"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            socket.create_connection(("127.0.0.1", backend_port),
                                     0.2).close()
            break
        except OSError:
            time.sleep(0.2)
    req = urllib.request.Request(
        base + "/api/endpoints",
        data=json.dumps({"name": "b", "base_url":
                         f"http://127.0.0.1:{backend_port}"}).encode(),
        headers={"Authorization": "Bearer t"})
    for _ in range(40):
        try:
            urllib.request.urlopen(req, timeout=10)
            break
        except Exception:
            time.sleep(0.3)
    yield {"base": base, "backend": f"http://127.0.0.1:{backend_port}"}
    gw.terminate()
    backend.terminate()


def test_responses_nonstream(stack):
    req = urllib.request.Request(
        stack["base"] + "/v1/responses",
        data=json.dumps({"model": "llama-tiny",
                         "input": "say something",
                         "max_tokens": 6}).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=30) as resp:
        obj = json.loads(resp.read())
    assert obj["object"] == "response"
    assert obj["status"] == "completed"
    text = obj["output"][0]["content"][0]["text"]
    assert text.strip()
    assert obj["usage"]["output_tokens"] == 6


def test_responses_streaming(stack):
    req = urllib.request.Request(
        stack["base"] + "/v1/responses",
        data=json.dumps({"model": "llama-tiny", "input": "go",
                         "max_tokens": 5, "stream": True}).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=30) as resp:
        assert "text/event-stream" in resp.headers.get("Content-Type", "")
        raw = resp.read().decode()
    assert '"response.created"' in raw
    assert raw.count('"response.output_text.delta"') == 5
    assert '"response.completed"' in raw
    assert '"output_tokens": 5' in raw or '"output_tokens":5' in raw
    assert "[DONE]" in raw


def test_responses_tokens_recorded(stack):
    time.sleep(0.5)
    req = urllib.request.Request(stack["base"] + "/api/requests",
                                 headers={"Authorization": "Bearer t"})
    with urllib.request.urlopen(req, timeout=10) as resp:
        recs = json.loads(resp.read())
    resp_recs = [r for r in recs if r["api_kind"] == "responses"
                 and r["status"] == 200]
    assert resp_recs
    assert any(r["completion_tokens"] > 0 for r in resp_recs)


def test_chat_logprobs_through_gateway(stack):
    """OpenAI chat logprobs: logprobs=true + top_logprobs=k returns a
    content array with per-token logprob and k alternatives."""
    body = json.dumps({"model": "llama-tiny",
                       "messages": [{"role": "user", "content": "hi"}],
                       "max_tokens": 3, "logprobs": True,
                       "top_logprobs": 2}).encode()
    req = urllib.request.Request(
        stack["base"] + "/v1/chat/completions", data=body,
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=30) as r:
        out = json.loads(r.read())
    lp = out["choices"][0]["logprobs"]
    assert lp and len(lp["content"]) == 3
    first = lp["content"][0]
    assert first["logprob"] <= 0.0
    assert len(first["top_logprobs"]) == 2


def test_n_choices(stack):
    """n>1 returns multiple choices; greedy choices are identical,
    sampled ones (temperature + seed) differ."""
    body = json.dumps({"model": "llama-tiny",
                       "messages": [{"role": "user", "content": "hi"}],
                       "max_tokens": 4, "n": 3, "temperature": 0.9,
                       "seed": 5}).encode()
    req = urllib.request.Request(
        stack["base"] + "/v1/chat/completions", data=body,
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=30) as r:
        out = json.loads(r.read())
    assert [c["index"] for c in out["choices"]] == [0, 1, 2]
    texts = {c["message"]["content"] for c in out["choices"]}
    assert len(texts) >= 2           # per-choice seeds differ
    assert out["usage"]["completion_tokens"] == 12


def test_embeddings_from_engine(stack):
    """/v1/embeddings served by the REAL backend (dense forward,
    unit-normalized last-token hidden state), routed by the gateway."""
    body = json.dumps({"model": "llama-tiny",
                       "input": ["hello world", "another text"]}).encode()
    req = urllib.request.Request(
        stack["base"] + "/v1/embeddings", data=body,
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=30) as r:
        out = json.loads(r.read())
    assert len(out["data"]) == 2
    v0 = out["data"][0]["embedding"]
    assert len(v0) == 256                # tiny hidden size
    import math
    assert abs(math.sqrt(sum(x * x for x in v0)) - 1.0) < 1e-3
    # deterministic: same input -> same vector
    with urllib.request.urlopen(urllib.request.Request(
            stack["base"] + "/v1/embeddings", data=body,
            headers={"Content-Type": "application/json"}),
            timeout=30) as r:
        out2 = json.loads(r.read())
    assert out2["data"][0]["embedding"] == v0


def test_disconnect_aborts_generation(stack):
    """Closing a streaming connection mid-generation aborts the
    request on the engine (pages freed, active count drops)."""
    import socket as _s
    host, port = stack["backend"].replace("http://", "").split(":")
    body = json.dumps({"model": "llama-tiny", "max_tokens": 500,
                       "stream": True,
                       "messages": [{"role": "user",
                                     "content": "hi"}]}).encode()
    c = _s.create_connection((host, int(port)), timeout=10)
    c.sendall((f"POST /v1/chat/completions HTTP/1.1\r\n"
               f"Host: {host}\r\nContent-Type: application/json\r\n"
               f"Content-Length: {len(body)}\r\n\r\n").encode() + body)
    c.recv(4096)          # first bytes of the stream
    c.close()             # abandon mid-generation
    deadline = time.time() + 20
    while time.time() < deadline:
        with urllib.request.urlopen(stack["backend"] + "/api/health",
                                    timeout=5) as r:
            h = json.loads(r.read())
        if h["active_requests"] == 0 and h["queued_requests"] == 0:
            return
        time.sleep(0.3)
    raise AssertionError(f"generation not aborted: {h}")


def test_context_length_exceeded(stack):
    body = json.dumps({"model": "llama-tiny", "max_tokens": 4,
                       "messages": [{"role": "user",
                                     "content": " ".join(
                                         f"w{i}" for i in range(600))}]
                       }).encode()
    req = urllib.request.Request(
        stack["backend"] + "/v1/chat/completions", data=body,
        headers={"Content-Type": "application/json"})
    try:
        urllib.request.urlopen(req, timeout=15)
        raise AssertionError("expected 400")
    except urllib.error.HTTPError as e:
        assert e.code == 400
        assert b"context_length_exceeded" in e.read()


def test_stop_strings(stack):
    """OpenAI `stop`: generation truncates at the stop sequence and
    reports finish_reason=stop (synthetic tokenizer emits 'tokN '
    pieces, so stopping on a piece we know will appear works)."""
    # discover the first generated piece for this prompt
    base_req = {"model": "llama-tiny", "max_tokens": 8,
                "messages": [{"role": "user", "content": "stop test"}]}
    r = urllib.request.Request(
        stack["backend"] + "/v1/chat/completions",
        data=json.dumps(base_req).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(r, timeout=30) as resp:
        full = json.loads(resp.read())["choices"][0]["message"]["content"]
    pieces = full.split()
    assert len(pieces) >= 3
    stop_piece = pieces[2]
    r2 = urllib.request.Request(
        stack["backend"] + "/v1/chat/completions",
        data=json.dumps({**base_req, "stop": stop_piece}).encode(),
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(r2, timeout=30) as resp:
        out = json.loads(resp.read())
    ch = out["choices"][0]
    assert ch["finish_reason"] == "stop"
    assert stop_piece not in ch["message"]["content"]
    # truncation happens at the FIRST occurrence of the stop string
    # anywhere in the decoded text (OpenAI substring semantics)
    assert ch["message"]["content"] == full[:full.index(stop_piece)]


def test_responses_instructions_and_model_detail(stack):
    body = json.dumps({"model": "llama-tiny", "max_tokens": 3,
                       "instructions": "be terse",
                       "input": "hello"}).encode()
    r = urllib.request.Request(stack["backend"] + "/v1/responses",
                               data=body,
                               headers={"Content-Type":
                                        "application/json"})
    with urllib.request.urlopen(r, timeout=30) as resp:
        out = json.loads(resp.read())
    assert out["object"] == "response"
    # instructions change the prompt -> (usually) different tokens than
    # without; at minimum the call succeeds and usage reflects a longer
    # prompt
    assert out["usage"]["input_tokens"] >= 3
    with urllib.request.urlopen(
            stack["backend"] + "/v1/models/llama-tiny", timeout=10) as m:
        detail = json.loads(m.read())
    assert detail["id"] == "llama-tiny"


def test_completions_echo(stack):
    body = json.dumps({"model": "llama-tiny", "prompt": "abc def",
                       "max_tokens": 3, "echo": True}).encode()
    r = urllib.request.Request(
        stack["backend"] + "/v1/completions", data=body,
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(r, timeout=30) as resp:
        out = json.loads(resp.read())
    assert out["choices"][0]["text"].startswith("abc def")
