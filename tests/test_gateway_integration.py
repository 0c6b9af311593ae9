"""Integration tests: real llmlb-gw + llmlb-mock binaries over loopback
(the reference tests the same flows with wiremock backends).
CPU-only; binaries are built by __graft_entry__.build() / make.
"""
import json
import os
import socket
import subprocess
import tempfile
import time
import urllib.error
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


def _req(method, url, body=None, headers=None, timeout=10):
    data = json.dumps(body).encode() if isinstance(body, (dict, list)) \
        else (body.encode() if isinstance(body, str) else body)
    r = urllib.request.Request(url, data=data, method=method,
                               headers=headers or {})
    try:
        with urllib.request.urlopen(r, timeout=timeout) as resp:
            return resp.status, resp.read().decode()
    except urllib.error.HTTPError as e:
        return e.code, e.read().decode()


@pytest.fixture(scope="module")
def stack():
    """mock backend + gateway with an admin token and one API key."""
    if not (BIN / "llmlb-gw").exists():
        subprocess.run(["make", "-C", str(REPO / "gateway"), "-j", "4"],
                       check=True, stdout=subprocess.DEVNULL)
    mock_port, gw_port, cloud_port = _free_port(), _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "mock-model,bge-embed"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    cloud = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(cloud_port), "--workers",
         "1", "--models", "gpt-test"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({
        "LLMLB_PORT": str(gw_port),
        "LLMLB_DATABASE_URL": db,
        "LLMLB_ADMIN_TOKEN": "admintok",
        "LLMLB_HEALTH_CHECK_INTERVAL": "2",
        "LLMLB_OPENAI_BASE_URL": f"http://127.0.0.1:{cloud_port}",
        "OPENAI_API_KEY": "sk-cloudkey",
    })
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "2"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    admin = {"Authorization": "Bearer admintok"}
    # wait for the gateway then register the mock
    for _ in range(50):
        try:
            _req("GET", base + "/health", timeout=2)
            break
        except Exception:
            time.sleep(0.1)
    status, _ = _req("POST", base + "/api/endpoints",
                     {"name": "mock1",
                      "base_url": f"http://127.0.0.1:{mock_port}"},
                     headers=admin)
    assert status == 201, status
    # create an inference API key (switches off open mode)
    status, body = _req("POST", base + "/api/api-keys",
                        {"name": "t", "permissions": "inference"},
                        headers=admin)
    key = json.loads(body)["api_key"]
    yield {"base": base, "admin": admin,
           "auth": {"Authorization": f"Bearer {key}",
                    "Content-Type": "application/json"},
           "mock_port": mock_port, "gw": gw}
    gw.terminate()
    mock.terminate()
    cloud.terminate()
    try:
        gw.wait(timeout=5)
    except subprocess.TimeoutExpired:
        gw.kill()
    for p in Path("/tmp").glob(Path(db).name + "*"):
        p.unlink(missing_ok=True)


def test_models_list(stack):
    status, body = _req("GET", stack["base"] + "/v1/models",
                        headers=stack["auth"])
    assert status == 200
    ids = [m["id"] for m in json.loads(body)["data"]]
    assert "mock-model" in ids and "bge-embed" in ids


def test_chat_roundtrip(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/chat/completions",
        {"model": "mock-model",
         "messages": [{"role": "user", "content": "hi"}]},
        headers=stack["auth"])
    assert status == 200, body
    obj = json.loads(body)
    assert obj["model"] == "mock-model"
    assert obj["usage"]["completion_tokens"] == 8
    assert obj["choices"][0]["message"]["content"]


def test_auth_required(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/chat/completions",
        {"model": "mock-model", "messages": []})
    assert status == 401
    assert "invalid_api_key" in body


def test_unknown_model_404(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/chat/completions",
        {"model": "no-such-model", "messages": []},
        headers=stack["auth"])
    assert status == 404
    assert "model_not_found" in body


def test_streaming_sse_passthrough(stack):
    req = urllib.request.Request(
        stack["base"] + "/v1/chat/completions",
        data=json.dumps({"model": "mock-model", "stream": True,
                         "messages": []}).encode(),
        headers=stack["auth"])
    with urllib.request.urlopen(req, timeout=10) as resp:
        assert resp.status == 200
        assert "text/event-stream" in resp.headers.get("Content-Type", "")
        raw = resp.read().decode()
    assert raw.count("data:") >= 9
    assert "[DONE]" in raw
    assert '"tok0 "' in raw


def test_embeddings(stack):
    status, body = _req("POST", stack["base"] + "/v1/embeddings",
                        {"model": "bge-embed", "input": "hello"},
                        headers=stack["auth"])
    assert status == 200, body
    assert json.loads(body)["data"][0]["embedding"]


def test_responses_api(stack):
    status, body = _req("POST", stack["base"] + "/v1/responses",
                        {"model": "mock-model", "input": "hello"},
                        headers=stack["auth"])
    assert status == 200, body


def test_anthropic_messages_nonstream(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/messages",
        {"model": "mock-model", "max_tokens": 32,
         "messages": [{"role": "user", "content": "hi"}]},
        headers={**stack["auth"], "anthropic-version": "2023-06-01"})
    assert status == 200, body
    obj = json.loads(body)
    assert obj["type"] == "message"
    assert obj["role"] == "assistant"
    assert obj["content"][0]["type"] == "text"
    assert obj["stop_reason"] == "end_turn"
    assert obj["usage"]["output_tokens"] == 8


def test_anthropic_messages_requires_version_header(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/messages",
        {"model": "mock-model", "max_tokens": 32, "messages": []},
        headers=stack["auth"])
    assert status == 400
    assert "anthropic-version" in body


def test_anthropic_messages_stream_reencoded(stack):
    req = urllib.request.Request(
        stack["base"] + "/v1/messages",
        data=json.dumps({"model": "mock-model", "max_tokens": 32,
                         "stream": True,
                         "messages": [{"role": "user",
                                       "content": "hi"}]}).encode(),
        headers={**stack["auth"], "anthropic-version": "2023-06-01"})
    with urllib.request.urlopen(req, timeout=10) as resp:
        raw = resp.read().decode()
    # re-encoded Anthropic event stream, in order
    for ev in ["message_start", "content_block_start",
               "content_block_delta", "content_block_stop",
               "message_delta", "message_stop"]:
        assert f"event: {ev}" in raw, raw[:800]
    assert raw.index("message_start") < raw.index("content_block_delta") \
        < raw.index("message_stop")
    assert '"text_delta"' in raw


def test_cloud_prefix_openai(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/chat/completions",
        {"model": "openai:gpt-test",
         "messages": [{"role": "user", "content": "hi"}]},
        headers=stack["auth"])
    assert status == 200, body
    # model rewritten back? cloud providers echo the bare name; the
    # gateway passes the body through with the prefix stripped upstream
    obj = json.loads(body)
    assert obj["choices"][0]["message"]["content"]


def test_cloud_prefix_unconfigured_provider(stack):
    status, body = _req(
        "POST", stack["base"] + "/v1/chat/completions",
        {"model": "google:gemini-x", "messages": []},
        headers=stack["auth"])
    assert status == 502
    assert "cloud_provider_unconfigured" in body


def test_audio_capability_routing_503_when_absent(stack):
    status, body = _req("POST", stack["base"] + "/v1/audio/speech",
                        {"model": "tts-1", "input": "hello"},
                        headers=stack["auth"])
    # the mock's models don't advertise audio capability
    assert status == 503
    assert "no_capable_endpoint" in body


def test_request_id_in_history(stack):
    """The minted/echoed request id lands in /api/requests records."""
    body = {"model": "mock-model",
            "messages": [{"role": "user", "content": "hi"}]}
    r = urllib.request.Request(
        stack["base"] + "/v1/chat/completions",
        data=json.dumps(body).encode(),
        headers={**stack["auth"], "X-Request-Id": "hist-trace-7"})
    urllib.request.urlopen(r, timeout=10).read()
    deadline = time.time() + 5
    while time.time() < deadline:
        _, recs = _req("GET", stack["base"] + "/api/requests",
                       headers=stack["admin"])
        if any(rec.get("request_id") == "hist-trace-7"
               for rec in json.loads(recs)):
            return
        time.sleep(0.2)
    raise AssertionError("request id not recorded")


def test_request_history_recorded(stack):
    _req("POST", stack["base"] + "/v1/chat/completions",
         {"model": "mock-model", "messages": []}, headers=stack["auth"])
    time.sleep(0.5)  # batched flush
    status, body = _req("GET", stack["base"] + "/api/requests",
                        headers=stack["admin"])
    assert status == 200
    recs = json.loads(body)
    assert any(r["model"] == "mock-model" and r["status"] == 200
               for r in recs)


def test_dashboard_summary(stack):
    status, body = _req("GET", stack["base"] + "/api/dashboard/summary",
                        headers=stack["admin"])
    assert status == 200
    obj = json.loads(body)
    assert obj["endpoints"][0]["status"] == "online"
    assert obj["last_hour"]["requests"] >= 1


def test_prometheus_per_endpoint_series(stack):
    _, text = _req("GET", stack["base"] + "/metrics")
    assert 'llmlb_endpoint_online{endpoint="mock1"} 1' in text
    assert 'llmlb_endpoint_requests_total{endpoint="mock1"}' in text


def test_prometheus_metrics(stack):
    status, body = _req("GET", stack["base"] + "/api/metrics")
    assert status == 200
    assert "llmlb_requests_total" in body


def test_audit_chain_verifies(stack):
    time.sleep(0.6)
    status, body = _req("GET", stack["base"] + "/api/audit/verify",
                        headers=stack["admin"])
    assert status == 200
    assert json.loads(body)["valid"] is True


def test_drain_gate(stack):
    _req("POST", stack["base"] + "/api/system/drain", {},
         headers=stack["admin"])
    status, body = _req("POST", stack["base"] + "/v1/chat/completions",
                        {"model": "mock-model", "messages": []},
                        headers=stack["auth"])
    assert status == 503 and "draining" in body
    _req("POST", stack["base"] + "/api/system/resume", {},
         headers=stack["admin"])
    status, _ = _req("POST", stack["base"] + "/v1/chat/completions",
                     {"model": "mock-model", "messages": []},
                     headers=stack["auth"])
    assert status == 200


def test_endpoint_failure_ejected_and_recovers(stack):
    """Register an endpoint that dies -> health marks it offline and
    routing stops using it; its model 404s (config 5 behavior)."""
    port = _free_port()
    victim = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(port), "--workers", "1",
         "--models", "ephemeral-model"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    for _ in range(50):  # wait for the victim to accept connections
        try:
            socket.create_connection(("127.0.0.1", port), 0.2).close()
            break
        except OSError:
            time.sleep(0.1)
    status, body = _req("POST", stack["base"] + "/api/endpoints",
                        {"name": "victim",
                         "base_url": f"http://127.0.0.1:{port}"},
                        headers=stack["admin"])
    assert status == 201
    ep_id = json.loads(body)["id"]
    status, _ = _req("POST", stack["base"] + "/v1/chat/completions",
                     {"model": "ephemeral-model", "messages": []},
                     headers=stack["auth"])
    assert status == 200
    victim.terminate()
    victim.wait()
    # 2-fail hysteresis at 2 s interval -> offline within ~8 s
    deadline = time.time() + 15
    while time.time() < deadline:
        status, _ = _req("POST", stack["base"] + "/v1/chat/completions",
                         {"model": "ephemeral-model", "messages": []},
                         headers=stack["auth"])
        if status in (404, 503):
            break
        time.sleep(1)
    assert status in (404, 503), f"endpoint never ejected (last {status})"
    _req("DELETE", stack["base"] + f"/api/endpoints/{ep_id}",
         headers=stack["admin"])


def test_expect_100_continue(stack):
    """curl sends Expect: 100-continue for large bodies and stalls ~1s
    unless the server acknowledges with an interim 100 response."""
    host, port = stack["base"].replace("http://", "").split(":")
    body = json.dumps({"model": "mock-model",
                       "messages": [{"role": "user",
                                     "content": "x" * 2048}]}).encode()
    s = socket.create_connection((host, int(port)), timeout=10)
    try:
        head = (f"POST /v1/chat/completions HTTP/1.1\r\n"
                f"Host: {host}\r\n"
                f"Authorization: {stack['auth']['Authorization']}\r\n"
                f"Content-Type: application/json\r\n"
                f"Expect: 100-continue\r\n"
                f"Content-Length: {len(body)}\r\n\r\n").encode()
        s.sendall(head)  # body deliberately withheld
        s.settimeout(5)
        interim = s.recv(4096)
        assert interim.startswith(b"HTTP/1.1 100 Continue"), interim[:60]
        s.sendall(body)
        buf = b""
        while b"\"chat.completion\"" not in buf:
            chunk = s.recv(65536)
            assert chunk, buf[:200]
            buf += chunk
        assert b" 200 " in buf.split(b"\r\n", 1)[0]
    finally:
        s.close()


def test_dashboard_spa_served(stack):
    """The gateway serves the embedded single-file dashboard at / and
    /dashboard (reference: include_dir-embedded React SPA)."""
    for path in ("/", "/dashboard"):
        status, body = _req("GET", stack["base"] + path)
        assert status == 200
        assert "<html" in body and "llmlb" in body
    # the APIs the page calls stay token-gated
    status, _ = _req("GET", stack["base"] + "/api/dashboard/summary")
    assert status in (401, 403)


def test_request_id_tracing(stack):
    """Every response carries X-Request-Id: echoed when the client
    supplies one, minted otherwise; streamed responses included."""
    body = {"model": "mock-model",
            "messages": [{"role": "user", "content": "hi"}]}
    r = urllib.request.Request(
        stack["base"] + "/v1/chat/completions",
        data=json.dumps(body).encode(),
        headers={**stack["auth"], "X-Request-Id": "trace-me-42"})
    with urllib.request.urlopen(r, timeout=10) as resp:
        assert resp.headers["X-Request-Id"] == "trace-me-42"
    r2 = urllib.request.Request(
        stack["base"] + "/v1/chat/completions",
        data=json.dumps({**body, "stream": True}).encode(),
        headers=stack["auth"])
    with urllib.request.urlopen(r2, timeout=10) as resp:
        rid = resp.headers["X-Request-Id"]
        assert rid and rid.startswith("req_")
        resp.read()
    # local API responses carry one too
    r3 = urllib.request.Request(stack["base"] + "/v1/models",
                                headers=stack["auth"])
    with urllib.request.urlopen(r3, timeout=10) as resp:
        assert resp.headers["X-Request-Id"]


def test_inference_timeout_sweep():
    """An endpoint whose backend stalls past inference_timeout_secs
    gets its request killed by the deadline sweep (504) instead of
    hanging the client."""
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    slow = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "slow-model", "--delay-ms", "30000",
         "--delay-inference-only"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    admin = {"Authorization": "Bearer t"}
    try:
        for _ in range(50):
            try:
                _req("GET", base + "/health", timeout=2)
                break
            except Exception:
                time.sleep(0.1)
        status, body = _req("POST", base + "/api/endpoints",
                            {"name": "slow",
                             "base_url": f"http://127.0.0.1:{mock_port}"},
                            headers=admin)
        assert status == 201, body
        ep_id = json.loads(body)["id"]
        status, _ = _req("PUT", base + f"/api/endpoints/{ep_id}",
                         {"inference_timeout_secs": 1}, headers=admin)
        assert status == 200
        # wait for model sync (the slow mock delays /v1/models too)
        deadline = time.time() + 30
        while time.time() < deadline:
            _, models = _req("GET", base + "/v1/models")
            if "slow-model" in models:
                break
            time.sleep(0.5)
        t0 = time.time()
        status, body = _req("POST", base + "/v1/chat/completions",
                            {"model": "slow-model",
                             "messages": [{"role": "user",
                                           "content": "hi"}]},
                            timeout=20)
        dt = time.time() - t0
        assert status == 504, (status, body)
        assert dt < 8, dt
    finally:
        gw.terminate()
        slow.terminate()
        for p in (gw, slow):
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


def test_admission_429_when_saturated():
    """LLMLB_QUEUE_MAX=2: with both slots held by stalled requests, a
    third gets a fast 429 (capacity) instead of queueing forever."""
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    slow = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "busy-model", "--delay-ms", "8000",
         "--delay-inference-only"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t", "LLMLB_QUEUE_MAX": "2",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    try:
        for _ in range(50):
            try:
                _req("GET", base + "/health", timeout=2)
                break
            except Exception:
                time.sleep(0.1)
        status, _ = _req("POST", base + "/api/endpoints",
                         {"name": "busy",
                          "base_url": f"http://127.0.0.1:{mock_port}"},
                         headers={"Authorization": "Bearer t"})
        assert status == 201
        import threading
        body = {"model": "busy-model",
                "messages": [{"role": "user", "content": "hi"}]}
        holders = [threading.Thread(
            target=lambda: _req("POST", base + "/v1/chat/completions",
                                body, timeout=30)) for _ in range(2)]
        for h in holders:
            h.start()
        time.sleep(1.0)   # both slots occupied (0.8 * 2 = admission cap)
        t0 = time.time()
        status, resp = _req("POST", base + "/v1/chat/completions", body,
                            timeout=10)
        dt = time.time() - t0
        assert status == 429, (status, resp)
        assert dt < 3, dt
        for h in holders:
            h.join(timeout=30)
    finally:
        gw.terminate()
        slow.terminate()
        for p in (gw, slow):
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


def test_endpoint_persistence_across_restart():
    """Registered endpoints survive a gateway restart (SQLite-backed
    registry reload + health re-check brings them back online)."""
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "persist-model"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "2"})

    def start_gw():
        return subprocess.Popen(
            [str(BIN / "llmlb-gw"), "--workers", "1"], env=env,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)

    def wait_health(timeout=15):
        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                _req("GET", f"http://127.0.0.1:{gw_port}/health",
                     timeout=2)
                return True
            except Exception:
                time.sleep(0.1)
        return False

    gw = start_gw()
    base = f"http://127.0.0.1:{gw_port}"
    try:
        assert wait_health()
        status, _ = _req("POST", base + "/api/endpoints",
                         {"name": "p1",
                          "base_url": f"http://127.0.0.1:{mock_port}"},
                         headers={"Authorization": "Bearer t"})
        assert status == 201
        status, _ = _req("POST", base + "/v1/chat/completions",
                         {"model": "persist-model",
                          "messages": [{"role": "user",
                                        "content": "x"}]})
        assert status == 200
        gw.terminate()
        gw.wait(timeout=10)
        gw = start_gw()
        assert wait_health()
        deadline = time.time() + 20
        ok = False
        while time.time() < deadline:
            status, _ = _req("POST", base + "/v1/chat/completions",
                             {"model": "persist-model",
                              "messages": [{"role": "user",
                                            "content": "x"}]})
            if status == 200:
                ok = True
                break
            time.sleep(0.5)
        assert ok, f"model not routable after restart (last {status})"
        # boot reseed (bootstrap.rs:127-159 analog): the restarted
        # gateway rebuilds the 60-min history ring from SQLite and
        # seeds TPS EMAs from today's daily stats — the summary shows
        # pre-restart traffic immediately and selections are not
        # score-blind
        status, body = _req("GET", base + "/api/dashboard/summary",
                            headers={"Authorization": "Bearer t"})
        assert status == 200
        summary = json.loads(body)
        assert summary["last_hour"]["requests"] >= 1, \
            "history ring not reseeded after restart"
        found_tps = False
        for ep in summary["endpoints"]:
            for m in ep.get("models", []):
                if m.get("tokens_per_sec"):
                    found_tps = True
        assert found_tps, "TPS EMA not reseeded from daily stats"
    finally:
        gw.terminate()
        mock.terminate()
        for p in (gw, mock):
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


def test_cloud_prefix_typo_alias(stack):
    """`ahtnorpic:` is the reference's documented typo-alias for
    anthropic (api/openai.rs:637-655); unconfigured here -> 502-class
    error naming the anthropic provider, not a 404."""
    status, body = _req("POST", stack["base"] + "/v1/chat/completions",
                        {"model": "ahtnorpic:claude-3-5-haiku",
                         "messages": [{"role": "user",
                                       "content": "hi"}]},
                        headers=stack["auth"])
    assert status in (400, 502, 503), (status, body)
    assert "anthropic" in body.lower()


def test_garbage_storm_survival(stack):
    """Random garbage, truncated heads, oversized lines and abrupt
    closes must not crash or wedge the gateway."""
    import random
    host, port = stack["base"].replace("http://", "").split(":")
    rng = random.Random(7)
    payloads = [
        b"\x00\xff\xfe\x01" * 64,
        b"GET / HTTP/9.9\r\n\r\n",
        b"POST /v1/chat/completions HTTP/1.1\r\nContent-Length: "
        b"99999999999999999999\r\n\r\n",
        b"A" * 70000,
        b"GET " + b"/x" * 8000 + b" HTTP/1.1\r\n\r\n",
        b"POST /v1/chat/completions HTTP/1.1\r\nTransfer-Encoding: "
        b"chunked\r\n\r\nzzz\r\n",
    ]
    for _ in range(16):
        p = rng.choice(payloads)
        try:
            s = socket.create_connection((host, int(port)), timeout=5)
            s.sendall(p[:rng.randrange(1, len(p) + 1)])
            if rng.random() < 0.5:
                s.close()
            else:
                s.settimeout(0.2)
                try:
                    s.recv(4096)
                except OSError:
                    pass
                s.close()
        except OSError:
            pass
    # gateway still serves normal traffic afterwards
    status, _ = _req("POST", stack["base"] + "/v1/chat/completions",
                     {"model": "mock-model",
                      "messages": [{"role": "user", "content": "ok"}]},
                     headers=stack["auth"])
    assert status == 200


def test_capture_bodies_flag():
    """LLMLB_CAPTURE_BODIES=1 persists truncated request/response
    bodies into request_history (off by default for throughput)."""
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "cap-model"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t", "LLMLB_CAPTURE_BODIES": "1",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    try:
        for _ in range(50):
            try:
                _req("GET", base + "/health", timeout=2)
                break
            except Exception:
                time.sleep(0.1)
        _req("POST", base + "/api/endpoints",
             {"name": "c", "base_url": f"http://127.0.0.1:{mock_port}"},
             headers={"Authorization": "Bearer t"})
        _req("POST", base + "/v1/chat/completions",
             {"model": "cap-model",
              "messages": [{"role": "user",
                            "content": "capture-me-marker"}]})
        import sqlite3 as sq
        deadline = time.time() + 10
        found = False
        while time.time() < deadline and not found:
            try:
                conn = sq.connect(db)
                row = conn.execute(
                    "SELECT request_body, response_body FROM "
                    "request_history ORDER BY id DESC LIMIT 1"
                ).fetchone()
                conn.close()
                if row and row[0] and "capture-me-marker" in row[0] \
                        and row[1]:
                    found = True
                    break
            except Exception:
                pass
            time.sleep(0.5)
        assert found
    finally:
        gw.terminate()
        mock.terminate()
        for p in (gw, mock):
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


def test_cloud_prefix_google(stack):
    """google: prefix proxies to the configured Google-compatible base
    (OpenAI-compat endpoint in this environment) with model rewrite."""
    # the module fixture only configures the openai provider; a google:
    # model against this stack reports the provider as unconfigured
    status, body = _req("POST", stack["base"] + "/v1/chat/completions",
                        {"model": "google:gemini-pro",
                         "messages": [{"role": "user",
                                       "content": "hi"}]},
                        headers=stack["auth"])
    assert status == 502
    assert "google" in body.lower()
    assert "LLMLB_GOOGLE_BASE_URL".lower() in body.lower()


def test_anthropic_unicode_roundtrip(stack):
    """Multibyte content survives request conversion + response
    re-encode (the mock echoes a fixed reply; the conversion path must
    not mangle or reject the UTF-8 request)."""
    body = {"model": "mock-model", "max_tokens": 16,
            "messages": [{"role": "user",
                          "content": "héllo 世界 🚀 \"quoted\" \\ backslash"}]}
    r = urllib.request.Request(
        stack["base"] + "/v1/messages",
        data=json.dumps(body, ensure_ascii=False).encode("utf-8"),
        headers={**stack["auth"],
                 "anthropic-version": "2023-06-01"})
    with urllib.request.urlopen(r, timeout=10) as resp:
        out = json.loads(resp.read())
    assert out["type"] == "message"
    assert out["content"][0]["type"] == "text"
    assert out["usage"]["input_tokens"] > 0


def test_delayed_admission_between_50_and_80pct():
    """LLMLB_QUEUE_MAX=4 with 2 slots held (50%): the next request is
    HELD 10-100 ms on the event loop and then served (reference
    admission_control AcceptWithDelay), visible both in its latency
    and in the llmlb_admission_delays_total counter."""
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    slow = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "dm", "--delay-ms", "4000",
         "--delay-inference-only"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t", "LLMLB_QUEUE_MAX": "4",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    try:
        for _ in range(50):
            try:
                _req("GET", base + "/health", timeout=2)
                break
            except Exception:
                time.sleep(0.1)
        status, _ = _req("POST", base + "/api/endpoints",
                         {"name": "d",
                          "base_url": f"http://127.0.0.1:{mock_port}"},
                         headers={"Authorization": "Bearer t"})
        assert status == 201
        import threading
        body = {"model": "dm",
                "messages": [{"role": "user", "content": "hi"}]}
        holders = [threading.Thread(
            target=lambda: _req("POST", base + "/v1/chat/completions",
                                body, timeout=30)) for _ in range(2)]
        for h in holders:
            h.start()
        time.sleep(1.0)   # 2/4 active = 50% -> delay band
        r = urllib.request.Request(
            base + "/v1/chat/completions",
            data=json.dumps(body).encode(), method="POST")
        with urllib.request.urlopen(r, timeout=30) as resp:
            status = resp.status
            delay_hdr = resp.headers.get("X-Admission-Delay-Ms")
        assert status == 200
        assert delay_hdr and 10 <= int(delay_hdr) <= 100, delay_hdr
        status, text = _req("GET", base + "/api/metrics")
        delays = [ln for ln in text.splitlines()
                  if ln.startswith("llmlb_admission_delays_total")]
        assert delays and int(delays[0].split()[-1]) >= 1, delays
        for h in holders:
            h.join(timeout=30)
    finally:
        gw.terminate()
        slow.terminate()
        for p in (gw, slow):
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


def test_connect_failure_retries_next_endpoint():
    """A killed backend whose TPS score still tops selection must not
    surface errors: the gateway retries the request once against the
    next-best endpoint (deliberate improvement over the reference,
    which 502s until the health eject)."""
    m1, m2, gw_port = _free_port(), _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    fast = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(m1), "--workers", "1",
         "--models", "ha-model"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    backup = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(m2), "--workers", "1",
         "--models", "ha-model"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    try:
        for _ in range(50):
            try:
                _req("GET", base + "/health", timeout=2)
                break
            except Exception:
                time.sleep(0.1)
        for name, port in (("fast", m1), ("backup", m2)):
            status, _ = _req("POST", base + "/api/endpoints",
                             {"name": name,
                              "base_url": f"http://127.0.0.1:{port}"},
                             headers={"Authorization": "Bearer t"})
            assert status == 201
        body = {"model": "ha-model",
                "messages": [{"role": "user", "content": "x"}]}
        # earn the first endpoint a TPS score so it keeps winning
        for _ in range(6):
            assert _req("POST", base + "/v1/chat/completions",
                        body)[0] == 200
        fast.terminate()
        fast.wait(timeout=5)
        # health interval is 60 s: no eject will save us — every
        # request must succeed via the transparent connect retry
        for i in range(10):
            status, resp = _req("POST", base + "/v1/chat/completions",
                                body, timeout=10)
            assert status == 200, (i, status, resp)
        # streaming too
        status, resp = _req("POST", base + "/v1/chat/completions",
                            dict(body, stream=True), timeout=10)
        assert status == 200
    finally:
        gw.terminate()
        backup.terminate()
        for p in (gw, fast, backup):
            try:
                p.wait(timeout=5)
            except Exception:
                pass


def test_pipelined_request_does_not_dangle_head_views(stack):
    """Regression for the r2 use-after-free: a pipelined second request
    reallocates the connection's input buffer while an upstream op is
    in flight; the audit path then read the FIRST request's parsed
    method/path string_views (which pointed into the old buffer).
    Fixed with owned per-request copies (gateway.cpp cur_method/path).
    This drives the exact sequence: slow upstream -> op in flight ->
    large pipelined request forces the realloc -> first response's
    audit runs after it.  Under an ASAN build this test crashes the
    old gateway; on a normal build it asserts both responses and a
    live gateway (profiles/r02_gateway_uaf_fix.md)."""
    import re as _re
    base = stack["base"]
    host, port = base.replace("http://", "").split(":")
    # slow mock: 400 ms inference latency keeps the op window open
    slow_port = _free_port()
    slow = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(slow_port), "--workers",
         "1", "--models", "uaf-model", "--delay-ms", "400",
         "--delay-inference-only"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        _req("POST", base + "/api/endpoints",
             {"name": "uafslow", "base_url": f"http://127.0.0.1:{slow_port}"},
             headers=stack["admin"])
        deadline = time.time() + 15
        while time.time() < deadline:
            st, body = _req("GET", base + "/v1/models",
                            headers=stack["auth"])
            if st == 200 and "uaf-model" in body:
                break
            time.sleep(0.3)
        assert "uaf-model" in body

        key = stack["auth"]["Authorization"]
        def chat_req(model, pad):
            payload = json.dumps({
                "model": model,
                "messages": [{"role": "user", "content": "x" * pad}]})
            return (f"POST /v1/chat/completions HTTP/1.1\r\n"
                    f"Host: {host}\r\nAuthorization: {key}\r\n"
                    f"Content-Type: application/json\r\n"
                    f"Content-Length: {len(payload)}\r\n\r\n"
                    ).encode() + payload.encode()

        c = socket.create_connection((host, int(port)), timeout=10)
        try:
            # request A: slow upstream -> async op stays in flight
            c.sendall(chat_req("uaf-model", 8))
            time.sleep(0.1)          # let the gateway dispatch A
            # request B: ~200 KB body pipelined on the same conn ->
            # guaranteed input-buffer reallocation
            c.sendall(chat_req("uaf-model", 200_000))
            c.settimeout(15)
            data = b""
            while data.count(b"HTTP/1.1") < 2 or not data.endswith(b"}"):
                chunk = c.recv(65536)
                if not chunk:
                    break
                data = data + chunk
            statuses = _re.findall(rb"HTTP/1\.1 (\d+)", data)
            assert len(statuses) == 2, data[:200]
            assert statuses[0] == b"200", data[:200]
        finally:
            c.close()
        # the gateway survived and still serves
        st, _ = _req("GET", base + "/health")
        assert st == 200
    finally:
        slow.terminate()


def _mock1_id(stack):
    _, body = _req("GET", stack["base"] + "/api/endpoints",
                   headers=stack["admin"])
    eps = json.loads(body)
    # the module-scoped stack accumulates endpoints from earlier
    # tests; the traffic-bearing one is the original "mock1"
    return next(e["id"] for e in eps if e["name"] == "mock1")


def test_per_endpoint_stats_routes(stack):
    """Per-endpoint statistics family (reference api/dashboard.rs:919-976):
    today / daily-stats / model-stats on one endpoint, plus the
    cross-endpoint /api/dashboard/all-model-stats rollup."""
    _req("POST", stack["base"] + "/v1/chat/completions",
         {"model": "mock-model", "messages": []}, headers=stack["auth"])
    time.sleep(0.6)  # batched daily-stats flush
    base = stack["base"] + "/api/endpoints/" + _mock1_id(stack)

    status, body = _req("GET", base + "/today", headers=stack["admin"])
    assert status == 200
    today = json.loads(body)
    assert today["total_requests"] >= 1
    assert today["total_requests"] == (today["successful_requests"]
                                       + today["failed_requests"])

    status, body = _req("GET", base + "/daily-stats?days=3",
                        headers=stack["admin"])
    assert status == 200
    days = json.loads(body)
    assert days and days[0]["date"] == today["date"]
    assert days[0]["total_requests"] >= 1

    status, body = _req("GET", base + "/model-stats",
                        headers=stack["admin"])
    assert status == 200
    ms = json.loads(body)
    row = next(m for m in ms if m["model_id"] == "mock-model")
    assert row["total_requests"] >= 1
    assert "total_output_tokens" in row and "total_duration_ms" in row

    status, body = _req("GET",
                        stack["base"] + "/api/dashboard/all-model-stats",
                        headers=stack["admin"])
    assert status == 200
    allms = json.loads(body)
    assert any(m["model_id"] == "mock-model"
               and m["total_requests"] >= row["total_requests"]
               for m in allms)


def test_endpoint_model_tps_route(stack):
    """Live per-model TPS EMAs for one endpoint
    (reference api/dashboard.rs:1206-1225)."""
    _req("POST", stack["base"] + "/v1/chat/completions",
         {"model": "mock-model", "messages": []}, headers=stack["auth"])
    time.sleep(0.6)
    status, body = _req("GET",
                        stack["base"] + "/api/endpoints/"
                        + _mock1_id(stack) + "/model-tps",
                        headers=stack["admin"])
    assert status == 200
    entries = json.loads(body)
    row = next(e for e in entries if e["model_id"] == "mock-model")
    assert row["tps"] > 0 and row["request_count"] >= 1
    assert row["source"] == "Production"
    assert row["api_kind"] == "chat_completions"
    assert "average_duration_ms" in row


def test_node_metrics_history_route(stack):
    """Per-endpoint health-metrics history (reference
    api/dashboard.rs:205-211): the health loop's persisted probes."""
    url = (stack["base"] + "/api/dashboard/metrics/"
           + _mock1_id(stack))
    deadline = time.time() + 8
    while time.time() < deadline:
        status, body = _req("GET", url, headers=stack["admin"])
        assert status == 200
        rows = json.loads(body)
        if rows:
            assert rows[0]["ok"] is True
            assert rows[0]["latency_ms"] >= 0
            return
        time.sleep(0.5)
    raise AssertionError("no health_checks rows recorded")


def test_request_detail_route(stack):
    """Request-history detail by row id or X-Request-Id (reference
    api/dashboard.rs:692-705)."""
    _req("POST", stack["base"] + "/v1/chat/completions",
         {"model": "mock-model", "messages": []},
         headers={**stack["auth"], "X-Request-Id": "detail-route-1"})
    time.sleep(0.6)
    status, body = _req("GET",
                        stack["base"] + "/api/requests/detail-route-1",
                        headers=stack["admin"])
    assert status == 200
    rec = json.loads(body)
    assert rec["request_id"] == "detail-route-1"
    assert rec["model"] == "mock-model" and rec["status"] == 200
    # numeric row id resolves to the same record
    status, body = _req("GET",
                        stack["base"] + "/api/requests/" + str(rec["id"]),
                        headers=stack["admin"])
    assert status == 200
    assert json.loads(body)["request_id"] == "detail-route-1"
    # unknown id -> 404
    status, _ = _req("GET", stack["base"] + "/api/requests/nope-404",
                     headers=stack["admin"])
    assert status == 404


def test_requests_list_filters(stack):
    """Request-history list filters model=/status=/limit= (reference
    api/dashboard.rs:661-689 RequestHistoryQuery)."""
    _req("POST", stack["base"] + "/v1/chat/completions",
         {"model": "mock-model", "messages": []}, headers=stack["auth"])
    _req("POST", stack["base"] + "/v1/chat/completions",
         {"model": "ghost-model", "messages": []}, headers=stack["auth"])
    time.sleep(0.3)
    status, body = _req("GET",
                        stack["base"] + "/api/requests?model=mock-model",
                        headers=stack["admin"])
    assert status == 200
    recs = json.loads(body)
    assert recs and all(r["model"] == "mock-model" for r in recs)

    status, body = _req("GET", stack["base"] + "/api/requests?status=200",
                        headers=stack["admin"])
    recs = json.loads(body)
    assert recs and all(r["status"] == 200 for r in recs)
    status, body = _req("GET",
                        stack["base"] + "/api/requests?model=no-such",
                        headers=stack["admin"])
    assert json.loads(body) == []

    status, body = _req("GET",
                        stack["base"]
                        + "/api/requests?model=mock-model&limit=1",
                        headers=stack["admin"])
    recs = json.loads(body)
    assert len(recs) == 1 and recs[0]["model"] == "mock-model"
    # the newest matching record is kept, not the oldest
    status, body = _req("GET",
                        stack["base"] + "/api/requests?model=mock-model",
                        headers=stack["admin"])
    assert recs[0]["ts_ms"] == json.loads(body)[-1]["ts_ms"]
