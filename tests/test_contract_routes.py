"""Contract sweep: every data-plane route answers with its documented
status + JSON shape (reference tests/contract/* tier).
One shared stack; each case asserts status, content type and key
fields — shapes, not values."""
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


def _req(method, url, body=None, headers=None, timeout=10, raw=False):
    data = (json.dumps(body).encode() if isinstance(body, (dict, list))
            else body)
    r = urllib.request.Request(url, data=data, method=method,
                               headers=headers or {})
    try:
        with urllib.request.urlopen(r, timeout=timeout) as resp:
            return resp.status, resp.read().decode(), dict(resp.headers)
    except urllib.error.HTTPError as e:
        return e.code, e.read().decode(), dict(e.headers)


@pytest.fixture(scope="module")
def stack():
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "cm,bge-embed"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, cwd=str(REPO),
                          stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    for _ in range(50):
        try:
            _req("GET", base + "/health")
            break
        except Exception:
            time.sleep(0.1)
    status, body, _ = _req("POST", base + "/api/endpoints",
                           {"name": "m",
                            "base_url": f"http://127.0.0.1:{mock_port}"},
                           headers={"Authorization": "Bearer t"})
    assert status == 201
    epid = json.loads(body)["id"]
    # mint a real key: CLOSES open mode so unauth cases get 401s
    status, body, _ = _req("POST", base + "/api/api-keys",
                           {"name": "contract",
                            "permissions": "inference"},
                           headers={"Authorization": "Bearer t"})
    key = json.loads(body)["api_key"]
    auth = {"Authorization": f"Bearer {key}"}
    # one request so history/analytics have data
    _req("POST", base + "/v1/chat/completions",
         {"model": "cm", "messages": [{"role": "user", "content": "x"}]},
         headers=auth)
    time.sleep(0.5)
    yield {"base": base, "epid": epid, "auth": auth,
           "admin": {"Authorization": "Bearer t"}}
    gw.terminate()
    mock.terminate()
    for p in (gw, mock):
        try:
            p.wait(timeout=5)
        except subprocess.TimeoutExpired:
            p.kill()


ADMIN = object()  # sentinel: needs admin header
KEY = object()    # sentinel: needs the inference API key

CASES = [
    # (method, path, body, auth?, expected status, required JSON keys)
    ("GET", "/health", None, None, 200, {"status", "version"}),
    ("GET", "/api/health", None, None, 200, {"status"}),
    ("GET", "/v1/models", None, KEY, 200, {"object", "data"}),
    ("GET", "/v1/models/cm", None, KEY, 200, {"id", "object"}),
    ("GET", "/v1/models/nope", None, KEY, 404, {"error"}),
    ("POST", "/v1/chat/completions",
     {"model": "cm", "messages": []}, KEY, 200,
     {"id", "choices", "usage", "model"}),
    ("POST", "/v1/completions", {"model": "cm", "prompt": "x"}, KEY,
     200, {"choices"}),
    ("POST", "/v1/embeddings", {"model": "bge-embed", "input": "x"},
     KEY, 200, {"data"}),
    ("POST", "/v1/responses", {"model": "cm", "input": "x"}, KEY, 200,
     None),
    ("POST", "/v1/chat/completions", {"messages": []}, KEY, 400,
     {"error"}),
    ("POST", "/v1/chat/completions", {"model": "ghost",
                                      "messages": []}, KEY, 404,
     {"error"}),
    ("GET", "/api/endpoints", None, ADMIN, 200, None),
    ("GET", "/api/endpoints/nope", None, ADMIN, 404, {"error"}),
    ("GET", "/api/api-keys", None, ADMIN, 200, None),
    ("GET", "/api/dashboard/summary", None, ADMIN, 200,
     {"endpoints", "active_requests", "last_hour"}),
    ("GET", "/api/dashboard/daily-stats", None, ADMIN, 200, None),
    ("GET", "/api/dashboard/overview", None, ADMIN, 200,
     {"summary", "history", "endpoint_tps", "generated_at",
      "generation_time_ms"}),
    ("GET", "/api/dashboard/all-model-stats", None, ADMIN, 200, None),
    ("GET", "/api/dashboard/stats/tokens", None, ADMIN, 200,
     {"total_input_tokens", "total_output_tokens", "total_tokens"}),
    ("GET", "/api/dashboard/stats/tokens/daily?days=7", None, ADMIN,
     200, None),
    ("GET", "/api/dashboard/clients", None, ADMIN, 200, None),
    ("GET", "/api/dashboard/clients/heatmap", None, ADMIN, 200,
     {"requests"}),
    ("GET", "/api/dashboard/token-stats/monthly", None, ADMIN, 200,
     None),
    ("GET", "/api/requests", None, ADMIN, 200, None),
    ("GET", "/api/events", None, ADMIN, 200, None),
    ("GET", "/api/audit", None, ADMIN, 200, {"entries"}),
    ("GET", "/api/audit/verify", None, ADMIN, 200, {"valid"}),
    ("GET", "/api/settings", None, ADMIN, 200, None),
    ("GET", "/api/system/info", None, ADMIN, 200,
     {"version", "workers", "uptime_ms"}),
    ("GET", "/api/gpu", None, ADMIN, 200, None),
    ("GET", "/api/metrics", None, None, 200, None),
    ("GET", "/api/endpoints", None, None, 401, {"error"}),
    ("POST", "/api/system/drain", None, None, 401, {"error"}),
    ("GET", "/v1/nope", None, KEY, 404, {"error"}),
    ("GET", "/totally/unknown", None, None, 404, {"error"}),
    ("POST", "/v1/chat/completions", {"model": "cm", "messages": []},
     None, 401, {"error"}),
]


@pytest.mark.parametrize("method,path,body,auth,want,keys", CASES,
                         ids=[f"{m} {p} -> {w}"
                              for m, p, body, a, w, keys in CASES])
def test_route_contract(stack, method, path, body, auth, want, keys):
    headers = (dict(stack["admin"]) if auth is ADMIN
               else dict(stack["auth"]) if auth is KEY else {})
    status, text, hdrs = _req(method, stack["base"] + path, body,
                              headers=headers)
    assert status == want, (status, text[:200])
    ctype = hdrs.get("Content-Type", "")
    if path != "/api/metrics":
        assert "json" in ctype, ctype
        obj = json.loads(text)
        if keys:
            assert keys <= set(obj), (keys - set(obj), text[:200])
    else:
        assert "text/plain" in ctype
        assert "llmlb_requests_total" in text


def test_drain_contract(stack):
    base, admin = stack["base"], stack["admin"]
    status, body, _ = _req("POST", base + "/api/system/drain",
                           headers=admin)
    assert status == 200 and json.loads(body)["draining"] is True
    status, body, hdrs = _req(
        "POST", base + "/v1/chat/completions",
        {"model": "cm", "messages": []}, headers=stack["auth"])
    assert status == 503
    assert hdrs.get("Retry-After")
    status, body, _ = _req("POST", base + "/api/system/resume",
                           headers=admin)
    assert status == 200 and json.loads(body)["draining"] is False
