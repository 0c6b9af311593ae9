"""Audit query/FTS search endpoint + cloud models merged into
/v1/models (reference migrations/026 and api/openai.rs:451-467)."""
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
    data = json.dumps(body).encode() if body is not None else None
    r = urllib.request.Request(url, data=data, method=method,
                               headers=headers or {})
    try:
        with urllib.request.urlopen(r, timeout=timeout) as resp:
            return resp.status, resp.read().decode()
    except urllib.error.HTTPError as e:
        return e.code, e.read().decode()


@pytest.fixture(scope="module")
def gw():
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "localm,cloudm"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "1",
                # the same mock serves as the "openai" cloud endpoint:
                # its /v1/models list becomes openai:-prefixed entries
                "LLMLB_OPENAI_BASE_URL":
                    f"http://127.0.0.1:{mock_port}/v1",
                "LLMLB_OPENAI_API_KEY": "sk-cloud"})
    proc = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
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
    status, _ = _req("POST", base + "/api/endpoints",
                     {"name": "m",
                      "base_url": f"http://127.0.0.1:{mock_port}"},
                     headers={"Authorization": "Bearer t"})
    assert status == 201
    yield base
    proc.terminate()
    mock.terminate()
    for p in (proc, mock):
        try:
            p.wait(timeout=5)
        except subprocess.TimeoutExpired:
            p.kill()


def test_cloud_models_in_v1_models(gw):
    deadline = time.time() + 15
    ids, body = [], "{}"
    while time.time() < deadline:
        status, body = _req("GET", gw + "/v1/models")
        assert status == 200
        ids = [m["id"] for m in json.loads(body)["data"]]
        if any(i.startswith("openai:") for i in ids):
            break
        time.sleep(0.5)
    assert "localm" in ids
    assert "openai:localm" in ids, ids
    cm = next(m for m in json.loads(body)["data"]
              if m["id"] == "openai:localm")
    assert cm["owned_by"] == "openai"
    assert cm["endpoint_ids"] == []


def test_audit_query_and_search(gw):
    admin = {"Authorization": "Bearer t"}
    for _ in range(3):
        _req("GET", gw + "/api/endpoints", headers=admin)
    _req("GET", gw + "/api/dashboard/summary", headers=admin)
    time.sleep(0.6)  # writer flush
    status, body = _req("GET", gw + "/api/audit?limit=50",
                        headers=admin)
    assert status == 200, body
    out = json.loads(body)
    assert len(out["entries"]) >= 3
    e = out["entries"][0]
    assert {"ts_ms", "method", "path", "status", "client_ip",
            "hash"} <= set(e)
    status, body = _req("GET", gw + "/api/audit?path=/api/dashboard",
                        headers=admin)
    out = json.loads(body)
    assert out["entries"]
    assert all(x["path"].startswith("/api/dashboard")
               for x in out["entries"])
    status, body = _req("GET", gw + "/api/audit?q=dashboard",
                        headers=admin)
    out = json.loads(body)
    assert any("dashboard" in x["path"] for x in out["entries"])
    status, body = _req("GET", gw + "/api/audit/verify", headers=admin)
    assert json.loads(body)["valid"] is True


def test_client_analytics_dataplane(gw):
    """C++ data-plane analytics: rankings, heatmap grid shape, per-IP
    drilldown, monthly rollups (reference api/dashboard.rs)."""
    admin = {"Authorization": "Bearer t"}
    chat = {"model": "localm",
            "messages": [{"role": "user", "content": "x"}]}
    for _ in range(4):
        assert _req("POST", gw + "/v1/chat/completions", chat)[0] == 200
    time.sleep(0.6)
    status, body = _req("GET", gw + "/api/dashboard/clients",
                        headers=admin)
    assert status == 200
    clients = json.loads(body)
    assert clients and clients[0]["client_ip"] == "127.0.0.1"
    assert clients[0]["requests"] >= 4

    status, body = _req("GET", gw + "/api/dashboard/clients/heatmap",
                        headers=admin)
    hm = json.loads(body)
    assert len(hm["requests"]) == 7
    assert all(len(r) == 24 for r in hm["requests"])
    assert sum(sum(r) for r in hm["requests"]) >= 4

    status, body = _req("GET", gw + "/api/dashboard/clients/127.0.0.1",
                        headers=admin)
    assert status == 200
    d = json.loads(body)
    assert d["totals"]["requests"] >= 4
    assert d["models"][0]["model"] == "localm"
    assert d["recent"]
    assert _req("GET", gw + "/api/dashboard/clients/9.9.9.9",
                headers=admin)[0] == 404

    status, body = _req("GET",
                        gw + "/api/dashboard/token-stats/monthly",
                        headers=admin)
    assert status == 200
    months = json.loads(body)
    assert months and months[0]["requests"] >= 4


def test_settings_kv_dataplane(gw):
    admin = {"Authorization": "Bearer t"}
    status, body = _req("GET", gw + "/api/settings", headers=admin)
    assert status == 200
    status, body = _req("PUT", gw + "/api/settings",
                        {"theme": "dark", "rows": "25"}, headers=admin)
    assert status == 200 and json.loads(body)["updated"] == 2
    status, body = _req("GET", gw + "/api/settings", headers=admin)
    got = json.loads(body)
    assert got["theme"] == "dark" and got["rows"] == "25"
