"""Per-engine model adapters (reference metadata/*.rs, download/mod.rs,
ollama-only delete): metadata/show, pull + auto-resync, delete."""
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
         "1", "--models", "llama3:8b", "--engine", "ollama"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "t",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
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
    status, body = _req("POST", base + "/api/endpoints",
                        {"name": "oll",
                         "base_url": f"http://127.0.0.1:{mock_port}"},
                        headers={"Authorization": "Bearer t"})
    assert status == 201, body
    ep = json.loads(body)
    yield base, ep["id"]
    proc.terminate()
    mock.terminate()
    for p in (proc, mock):
        try:
            p.wait(timeout=5)
        except subprocess.TimeoutExpired:
            p.kill()


ADMIN = {"Authorization": "Bearer t"}


def test_detected_as_ollama(gw):
    base, epid = gw
    status, body = _req("GET", f"{base}/api/endpoints/{epid}",
                        headers=ADMIN)
    assert json.loads(body)["endpoint_type"] == "ollama"


def test_metadata_adapter(gw):
    base, epid = gw
    status, body = _req(
        "GET", f"{base}/api/endpoints/{epid}/models/llama3:8b/metadata",
        headers=ADMIN)
    assert status == 200, body
    meta = json.loads(body)
    assert meta["details"]["family"] == "llama"
    status, _ = _req(
        "GET", f"{base}/api/endpoints/{epid}/models/nope/metadata",
        headers=ADMIN)
    assert status == 404


def test_download_and_delete_adapters(gw):
    base, epid = gw
    status, body = _req(
        "POST", f"{base}/api/endpoints/{epid}/models/download",
        {"model": "qwen:4b"}, headers=ADMIN)
    assert status == 200, body
    status, body = _req("GET", f"{base}/api/endpoints/{epid}",
                        headers=ADMIN)
    ids = [m["id"] for m in json.loads(body)["models"]]
    assert "qwen:4b" in ids
    status, body = _req(
        "DELETE", f"{base}/api/endpoints/{epid}/models/qwen:4b",
        headers=ADMIN)
    assert status == 200, body
    status, body = _req("GET", f"{base}/api/endpoints/{epid}",
                        headers=ADMIN)
    ids = [m["id"] for m in json.loads(body)["models"]]
    assert "qwen:4b" not in ids
