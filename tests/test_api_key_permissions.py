"""API-key permission matrix on the data plane: inference keys cannot
touch admin surfaces; admin token can do both; revoked keys die."""
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


def _req(method, url, body=None, headers=None):
    data = json.dumps(body).encode() if body is not None else None
    r = urllib.request.Request(url, data=data, method=method,
                               headers=headers or {})
    try:
        with urllib.request.urlopen(r, timeout=10) as resp:
            return resp.status, resp.read().decode()
    except urllib.error.HTTPError as e:
        return e.code, e.read().decode()


@pytest.fixture(scope="module")
def gw():
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "pm"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "adm",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    proc = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                            env=env, stdout=subprocess.DEVNULL,
                            stderr=subprocess.DEVNULL)
    base = f"http://127.0.0.1:{gw_port}"
    for _ in range(50):
        try:
            _req("GET", base + "/health")
            break
        except Exception:
            time.sleep(0.1)
    _req("POST", base + "/api/endpoints",
         {"name": "m", "base_url": f"http://127.0.0.1:{mock_port}"},
         headers={"Authorization": "Bearer adm"})
    yield base
    proc.terminate()
    mock.terminate()
    for p in (proc, mock):
        try:
            p.wait(timeout=5)
        except subprocess.TimeoutExpired:
            p.kill()


def test_permission_matrix(gw):
    admin = {"Authorization": "Bearer adm"}
    status, body = _req("POST", gw + "/api/api-keys",
                        {"name": "inf", "permissions": "inference"},
                        headers=admin)
    assert status == 200 or status == 201, body
    key = json.loads(body)["api_key"]
    inf = {"Authorization": f"Bearer {key}"}
    chat = {"model": "pm",
            "messages": [{"role": "user", "content": "x"}]}
    # inference key: inference yes, admin no
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers=inf)[0] == 200
    assert _req("GET", gw + "/api/endpoints", headers=inf)[0] in (401,
                                                                  403)
    assert _req("POST", gw + "/api/api-keys", {"name": "x"},
                headers=inf)[0] in (401, 403)
    # admin token can do inference surfaces too
    assert _req("GET", gw + "/api/endpoints", headers=admin)[0] == 200
    # bad key rejected
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers={"Authorization": "Bearer nope"})[0] in (401,
                                                                 403)
    # revocation: the revoked key stops working while another key
    # still exists (revoking the LAST key returns the store to open
    # mode — fresh-install semantics — so keep a second key alive)
    status, body = _req("POST", gw + "/api/api-keys",
                        {"name": "keep", "permissions": "inference"},
                        headers=admin)
    keep = json.loads(body)["api_key"]
    status, body = _req("GET", gw + "/api/api-keys", headers=admin)
    keys = json.loads(body)
    kid = next(k["id"] for k in keys if k["name"] == "inf")
    assert _req("DELETE", gw + f"/api/api-keys/{kid}",
                headers=admin)[0] == 200
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers=inf)[0] in (401, 403)
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers={"Authorization":
                         f"Bearer {keep}"})[0] == 200


def test_scoped_permission_matrix(gw):
    """Reference 11-scope matrix (common/auth.rs:60-95): scoped keys
    reach exactly their routes; substring tokens grant nothing."""
    admin = {"Authorization": "Bearer adm"}

    def mint(perms):
        status, body = _req("POST", gw + "/api/api-keys",
                            {"name": perms, "permissions": perms},
                            headers=admin)
        assert status in (200, 201), body
        return {"Authorization": f"Bearer {json.loads(body)['api_key']}"}

    chat = {"model": "pm", "messages": [{"role": "user", "content": "x"}]}

    ep_read = mint("endpoints.read")
    assert _req("GET", gw + "/api/endpoints", headers=ep_read)[0] == 200
    assert _req("POST", gw + "/api/endpoints", {"name": "x"},
                headers=ep_read)[0] in (401, 403)          # no manage
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers=ep_read)[0] in (401, 403)          # no inference

    models_read = mint("openai.models.read")
    assert _req("GET", gw + "/v1/models", headers=models_read)[0] == 200
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers=models_read)[0] in (401, 403)

    metrics = mint("metrics.read")
    assert _req("GET", gw + "/api/dashboard/summary",
                headers=metrics)[0] == 200
    assert _req("GET", gw + "/api/endpoints",
                headers=metrics)[0] in (401, 403)

    # legacy "inference" implies models.read but not admin reads
    inf = mint("inference")
    assert _req("GET", gw + "/v1/models", headers=inf)[0] == 200
    assert _req("GET", gw + "/api/dashboard/summary",
                headers=inf)[0] in (401, 403)

    # substring tokens must not grant anything
    bad = mint("badminton")
    assert _req("POST", gw + "/v1/chat/completions", chat,
                headers=bad)[0] in (401, 403)
    assert _req("GET", gw + "/api/endpoints", headers=bad)[0] in (401,
                                                                  403)
