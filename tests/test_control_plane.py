"""Control-plane tests (FastAPI TestClient against an in-memory DB)."""
import sqlite3
import time

import pytest
from fastapi.testclient import TestClient

from llmlb_amd.api.control_plane import (create_app, hash_password,
                                         jwt_decode, jwt_encode,
                                         verify_password)


@pytest.fixture()
def app_client(tmp_path, monkeypatch):
    db = str(tmp_path / "cp.sqlite3")
    # create the schema pieces the control plane reads/writes
    conn = sqlite3.connect(db)
    conn.executescript("""
    CREATE TABLE users (id INTEGER PRIMARY KEY AUTOINCREMENT,
      username TEXT UNIQUE, password_hash TEXT, role TEXT DEFAULT 'user',
      must_change_password INTEGER DEFAULT 0, created_at TEXT);
    CREATE TABLE invitations (id INTEGER PRIMARY KEY AUTOINCREMENT,
      token TEXT UNIQUE, email TEXT, role TEXT, created_by INTEGER,
      created_at TEXT, expires_at TEXT, used_at TEXT);
    CREATE TABLE audit_log (id INTEGER PRIMARY KEY AUTOINCREMENT,
      ts_ms INTEGER, method TEXT, path TEXT, status INTEGER,
      client_ip TEXT, user TEXT, duration_ms REAL, prev_hash TEXT,
      hash TEXT);
    CREATE TABLE endpoint_daily_stats (date TEXT, endpoint_id TEXT,
      model TEXT, api_kind TEXT, requests INTEGER, failures INTEGER,
      prompt_tokens INTEGER, completion_tokens INTEGER,
      total_duration_ms REAL);
    CREATE TABLE request_history (id INTEGER PRIMARY KEY AUTOINCREMENT,
      ts_ms INTEGER, endpoint_id TEXT, model TEXT, api_kind TEXT,
      status INTEGER, duration_ms REAL, prompt_tokens INTEGER,
      completion_tokens INTEGER, client_ip TEXT, error TEXT,
      request_body TEXT, response_body TEXT);
    """)
    conn.commit()
    conn.close()
    monkeypatch.setenv("LLMLB_ADMIN_USER", "root")
    monkeypatch.setenv("LLMLB_ADMIN_PASSWORD", "rootpw123")
    app = create_app(db_path=db, jwt_secret="testsecret",
                     admin_token="admtok")
    return TestClient(app)


def _login(client, username="root", password="rootpw123"):
    r = client.post("/api/auth/login",
                    json={"username": username, "password": password})
    assert r.status_code == 200, r.text
    return {"Authorization": f"Bearer {r.json()['token']}"}


def test_jwt_roundtrip():
    tok = jwt_encode({"sub": 1, "role": "admin",
                      "exp": time.time() + 60}, "s")
    assert jwt_decode(tok, "s")["role"] == "admin"
    assert jwt_decode(tok, "wrong") is None
    expired = jwt_encode({"exp": time.time() - 10}, "s")
    assert jwt_decode(expired, "s") is None


def test_password_hashing():
    h = hash_password("secret")
    assert verify_password("secret", h)
    assert not verify_password("other", h)


def test_login_and_me(app_client):
    hdrs = _login(app_client)
    r = app_client.get("/api/auth/me", headers=hdrs)
    assert r.json()["role"] == "admin"


def test_login_rejects_bad_password(app_client):
    r = app_client.post("/api/auth/login",
                        json={"username": "root", "password": "nope"})
    assert r.status_code == 401


def test_user_crud_requires_admin(app_client):
    hdrs = _login(app_client)
    r = app_client.post("/api/users", headers=hdrs,
                        json={"username": "alice", "password": "pw12345",
                              "role": "user"})
    assert r.status_code == 201
    # alice (non-admin) cannot create users
    alice = _login(app_client, "alice", "pw12345")
    r = app_client.post("/api/users", headers=alice,
                        json={"username": "bob", "password": "x"})
    assert r.status_code == 403
    r = app_client.get("/api/users", headers=hdrs)
    assert {u["username"] for u in r.json()} >= {"root", "alice"}


def test_invitation_flow(app_client):
    hdrs = _login(app_client)
    r = app_client.post("/api/invitations", headers=hdrs,
                        json={"role": "user"})
    token = r.json()["token"]
    r = app_client.post("/api/auth/register",
                        json={"token": token, "username": "invitee",
                              "password": "pw12345"})
    assert r.status_code == 201
    # token single-use
    r = app_client.post("/api/auth/register",
                        json={"token": token, "username": "x2",
                              "password": "pw"})
    assert r.status_code == 400
    assert _login(app_client, "invitee", "pw12345")


def test_change_password(app_client):
    hdrs = _login(app_client)
    app_client.post("/api/users", headers=hdrs,
                    json={"username": "carol", "password": "old12345"})
    carol = _login(app_client, "carol", "old12345")
    r = app_client.post("/api/auth/change-password", headers=carol,
                        json={"old_password": "old12345",
                              "new_password": "new12345"})
    assert r.status_code == 200
    assert _login(app_client, "carol", "new12345")


def test_admin_token_accepted(app_client):
    r = app_client.get("/api/users",
                       headers={"Authorization": "Bearer admtok"})
    assert r.status_code == 200


def test_catalog_search(app_client):
    r = app_client.get("/api/catalog/search?q=llama",
                       headers={"Authorization": "Bearer admtok"})
    assert r.status_code == 200
    assert any("llama" in m["id"].lower() for m in r.json())


def test_token_stats_empty_ok(app_client):
    r = app_client.get("/api/dashboard/token-stats",
                       headers={"Authorization": "Bearer admtok"})
    assert r.status_code == 200


def test_update_manager_drain_apply_rollback(app_client, tmp_path,
                                             monkeypatch):
    """Self-update flow against a stub gateway (drain -> replace ->
    resume; rollback restores the .bak)."""
    import http.server
    import threading as th

    calls = []

    class StubGW(http.server.BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_POST(self):
            calls.append(self.path)
            self.send_response(200)
            self.send_header("Content-Length", "2")
            self.end_headers()
            self.wfile.write(b"{}")

        def do_GET(self):
            body = b"llmlb_active_requests 0\n"
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    import socket as sk
    s = sk.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    httpd = http.server.ThreadingHTTPServer(("127.0.0.1", port), StubGW)
    th.Thread(target=httpd.serve_forever, daemon=True).start()

    from llmlb_amd.api import control_plane as cp
    app = cp.create_app(db_path=":memory:",
                        gateway_url=f"http://127.0.0.1:{port}",
                        jwt_secret="s", admin_token="admtok")
    from fastapi.testclient import TestClient
    client = TestClient(app)
    hdrs = {"Authorization": "Bearer admtok"}

    old = tmp_path / "llmlb-gw"
    old.write_bytes(b"OLD BINARY")
    new = tmp_path / "artifact"
    new.write_bytes(b"NEW BINARY")
    r = client.post("/api/update/apply", headers=hdrs,
                    json={"artifact_path": str(new),
                          "target_path": str(old)})
    assert r.status_code == 202
    deadline = time.time() + 15
    while time.time() < deadline:
        st = client.get("/api/update/status", headers=hdrs).json()
        if st["state"] in ("applied", "failed"):
            break
        time.sleep(0.2)
    assert st["state"] == "applied", st
    assert old.read_bytes() == b"NEW BINARY"
    assert (tmp_path / "llmlb-gw.bak").read_bytes() == b"OLD BINARY"
    assert "/api/system/drain" in calls
    assert "/api/system/resume" in calls
    r = client.post("/api/update/rollback", headers=hdrs)
    assert r.status_code == 200
    assert old.read_bytes() == b"OLD BINARY"
    httpd.shutdown()


def test_ws_dashboard_auth_and_stream(app_client):
    """/ws/dashboard rejects bad tokens and streams typed events for a
    valid admin token (gateway absent -> error-typed events still show
    the loop is alive)."""
    import pytest as _pytest
    from starlette.websockets import WebSocketDisconnect
    with _pytest.raises(WebSocketDisconnect):
        with app_client.websocket_connect("/ws/dashboard?token=bad"):
            pass
    with app_client.websocket_connect(
            "/ws/dashboard?token=admtok") as ws:
        msg = ws.receive_json()
        assert msg["type"] in ("summary", "event", "error")


def test_logs_tail_reads_rotated_file(app_client, tmp_path,
                                      monkeypatch):
    """/api/logs tails the newest dated llmlb-gw-YYYYMMDD.jsonl."""
    import llmlb_amd.api.control_plane as cp
    logd = tmp_path / "logs"
    logd.mkdir()
    (logd / "llmlb-gw-20260101.jsonl").write_text('{"old":1}\n')
    (logd / "llmlb-gw-20260914.jsonl").write_text(
        '{"msg":"a"}\n{"msg":"b"}\n')
    monkeypatch.setenv("LLMLB_LOG_DIR", str(logd))
    from llmlb_amd.config import get_config
    get_config(refresh=True)
    app = cp.create_app(db_path=":memory:", jwt_secret="s",
                        admin_token="tok2")
    from fastapi.testclient import TestClient
    c = TestClient(app)
    r = c.get("/api/logs", headers={"Authorization": "Bearer tok2"})
    assert r.status_code == 200
    lines = r.json()["lines"]
    assert '{"msg":"b"}' in lines and '{"old":1}' not in lines


def test_requests_csv_export(app_client, tmp_path, monkeypatch):
    import llmlb_amd.api.control_plane as cp
    db = str(tmp_path / "csv.sqlite3")
    conn = sqlite3.connect(db)
    conn.executescript("""
    CREATE TABLE users (id INTEGER PRIMARY KEY AUTOINCREMENT,
      username TEXT UNIQUE, password_hash TEXT, role TEXT DEFAULT 'user',
      must_change_password INTEGER DEFAULT 0, created_at TEXT);
    CREATE TABLE request_history (id INTEGER PRIMARY KEY AUTOINCREMENT,
      ts_ms INTEGER, endpoint_id TEXT, model TEXT, api_kind TEXT,
      status INTEGER, duration_ms REAL, prompt_tokens INTEGER,
      completion_tokens INTEGER, client_ip TEXT, error TEXT,
      request_body TEXT, response_body TEXT, request_id TEXT);
    """)
    import time as _t
    now = int(_t.time() * 1000)
    conn.execute(
        "INSERT INTO request_history (ts_ms,endpoint_id,model,api_kind,"
        "status,duration_ms,prompt_tokens,completion_tokens,client_ip,"
        "error) VALUES (?,?,?,?,?,?,?,?,?,?)",
        (now, "e1", "m1", "chat_completions", 200, 12.5, 10, 20,
         "1.2.3.4", ""))
    conn.commit()
    conn.close()
    app = cp.create_app(db_path=db, jwt_secret="s", admin_token="tok3")
    from fastapi.testclient import TestClient
    c = TestClient(app)
    r = c.get("/api/requests/export.csv",
              headers={"Authorization": "Bearer tok3"})
    assert r.status_code == 200
    lines = r.text.strip().splitlines()
    assert lines[0].startswith("ts_ms,endpoint_id,model")
    assert any("m1" in ln and "chat_completions" in ln
               for ln in lines[1:])


def test_dashboard_analytics_and_registry(tmp_path):
    """clients / timeline / audit query / model-registry CRUD /
    control health on a seeded DB."""
    import llmlb_amd.api.control_plane as cp
    import time as _t
    db = str(tmp_path / "an.sqlite3")
    conn = sqlite3.connect(db)
    conn.executescript("""
    CREATE TABLE users (id INTEGER PRIMARY KEY AUTOINCREMENT,
      username TEXT UNIQUE, password_hash TEXT, role TEXT DEFAULT 'user',
      must_change_password INTEGER DEFAULT 0, created_at TEXT);
    CREATE TABLE request_history (id INTEGER PRIMARY KEY AUTOINCREMENT,
      ts_ms INTEGER, endpoint_id TEXT, model TEXT, api_kind TEXT,
      status INTEGER, duration_ms REAL, prompt_tokens INTEGER,
      completion_tokens INTEGER, client_ip TEXT, error TEXT,
      request_body TEXT, response_body TEXT, request_id TEXT);
    CREATE TABLE audit_log (id INTEGER PRIMARY KEY AUTOINCREMENT,
      ts_ms INTEGER, method TEXT, path TEXT, status INTEGER,
      client_ip TEXT, user TEXT, duration_ms REAL,
      prev_hash TEXT, hash TEXT);
    CREATE TABLE models (id TEXT PRIMARY KEY, name TEXT NOT NULL,
      source TEXT NOT NULL DEFAULT '', capabilities TEXT NOT NULL
      DEFAULT 'chat', size_bytes INTEGER, created_at TEXT NOT NULL);
    """)
    now = int(_t.time() * 1000)
    for i in range(5):
        conn.execute(
            "INSERT INTO request_history (ts_ms,endpoint_id,model,"
            "api_kind,status,duration_ms,prompt_tokens,"
            "completion_tokens,client_ip,error) VALUES "
            "(?,?,?,?,?,?,?,?,?,?)",
            (now - i * 60000, "e1", "m1", "chat_completions",
             200 if i else 500, 10.0, 5, 7, f"10.0.0.{i % 2}", ""))
    conn.execute("INSERT INTO audit_log (ts_ms,method,path,status,"
                 "client_ip,user,duration_ms,prev_hash,hash) VALUES "
                 "(?,?,?,?,?,?,?,?,?)",
                 (now, "GET", "/x", 200, "10.0.0.1", "", 1.0, "p", "h"))
    conn.commit()
    conn.close()
    app = cp.create_app(db_path=db, jwt_secret="s", admin_token="tk")
    from fastapi.testclient import TestClient
    c = TestClient(app)
    hdr = {"Authorization": "Bearer tk"}
    r = c.get("/api/dashboard/clients", headers=hdr)
    assert r.status_code == 200
    assert any(cl["client_ip"].startswith("10.0.0.")
               for cl in r.json())
    r = c.get("/api/dashboard/timeline?hours=2", headers=hdr)
    assert r.status_code == 200 and r.json()
    r = c.get("/api/audit?limit=10", headers=hdr)
    assert r.status_code == 200 and len(r.json()) == 1
    # model-registry CRUD (schema shared with the data plane's db.cpp)
    r = c.post("/api/models", headers=hdr,
               json={"id": "meta-llama/Meta-Llama-3-8B",
                     "capabilities": "chat"})
    assert r.status_code in (200, 201), r.text
    r = c.get("/api/models", headers=hdr)
    assert any(m["id"] == "meta-llama/Meta-Llama-3-8B"
               for m in r.json())
    r = c.delete("/api/models/meta-llama/Meta-Llama-3-8B", headers=hdr)
    assert r.status_code == 200, r.text
    assert c.get("/api/control/health").status_code == 200


def test_tps_benchmark_runner(tmp_path):
    """POST /api/benchmarks/tps drives real requests through a live
    gateway and records tokens/sec; GET returns the finished result."""
    import socket
    import subprocess
    import sys
    import tempfile
    import llmlb_amd.api.control_plane as cp
    from pathlib import Path
    REPO = Path(__file__).resolve().parents[1]
    BIN = REPO / "gateway" / "bin"

    def free():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    import os
    mock_port, gw_port = free(), free()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "bm"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    env = os.environ.copy()
    env.update({"LLMLB_PORT": str(gw_port), "LLMLB_DATABASE_URL": db,
                "LLMLB_ADMIN_TOKEN": "bt",
                "LLMLB_HEALTH_CHECK_INTERVAL": "60"})
    gw = subprocess.Popen([str(BIN / "llmlb-gw"), "--workers", "1"],
                          env=env, stdout=subprocess.DEVNULL,
                          stderr=subprocess.DEVNULL)
    import urllib.request
    import json as _json
    import time as _t
    base = f"http://127.0.0.1:{gw_port}"
    try:
        deadline = _t.time() + 15
        while _t.time() < deadline:
            try:
                urllib.request.urlopen(base + "/health", timeout=2)
                break
            except Exception:
                _t.sleep(0.1)
        req = urllib.request.Request(
            base + "/api/endpoints",
            data=_json.dumps({"name": "b",
                              "base_url":
                                  f"http://127.0.0.1:{mock_port}"
                              }).encode(),
            headers={"Authorization": "Bearer bt"})
        urllib.request.urlopen(req, timeout=10)

        # real control-plane process: background benchmark tasks need
        # a continuously-running event loop (TestClient suspends it)
        cp_port = free()
        cp_env = os.environ.copy()
        cp_env.update({"LLMLB_DATABASE_URL": tempfile.mktemp(),
                       "LLMLB_ADMIN_TOKEN": "bt",
                       "LLMLB_JWT_SECRET": "s",
                       "LLMLB_GATEWAY_URL": base})
        cpp = subprocess.Popen(
            [sys.executable, "-m", "llmlb_amd.api.control_plane",
             "--port", str(cp_port)], env=cp_env, cwd=str(REPO),
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        try:
            cp_base = f"http://127.0.0.1:{cp_port}"
            deadline = _t.time() + 30
            while _t.time() < deadline:
                try:
                    urllib.request.urlopen(cp_base
                                           + "/api/control/health",
                                           timeout=2)
                    break
                except Exception:
                    _t.sleep(0.3)
            req = urllib.request.Request(
                cp_base + "/api/benchmarks/tps",
                data=_json.dumps({"model": "bm", "requests": 6,
                                  "concurrency": 2,
                                  "max_tokens": 16}).encode(),
                headers={"Authorization": "Bearer bt",
                         "Content-Type": "application/json"})
            with urllib.request.urlopen(req, timeout=10) as r:
                bid = _json.loads(r.read())["benchmark_id"]
            deadline = _t.time() + 30
            res = None
            while _t.time() < deadline:
                q = urllib.request.Request(
                    cp_base + f"/api/benchmarks/tps/{bid}",
                    headers={"Authorization": "Bearer bt"})
                with urllib.request.urlopen(q, timeout=10) as r:
                    res = _json.loads(r.read())
                if res.get("status") != "running":
                    break
                _t.sleep(0.3)
            assert res and res["status"] == "done", res
            assert res["errors"] == 0 and res["tokens_per_sec"] > 0
            assert res["requests"] == 6
            assert res["source"] == "benchmark"
        finally:
            cpp.terminate()
            try:
                cpp.wait(timeout=5)
            except subprocess.TimeoutExpired:
                cpp.kill()
    finally:
        gw.terminate()
        mock.terminate()
        for p in (gw, mock):
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


def _seed_history(client_db_path=None, client=None):
    pass


def test_analytics_heatmap_drilldown_monthly(app_client):
    import sqlite3 as _sq
    hdrs = _login(app_client)
    # seed request history + daily stats through the CP's own db path
    db_path = app_client.app.state.db_path
    conn = _sq.connect(db_path)
    now = int(time.time() * 1000)
    for i in range(30):
        conn.execute(
            "INSERT INTO request_history (ts_ms, endpoint_id, model, "
            "api_kind, status, duration_ms, prompt_tokens, "
            "completion_tokens, client_ip) VALUES (?,?,?,?,?,?,?,?,?)",
            (now - i * 3600 * 1000, "e1", "m1", "chat_completions",
             200 if i % 5 else 500, 12.5, 10, 20,
             "10.0.0.1" if i % 2 else "10.0.0.2"))
    conn.execute(
        "INSERT INTO endpoint_daily_stats VALUES "
        "('2026-08-01','e1','m1','chat_completions',10,1,100,200,50)")
    conn.execute(
        "INSERT INTO endpoint_daily_stats VALUES "
        "('2026-09-01','e1','m1','chat_completions',20,0,300,400,60)")
    conn.commit()
    conn.close()

    r = app_client.get("/api/dashboard/clients/heatmap?days=7",
                       headers=hdrs)
    assert r.status_code == 200
    hm = r.json()
    assert len(hm["requests"]) == 7
    assert all(len(row) == 24 for row in hm["requests"])
    assert sum(sum(row) for row in hm["requests"]) >= 30

    r = app_client.get("/api/dashboard/clients/10.0.0.1", headers=hdrs)
    assert r.status_code == 200
    d = r.json()
    assert d["totals"]["requests"] >= 10
    assert d["models"][0]["model"] == "m1"
    assert d["hourly"] and d["recent"]
    assert app_client.get("/api/dashboard/clients/10.9.9.9",
                          headers=hdrs).status_code == 404

    r = app_client.get("/api/dashboard/token-stats/monthly",
                       headers=hdrs)
    assert r.status_code == 200
    months = {m["month"]: m for m in r.json()}
    assert months["2026-08"]["completion_tokens"] == 200
    assert months["2026-09"]["requests"] == 20


def test_settings_kv(app_client):
    hdrs = _login(app_client)
    assert app_client.get("/api/settings", headers=hdrs).json() == {}
    r = app_client.put("/api/settings",
                       json={"theme": "dark", "rows": "50"},
                       headers=hdrs)
    assert r.status_code == 200 and r.json()["updated"] == 2
    got = app_client.get("/api/settings", headers=hdrs).json()
    assert got == {"theme": "dark", "rows": "50"}


def test_cookie_session_requires_csrf(app_client):
    """Browser cookie sessions enforce CSRF double-submit on mutating
    requests (reference auth CSRF middleware)."""
    r = app_client.post("/api/auth/login",
                        json={"username": "root",
                              "password": "rootpw123"})
    assert r.status_code == 200
    csrf = r.json()["csrf_token"]
    assert "llmlb_session" in r.cookies
    # cookie-only GET works (TestClient persists cookies)
    assert app_client.get("/api/auth/me").status_code == 200
    # mutating request with cookie but no CSRF header -> 403
    r2 = app_client.put("/api/settings", json={"a": "1"})
    assert r2.status_code == 403
    # with the CSRF header -> accepted
    r3 = app_client.put("/api/settings", json={"a": "1"},
                        headers={"X-CSRF-Token": csrf})
    assert r3.status_code == 200
    # bearer-token requests never need CSRF
    hdrs = _login(app_client)
    assert app_client.put("/api/settings", json={"b": "2"},
                          headers=hdrs).status_code == 200


CP_CASES = [
    ("GET", "/api/auth/me", None, True, 200, {"username", "role"}),
    ("GET", "/api/users", None, True, 200, None),
    ("GET", "/api/invitations", None, True, 200, None),
    ("GET", "/api/audit", None, True, 200, None),
    ("GET", "/api/dashboard/token-stats", None, True, 200, None),
    ("GET", "/api/dashboard/token-stats/monthly", None, True, 200,
     None),
    ("GET", "/api/dashboard/clients", None, True, 200, None),
    ("GET", "/api/dashboard/clients/heatmap", None, True, 200,
     {"requests"}),
    ("GET", "/api/dashboard/timeline", None, True, 200, None),
    ("GET", "/api/settings", None, True, 200, None),
    ("GET", "/api/models", None, True, 200, None),
    ("GET", "/api/catalog/search?q=llama", None, True, 200, None),
    ("GET", "/api/logs", None, True, 200, {"lines"}),
    ("GET", "/api/update/status", None, True, 200, None),
    ("GET", "/api/users", None, False, 401, None),
    ("GET", "/api/auth/me", None, False, 401, None),
]


@pytest.mark.parametrize("method,path,body,authed,want,keys", CP_CASES,
                         ids=[f"{m} {p} -> {w}"
                              for m, p, b, a, w, keys in CP_CASES])
def test_cp_route_contract(app_client, method, path, body, authed,
                           want, keys):
    hdrs = _login(app_client) if authed else {}
    r = app_client.request(method, path,
                           json=body if body is not None else None,
                           headers=hdrs)
    assert r.status_code == want, (r.status_code, r.text[:200])
    if want == 200 and keys:
        assert keys <= set(r.json()), r.text[:200]


def test_scheduled_update(tmp_path):
    """Scheduled self-update (reference api/system.rs:218-330): create
    fires the drain->apply path at the scheduled time; get reflects
    the pending schedule; cancel stops it; one schedule at a time."""
    import http.server
    import socket as sk
    import threading as th

    class StubGW(http.server.BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_POST(self):
            self.send_response(200)
            self.send_header("Content-Length", "2")
            self.end_headers()
            self.wfile.write(b"{}")

        def do_GET(self):
            body = b"llmlb_active_requests 0\n"
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    s = sk.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    httpd = http.server.ThreadingHTTPServer(("127.0.0.1", port), StubGW)
    th.Thread(target=httpd.serve_forever, daemon=True).start()

    from llmlb_amd.api import control_plane as cp
    from fastapi.testclient import TestClient
    app = cp.create_app(db_path=":memory:",
                        gateway_url=f"http://127.0.0.1:{port}",
                        jwt_secret="s", admin_token="admtok")
    client = TestClient(app)
    hdrs = {"Authorization": "Bearer admtok"}

    old = tmp_path / "llmlb-gw"
    old.write_bytes(b"OLD BINARY")
    new = tmp_path / "artifact"
    new.write_bytes(b"NEW BINARY")

    # cancel without a schedule -> 404; past time -> 400
    assert client.delete("/api/update/schedule",
                         headers=hdrs).status_code == 404
    r = client.post("/api/update/schedule", headers=hdrs,
                    json={"artifact_path": str(new),
                          "target_path": str(old),
                          "at_epoch": time.time() - 10})
    assert r.status_code == 400

    # schedule, observe pending, duplicate -> 409, cancel works
    r = client.post("/api/update/schedule", headers=hdrs,
                    json={"artifact_path": str(new),
                          "target_path": str(old), "delay_s": 30})
    assert r.status_code == 202
    st = client.get("/api/update/schedule", headers=hdrs).json()
    assert st["scheduled"] and st["artifact_path"] == str(new)
    assert client.post("/api/update/schedule", headers=hdrs,
                       json={"artifact_path": str(new),
                             "delay_s": 30}).status_code == 409
    assert client.delete("/api/update/schedule",
                         headers=hdrs).status_code == 200
    assert client.get("/api/update/schedule",
                      headers=hdrs).json() == {"scheduled": False}
    assert old.read_bytes() == b"OLD BINARY"  # never fired

    # short schedule actually applies
    r = client.post("/api/update/schedule", headers=hdrs,
                    json={"artifact_path": str(new),
                          "target_path": str(old), "delay_s": 0.2})
    assert r.status_code == 202
    deadline = time.time() + 15
    st = {}
    while time.time() < deadline:
        st = client.get("/api/update/status", headers=hdrs).json()
        if st.get("state") in ("applied", "failed"):
            break
        time.sleep(0.2)
    assert st.get("state") == "applied", st
    assert old.read_bytes() == b"NEW BINARY"
    assert client.get("/api/update/schedule",
                      headers=hdrs).json() == {"scheduled": False}
    httpd.shutdown()
