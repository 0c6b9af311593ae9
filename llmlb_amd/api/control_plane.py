"""Python control plane for llmlb-gw.

The C++ data plane serves all /v1/* inference routes and the hot admin
surface; it proxies everything it does not handle (dashboard data
drill-downs, auth/users, invitations, audit queries, TPS benchmarks,
catalog, WS events) to this FastAPI app over loopback
(LLMLB_CONTROL_PLANE_URL).  Shares the gateway's SQLite database (WAL).

Reference coverage: components 20 (dashboard API), 21
(dashboard WS), 22 (benchmarks), 23 (catalog), 24 (users/api_keys/
invitations/audit API), 25 (auth: JWT login, password policy).
bcrypt is not in this image, so passwords use PBKDF2-HMAC-SHA256
(documented deviation; same role in the schema).
"""
from __future__ import annotations

import asyncio
import base64
import hashlib
import hmac
import json
import os
import secrets
import sqlite3
import time
import urllib.request
from contextlib import contextmanager

from fastapi import (Depends, FastAPI, Header, HTTPException, Query,
                     Request, WebSocket, WebSocketDisconnect)
from fastapi.responses import JSONResponse
from pydantic import BaseModel

# --------------------------------------------------------------- jwt
def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def jwt_encode(payload: dict, secret: str) -> str:
    header = _b64url(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
    body = _b64url(json.dumps(payload).encode())
    msg = f"{header}.{body}".encode()
    sig = _b64url(hmac.new(secret.encode(), msg, hashlib.sha256).digest())
    return f"{header}.{body}.{sig}"


def jwt_decode(token: str, secret: str) -> dict | None:
    try:
        header, body, sig = token.split(".")
        msg = f"{header}.{body}".encode()
        expect = _b64url(hmac.new(secret.encode(), msg,
                                  hashlib.sha256).digest())
        if not hmac.compare_digest(sig, expect):
            return None
        pad = "=" * (-len(body) % 4)
        payload = json.loads(base64.urlsafe_b64decode(body + pad))
        if payload.get("exp", 0) < time.time():
            return None
        return payload
    except Exception:
        return None


def hash_password(password: str, salt: bytes | None = None) -> str:
    salt = salt or secrets.token_bytes(16)
    dk = hashlib.pbkdf2_hmac("sha256", password.encode(), salt, 100_000)
    return f"pbkdf2${salt.hex()}${dk.hex()}"


def verify_password(password: str, stored: str) -> bool:
    try:
        _, salt_hex, dk_hex = stored.split("$")
        dk = hashlib.pbkdf2_hmac("sha256", password.encode(),
                                 bytes.fromhex(salt_hex), 100_000)
        return hmac.compare_digest(dk.hex(), dk_hex)
    except Exception:
        return False


# --------------------------------------------------------------- app
class LoginBody(BaseModel):
    username: str
    password: str


class UserBody(BaseModel):
    username: str
    password: str
    role: str = "user"


class ChangePasswordBody(BaseModel):
    old_password: str
    new_password: str


class InvitationBody(BaseModel):
    email: str = ""
    role: str = "user"
    expires_hours: int = 72


class RegisterBody(BaseModel):
    token: str
    username: str
    password: str


class ModelRegBody(BaseModel):
    id: str
    name: str = ""
    source: str = ""
    capabilities: str = "chat"
    size_bytes: int | None = None


class UpdateBody(BaseModel):
    artifact_path: str
    target_path: str = ""


class ScheduleBody(BaseModel):
    artifact_path: str
    target_path: str = ""
    at_epoch: float | None = None
    delay_s: float | None = None


class TpsBenchmarkBody(BaseModel):
    # reference api/benchmarks.rs:25-34 defaults and caps
    model: str
    requests: int = 20
    concurrency: int = 4
    max_tokens: int = 128
    temperature: float = 0.2


def create_app(db_path: str | None = None,
               gateway_url: str | None = None,
               jwt_secret: str | None = None,
               admin_token: str | None = None) -> FastAPI:
    from ..config import get_config
    cfg = get_config(refresh=True)
    db_path = db_path or cfg.database_url
    gateway_url = gateway_url or os.environ.get(
        "LLMLB_GATEWAY_URL", "http://127.0.0.1:32768")
    jwt_secret = jwt_secret or cfg.jwt_secret or secrets.token_hex(32)
    admin_token = admin_token or os.environ.get("LLMLB_ADMIN_TOKEN", "")

    app = FastAPI(title="llmlb control plane", docs_url=None,
                  redoc_url=None)
    app.state.db_path = db_path

    @contextmanager
    def db():
        conn = sqlite3.connect(db_path, timeout=10)
        conn.row_factory = sqlite3.Row
        try:
            yield conn
            conn.commit()
        finally:
            conn.close()

    def ensure_admin():
        admin_user = cfg.admin_user
        admin_pass = cfg.admin_password
        with db() as conn:
            row = conn.execute("SELECT COUNT(*) c FROM users").fetchone()
            if row["c"] == 0:
                pw = admin_pass or secrets.token_urlsafe(12)
                conn.execute(
                    "INSERT INTO users (username,password_hash,role,"
                    "must_change_password,created_at) VALUES (?,?,?,?,?)",
                    (admin_user, hash_password(pw), "admin",
                     0 if admin_pass else 1,
                     time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())))
                if not admin_pass:
                    print(f"[control-plane] created admin user "
                          f"'{admin_user}' with password: {pw}",
                          flush=True)

    try:
        ensure_admin()
    except sqlite3.OperationalError:
        pass  # gateway migrates the schema; first boot ordering

    # ---- auth dependency ----
    from fastapi import Header

    def auth_user(request: Request,
                  authorization: str | None = Header(None),
                  x_csrf_token: str | None = Header(None)):
        token = None
        from_cookie = False
        if authorization:
            token = authorization
            if token.lower().startswith("bearer "):
                token = token[7:]
        else:
            # browser cookie session (reference JWT-via-cookie path,
            # auth/middleware.rs).  Cookie auth on state-changing
            # methods requires the CSRF double-submit header to match
            # the csrf claim minted at login.
            token = request.cookies.get("llmlb_session")
            from_cookie = token is not None
        if not token:
            raise HTTPException(401, "missing Authorization")
        if admin_token and token == admin_token:
            return {"username": "admin-token", "role": "admin"}
        payload = jwt_decode(token, jwt_secret)
        if not payload:
            raise HTTPException(401, "invalid token")
        if from_cookie and request.method not in ("GET", "HEAD",
                                                  "OPTIONS"):
            if not x_csrf_token or x_csrf_token != payload.get("csrf"):
                raise HTTPException(403, "CSRF token missing or invalid")
        return payload

    def require_admin(user=Depends(auth_user)):
        if user.get("role") != "admin":
            raise HTTPException(403, "admin role required")
        return user

    # ---------------- auth ----------------
    @app.post("/api/auth/login")
    def login(body: LoginBody):
        with db() as conn:
            row = conn.execute(
                "SELECT id,username,password_hash,role,"
                "must_change_password FROM users WHERE username=?",
                (body.username,)).fetchone()
        if not row or not verify_password(body.password,
                                          row["password_hash"]):
            raise HTTPException(401, "invalid credentials")
        csrf = secrets.token_hex(16)
        token = jwt_encode(
            {"sub": row["id"], "username": row["username"],
             "role": row["role"], "csrf": csrf,
             "exp": time.time() + 24 * 3600},
            jwt_secret)
        resp = JSONResponse(
            {"token": token, "role": row["role"], "csrf_token": csrf,
             "must_change_password": bool(row["must_change_password"])})
        # httpOnly session cookie for browser flows; SPA keeps using
        # the bearer token, curl/browser forms get cookie + CSRF
        resp.set_cookie("llmlb_session", token, httponly=True,
                        samesite="lax", max_age=24 * 3600)
        return resp

    @app.post("/api/auth/change-password")
    def change_password(body: ChangePasswordBody, user=Depends(auth_user)):
        with db() as conn:
            row = conn.execute(
                "SELECT id,password_hash FROM users WHERE username=?",
                (user["username"],)).fetchone()
            if not row or not verify_password(body.old_password,
                                              row["password_hash"]):
                raise HTTPException(401, "wrong password")
            conn.execute(
                "UPDATE users SET password_hash=?,must_change_password=0 "
                "WHERE id=?", (hash_password(body.new_password),
                               row["id"]))
        return {"changed": True}

    @app.get("/api/auth/me")
    def me(user=Depends(auth_user)):
        return {"username": user.get("username"), "role": user.get("role")}

    # ---------------- users ----------------
    @app.get("/api/users")
    def list_users(user=Depends(require_admin)):
        with db() as conn:
            rows = conn.execute(
                "SELECT id,username,role,must_change_password,created_at "
                "FROM users").fetchall()
        return [dict(r) for r in rows]

    @app.post("/api/users", status_code=201)
    def create_user(body: UserBody, user=Depends(require_admin)):
        with db() as conn:
            try:
                conn.execute(
                    "INSERT INTO users (username,password_hash,role,"
                    "created_at) VALUES (?,?,?,?)",
                    (body.username, hash_password(body.password),
                     body.role,
                     time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())))
            except sqlite3.IntegrityError:
                raise HTTPException(409, "username exists")
        return {"created": body.username}

    @app.delete("/api/users/{username}")
    def delete_user(username: str, user=Depends(require_admin)):
        with db() as conn:
            cur = conn.execute("DELETE FROM users WHERE username=? AND "
                               "role != 'admin'", (username,))
            if cur.rowcount == 0:
                raise HTTPException(404, "no such (non-admin) user")
        return {"deleted": username}

    # ---------------- invitations ----------------
    @app.post("/api/invitations", status_code=201)
    def create_invitation(body: InvitationBody,
                          user=Depends(require_admin)):
        token = secrets.token_urlsafe(24)
        with db() as conn:
            conn.execute(
                "INSERT INTO invitations (token,email,role,created_at,"
                "expires_at) VALUES (?,?,?,?,?)",
                (token, body.email, body.role,
                 time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
                 time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime(
                     time.time() + body.expires_hours * 3600))))
        return {"token": token, "role": body.role}

    @app.get("/api/invitations")
    def list_invitations(user=Depends(require_admin)):
        with db() as conn:
            rows = conn.execute(
                "SELECT id,token,email,role,created_at,expires_at,used_at "
                "FROM invitations").fetchall()
        return [dict(r) for r in rows]

    @app.post("/api/auth/register", status_code=201)
    def register(body: RegisterBody):
        now = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        with db() as conn:
            inv = conn.execute(
                "SELECT id,role,expires_at,used_at FROM invitations "
                "WHERE token=?", (body.token,)).fetchone()
            if not inv or inv["used_at"]:
                raise HTTPException(400, "invalid invitation")
            if inv["expires_at"] and inv["expires_at"] < now:
                raise HTTPException(400, "invitation expired")
            try:
                conn.execute(
                    "INSERT INTO users (username,password_hash,role,"
                    "created_at) VALUES (?,?,?,?)",
                    (body.username, hash_password(body.password),
                     inv["role"], now))
            except sqlite3.IntegrityError:
                raise HTTPException(409, "username exists")
            conn.execute("UPDATE invitations SET used_at=? WHERE id=?",
                         (now, inv["id"]))
        return {"created": body.username, "role": inv["role"]}

    # ---------------- audit queries ----------------
    @app.get("/api/audit")
    def audit_log(limit: int = Query(100, le=1000), offset: int = 0,
                  q: str = "", user=Depends(require_admin)):
        sql = ("SELECT id,ts_ms,method,path,status,client_ip,user,"
               "duration_ms,hash FROM audit_log")
        params: list = []
        if q:
            sql += " WHERE path LIKE ? OR user LIKE ? OR client_ip LIKE ?"
            params += [f"%{q}%"] * 3
        sql += " ORDER BY id DESC LIMIT ? OFFSET ?"
        params += [limit, offset]
        with db() as conn:
            rows = conn.execute(sql, params).fetchall()
        return [dict(r) for r in rows]

    # ---------------- dashboard stats (drill-downs) ----------------
    @app.get("/api/dashboard/token-stats")
    def token_stats(days: int = Query(30, le=365),
                    user=Depends(auth_user)):
        with db() as conn:
            rows = conn.execute(
                "SELECT date, SUM(prompt_tokens) pt, "
                "SUM(completion_tokens) ct, SUM(requests) reqs, "
                "SUM(failures) fails FROM endpoint_daily_stats "
                "GROUP BY date ORDER BY date DESC LIMIT ?",
                (days,)).fetchall()
        return [dict(r) for r in rows]

    @app.get("/api/dashboard/clients")
    def client_analytics(limit: int = Query(20, le=100),
                         user=Depends(auth_user)):
        cutoff = int(time.time() * 1000) - 7 * 86400 * 1000
        with db() as conn:
            rows = conn.execute(
                "SELECT client_ip, COUNT(*) requests, "
                "SUM(completion_tokens) tokens, AVG(duration_ms) avg_ms "
                "FROM request_history WHERE ts_ms > ? GROUP BY client_ip "
                "ORDER BY requests DESC LIMIT ?",
                (cutoff, limit)).fetchall()
        return [dict(r) for r in rows]

    @app.get("/api/dashboard/timeline")
    def request_timeline(hours: int = Query(24, le=168),
                         bucket_minutes: int = 10,
                         user=Depends(auth_user)):
        cutoff = int(time.time() * 1000) - hours * 3600 * 1000
        bucket_ms = bucket_minutes * 60 * 1000
        with db() as conn:
            rows = conn.execute(
                "SELECT (ts_ms / ?) * ? bucket, COUNT(*) requests, "
                "SUM(CASE WHEN status >= 400 THEN 1 ELSE 0 END) failures, "
                "SUM(completion_tokens) tokens FROM request_history "
                "WHERE ts_ms > ? GROUP BY bucket ORDER BY bucket",
                (bucket_ms, bucket_ms, cutoff)).fetchall()
        return [dict(r) for r in rows]

    @app.get("/api/dashboard/clients/heatmap")
    def client_heatmap(days: int = Query(7, le=90),
                       client_ip: str | None = None,
                       user=Depends(auth_user)):
        """Weekday x hour activity heatmap (reference client-analytics
        heatmap, api/dashboard.rs): 7x24 request counts over the
        window, optionally for one client."""
        cutoff = int(time.time() * 1000) - days * 86400 * 1000
        q = ("SELECT CAST(strftime('%w', ts_ms/1000, 'unixepoch') AS "
             "INTEGER) dow, CAST(strftime('%H', ts_ms/1000, "
             "'unixepoch') AS INTEGER) hour, COUNT(*) requests, "
             "SUM(completion_tokens) tokens FROM request_history "
             "WHERE ts_ms > ?")
        args: list = [cutoff]
        if client_ip:
            q += " AND client_ip = ?"
            args.append(client_ip)
        q += " GROUP BY dow, hour"
        grid = [[0] * 24 for _ in range(7)]
        tokens = [[0] * 24 for _ in range(7)]
        with db() as conn:
            for r in conn.execute(q, args):
                grid[r["dow"]][r["hour"]] = r["requests"]
                tokens[r["dow"]][r["hour"]] = r["tokens"] or 0
        return {"days": days, "requests": grid, "tokens": tokens}

    @app.get("/api/dashboard/clients/{client_ip}")
    def client_drilldown(client_ip: str, days: int = Query(7, le=90),
                         user=Depends(auth_user)):
        """Per-IP drilldown (reference api/dashboard.rs per-client
        view): totals, per-model split, hourly timeline, recent
        requests."""
        cutoff = int(time.time() * 1000) - days * 86400 * 1000
        with db() as conn:
            tot = conn.execute(
                "SELECT COUNT(*) requests, SUM(prompt_tokens) pt, "
                "SUM(completion_tokens) ct, AVG(duration_ms) avg_ms, "
                "SUM(CASE WHEN status >= 400 THEN 1 ELSE 0 END) "
                "failures FROM request_history WHERE ts_ms > ? AND "
                "client_ip = ?", (cutoff, client_ip)).fetchone()
            models = conn.execute(
                "SELECT model, COUNT(*) requests, "
                "SUM(completion_tokens) tokens FROM request_history "
                "WHERE ts_ms > ? AND client_ip = ? GROUP BY model "
                "ORDER BY requests DESC", (cutoff, client_ip)).fetchall()
            hourly = conn.execute(
                "SELECT (ts_ms / 3600000) * 3600000 bucket, COUNT(*) "
                "requests FROM request_history WHERE ts_ms > ? AND "
                "client_ip = ? GROUP BY bucket ORDER BY bucket",
                (cutoff, client_ip)).fetchall()
            recent = conn.execute(
                "SELECT ts_ms, model, api_kind, status, duration_ms, "
                "completion_tokens FROM request_history WHERE "
                "client_ip = ? ORDER BY ts_ms DESC LIMIT 50",
                (client_ip,)).fetchall()
        if not tot or tot["requests"] == 0:
            raise HTTPException(404, "no requests from this client")
        return {"client_ip": client_ip, "totals": dict(tot),
                "models": [dict(m) for m in models],
                "hourly": [dict(h) for h in hourly],
                "recent": [dict(r) for r in recent]}

    @app.get("/api/dashboard/token-stats/monthly")
    def token_stats_monthly(months: int = Query(12, le=36),
                            user=Depends(auth_user)):
        """Monthly token rollups (reference api/dashboard.rs monthly
        token stats) aggregated from endpoint_daily_stats."""
        with db() as conn:
            rows = conn.execute(
                "SELECT substr(date, 1, 7) month, "
                "SUM(prompt_tokens) prompt_tokens, "
                "SUM(completion_tokens) completion_tokens, "
                "SUM(requests) requests, SUM(failures) failures "
                "FROM endpoint_daily_stats GROUP BY month "
                "ORDER BY month DESC LIMIT ?", (months,)).fetchall()
        return [dict(r) for r in rows]

    # ---------------- settings KV (reference db/settings.rs +
    # dashboard settings pages) ----------------
    @app.get("/api/settings")
    def get_settings(user=Depends(auth_user)):
        with db() as conn:
            conn.execute("CREATE TABLE IF NOT EXISTS settings ("
                         "key TEXT PRIMARY KEY, value TEXT NOT NULL)")
            rows = conn.execute("SELECT key, value FROM settings")
            return {r["key"]: r["value"] for r in rows}

    @app.put("/api/settings")
    def put_settings(body: dict, user=Depends(require_admin)):
        if not isinstance(body, dict) or len(body) > 100:
            raise HTTPException(400, "expected a flat key->value object")
        with db() as conn:
            conn.execute("CREATE TABLE IF NOT EXISTS settings ("
                         "key TEXT PRIMARY KEY, value TEXT NOT NULL)")
            for k, v in body.items():
                conn.execute(
                    "INSERT INTO settings (key, value) VALUES (?, ?) "
                    "ON CONFLICT(key) DO UPDATE SET value=excluded.value",
                    (str(k)[:128], str(v)[:4096]))
        return {"updated": len(body)}

    # ---------------- TPS benchmarks (api/benchmarks.rs) ----------------
    bench_results: dict[str, dict] = {}

    def _gateway_headers():
        return {"Authorization": f"Bearer {admin_token}",
                "Content-Type": "application/json"}

    @app.post("/api/benchmarks/tps", status_code=202)
    async def start_tps_benchmark(body: TpsBenchmarkBody,
                                  user=Depends(require_admin)):
        body.requests = min(body.requests, 500)       # caps (ref :25-34)
        body.concurrency = min(body.concurrency, 64)
        body.max_tokens = min(body.max_tokens, 4096)
        bench_id = secrets.token_hex(8)
        bench_results[bench_id] = {"status": "running",
                                   "model": body.model}

        async def run():
          try:
            import aiohttp
            payload = {"model": body.model,
                       "messages": [{"role": "user",
                                     "content": "Benchmark prompt."}],
                       "max_tokens": body.max_tokens,
                       "temperature": body.temperature}
            sem = asyncio.Semaphore(body.concurrency)
            tokens = 0
            errors = 0
            t0 = time.monotonic()

            async def one(session):
                nonlocal tokens, errors
                async with sem:
                    try:
                        async with session.post(
                                gateway_url + "/v1/chat/completions",
                                json=payload,
                                headers=_gateway_headers()) as resp:
                            data = await resp.json()
                            if resp.status == 200:
                                tokens += data.get("usage", {}).get(
                                    "completion_tokens", 0)
                            else:
                                errors += 1
                    except Exception:
                        errors += 1

            async with aiohttp.ClientSession() as session:
                await asyncio.gather(
                    *[one(session) for _ in range(body.requests)])
            dur = time.monotonic() - t0
            bench_results[bench_id] = {
                "status": "done", "model": body.model,
                "requests": body.requests, "errors": errors,
                "duration_s": round(dur, 3),
                "total_tokens": tokens,
                "tokens_per_sec": round(tokens / dur, 2) if dur else 0,
                "source": "benchmark",  # TpsSource::Benchmark
            }
          except Exception as e:  # a crashed run must not stay
            bench_results[bench_id] = {  # "running" forever
                "status": "failed", "model": body.model,
                "error": repr(e)}

        asyncio.get_event_loop().create_task(run())
        return {"benchmark_id": bench_id}

    @app.get("/api/benchmarks/tps/{bench_id}")
    def get_tps_benchmark(bench_id: str, user=Depends(require_admin)):
        if bench_id not in bench_results:
            raise HTTPException(404, "no such benchmark")
        return bench_results[bench_id]

    # ---------------- catalog (local, no egress) ----------------
    catalog_path = os.environ.get(
        "LLMLB_CATALOG_PATH",
        str(os.path.join(os.path.dirname(__file__), "model_catalog.json")))

    @app.get("/api/catalog/search")
    def catalog_search(q: str = "", limit: int = Query(20, le=100),
                       user=Depends(auth_user)):
        try:
            with open(catalog_path) as f:
                catalog = json.load(f)
        except OSError:
            catalog = []
        ql = q.lower()
        hits = [m for m in catalog
                if not ql or ql in m.get("id", "").lower()
                or ql in " ".join(m.get("tags", [])).lower()]
        return hits[:limit]

    # ---------------- dashboard WebSocket ----------------
    @app.websocket("/ws/dashboard")
    async def ws_dashboard(ws: WebSocket):
        token = ws.query_params.get("token", "")
        ok = (admin_token and token == admin_token) or \
            jwt_decode(token, jwt_secret)
        if not ok:
            await ws.close(code=4401)
            return
        await ws.accept()
        last_event_ms = int(time.time() * 1000)
        try:
            while True:
                # push a summary snapshot every 2 s plus any new gateway
                # events (EndpointStatusChanged / TpsUpdated — the
                # reference's DashboardEvents over /ws/dashboard)
                try:
                    req = urllib.request.Request(
                        gateway_url + "/api/dashboard/summary",
                        headers=_gateway_headers())
                    with urllib.request.urlopen(req, timeout=5) as resp:
                        summary = json.loads(resp.read())
                    await ws.send_json({"type": "summary",
                                        "data": summary,
                                        "ts": time.time()})
                    ereq = urllib.request.Request(
                        gateway_url +
                        f"/api/events?since={last_event_ms}",
                        headers=_gateway_headers())
                    with urllib.request.urlopen(ereq, timeout=5) as resp:
                        events = json.loads(resp.read())
                    for ev in events:
                        last_event_ms = max(last_event_ms, ev["ts_ms"])
                        await ws.send_json({"type": "event", "data": ev})
                except Exception as e:
                    await ws.send_json({"type": "error", "error": str(e)})
                await asyncio.sleep(2)
        except WebSocketDisconnect:
            return

    # ---------------- request-history CSV export (api/mod.rs:248) ----
    @app.get("/api/requests/export.csv")
    def export_requests_csv(hours: int = Query(24, le=720),
                            user=Depends(require_admin)):
        from fastapi.responses import PlainTextResponse
        cutoff = int(time.time() * 1000) - hours * 3600 * 1000
        with db() as conn:
            rows = conn.execute(
                "SELECT ts_ms,endpoint_id,model,api_kind,status,"
                "duration_ms,prompt_tokens,completion_tokens,client_ip,"
                "error FROM request_history WHERE ts_ms > ? ORDER BY ts_ms",
                (cutoff,)).fetchall()
        lines = ["ts_ms,endpoint_id,model,api_kind,status,duration_ms,"
                 "prompt_tokens,completion_tokens,client_ip,error"]
        for r in rows:
            lines.append(",".join(
                str(r[k]) if r[k] is not None else ""
                for k in r.keys()))
        return PlainTextResponse("\n".join(lines) + "\n",
                                 media_type="text/csv")

    # ---------------- log tail (api 'logs' component) ----------------
    @app.get("/api/logs")
    def tail_logs(lines: int = Query(200, le=5000),
                  user=Depends(require_admin)):
        log_dir = cfg.log_dir
        # newest dated file (daily rotation: llmlb-gw-YYYYMMDD.jsonl),
        # falling back to the legacy fixed name
        import glob as _glob
        dated = sorted(_glob.glob(
            os.path.join(log_dir, "llmlb-gw-*.jsonl")))
        path = dated[-1] if dated else os.path.join(log_dir,
                                                    "llmlb-gw.jsonl")
        out = []
        try:
            with open(path, "rb") as f:
                f.seek(0, 2)
                size = f.tell()
                f.seek(max(0, size - (1 << 20)))
                out = f.read().decode(errors="replace").splitlines()
        except OSError:
            pass
        return {"lines": out[-lines:]}

    # ---------------- model registry metadata (registry::models) -----
    def _ensure_models_table(conn):
        conn.execute(
            "CREATE TABLE IF NOT EXISTS models (id TEXT PRIMARY KEY, "
            "name TEXT, source TEXT, capabilities TEXT, "
            "size_bytes INTEGER, created_at TEXT)")

    @app.get("/api/models")
    def list_registered_models(user=Depends(auth_user)):
        with db() as conn:
            _ensure_models_table(conn)
            rows = conn.execute(
                "SELECT id,name,source,capabilities,size_bytes,created_at"
                " FROM models").fetchall()
        return [dict(r) for r in rows]

    @app.post("/api/models", status_code=201)
    def register_model(body: ModelRegBody, user=Depends(require_admin)):
        with db() as conn:
            _ensure_models_table(conn)
            conn.execute(
                "INSERT OR REPLACE INTO models (id,name,source,"
                "capabilities,size_bytes,created_at) VALUES (?,?,?,?,?,?)",
                (body.id, body.name or body.id, body.source,
                 body.capabilities, body.size_bytes,
                 time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())))
        return {"registered": body.id}

    @app.delete("/api/models/{model_id:path}")
    def delete_model(model_id: str, user=Depends(require_admin)):
        with db() as conn:
            cur = conn.execute("DELETE FROM models WHERE id=?",
                               (model_id,))
        if cur.rowcount == 0:
            raise HTTPException(404, "not registered")
        return {"deleted": model_id}

    # ---------------- self-update manager (update/ component) --------
    # No egress in this environment: updates come from a local artifact
    # path.  Flow mirrors the reference (update/mod.rs): check ->
    # drain the data plane -> apply (atomic binary replace with .bak
    # rollback) -> resume.  Restart is left to the process supervisor.
    update_state: dict = {"state": "idle", "history": []}

    @app.get("/api/update/status")
    def update_status(user=Depends(require_admin)):
        return update_state

    def _resolve_target(target_path: str) -> str:
        return target_path or os.path.join(
            os.path.dirname(os.path.dirname(os.path.dirname(
                os.path.abspath(__file__)))), "gateway", "bin", "llmlb-gw")

    def _start_apply(artifact_path: str, target: str) -> None:
        """Drain -> swap binary (+.bak) -> resume; shared by the
        immediate apply and the scheduler (reference
        api/system.rs:129-150, 218-330)."""
        import shutil
        update_state.update({"state": "draining"})

        def drain_and_apply():
            try:
                req = urllib.request.Request(
                    gateway_url + "/api/system/drain", data=b"{}",
                    headers=_gateway_headers())
                urllib.request.urlopen(req, timeout=10)
                # wait for in-flight requests to finish (300 s cap,
                # update/mod.rs:37)
                deadline = time.time() + 300
                while time.time() < deadline:
                    with urllib.request.urlopen(
                            gateway_url + "/api/metrics",
                            timeout=5) as resp:
                        text = resp.read().decode()
                    for line in text.splitlines():
                        if line.startswith("llmlb_active_requests"):
                            if int(float(line.split()[-1])) == 0:
                                deadline = 0
                    if deadline == 0:
                        break
                    time.sleep(1)
                update_state.update({"state": "applying"})
                bak = target + ".bak"
                if os.path.exists(target):
                    shutil.copy2(target, bak)
                shutil.copy2(artifact_path, target)
                os.chmod(target, 0o755)
                update_state.update({"state": "applied",
                                     "rollback": bak})
            except Exception as e:
                update_state.update({"state": "failed", "error": str(e)})
            finally:
                try:
                    req = urllib.request.Request(
                        gateway_url + "/api/system/resume", data=b"{}",
                        headers=_gateway_headers())
                    urllib.request.urlopen(req, timeout=10)
                except Exception:
                    pass
                update_state["history"] = (
                    update_state.get("history", [])
                    + [{"ts": time.time(),
                        "state": update_state["state"]}])[-20:]

        import threading as _threading
        _threading.Thread(target=drain_and_apply, daemon=True).start()

    @app.post("/api/update/apply", status_code=202)
    async def update_apply(body: UpdateBody, user=Depends(require_admin)):
        if not os.path.exists(body.artifact_path):
            raise HTTPException(400, "artifact not found")
        _start_apply(body.artifact_path, _resolve_target(body.target_path))
        return {"accepted": True}

    # scheduled update (reference api/system.rs:218-330): one pending
    # schedule at a time; a timer fires the same drain->apply path
    update_schedule: dict = {}

    @app.post("/api/update/schedule", status_code=202)
    def update_schedule_create(body: ScheduleBody,
                               user=Depends(require_admin)):
        import threading as _threading
        if update_schedule.get("timer") is not None:
            raise HTTPException(409, "a schedule already exists")
        if not os.path.exists(body.artifact_path):
            raise HTTPException(400, "artifact not found")
        delay = (body.at_epoch - time.time()) if body.at_epoch             else body.delay_s
        if delay is None or delay < 0:
            raise HTTPException(400, "schedule time is in the past")
        target = _resolve_target(body.target_path)

        def fire():
            update_schedule.clear()
            _start_apply(body.artifact_path, target)

        t = _threading.Timer(delay, fire)
        t.daemon = True
        update_schedule.update({"timer": t,
                                "artifact_path": body.artifact_path,
                                "target_path": target,
                                "scheduled_at": time.time() + delay})
        t.start()
        return {"scheduled": True,
                "scheduled_at": update_schedule["scheduled_at"]}

    @app.get("/api/update/schedule")
    def update_schedule_get(user=Depends(require_admin)):
        if update_schedule.get("timer") is None:
            return {"scheduled": False}
        return {"scheduled": True,
                "artifact_path": update_schedule["artifact_path"],
                "target_path": update_schedule["target_path"],
                "scheduled_at": update_schedule["scheduled_at"]}

    @app.delete("/api/update/schedule")
    def update_schedule_cancel(user=Depends(require_admin)):
        t = update_schedule.get("timer")
        if t is None:
            raise HTTPException(404, "no schedule")
        t.cancel()
        update_schedule.clear()
        return {"cancelled": True}

    @app.post("/api/update/rollback")
    def update_rollback(user=Depends(require_admin)):
        import shutil
        bak = update_state.get("rollback")
        if not bak or not os.path.exists(bak):
            raise HTTPException(400, "no rollback artifact")
        target = bak[:-4]
        shutil.copy2(bak, target)
        os.chmod(target, 0o755)
        update_state.update({"state": "rolled_back"})
        return {"rolled_back": True}

    @app.get("/api/control/health")
    def control_health():
        return {"status": "ok", "component": "control-plane"}

    # minimal dashboard landing page (the reference embeds a React SPA;
    # our ops surface is the REST + WS API above)
    @app.get("/dashboard")
    def dashboard_page():
        return JSONResponse({
            "dashboard": "llmlb_amd control plane",
            "endpoints": ["/api/dashboard/summary (data plane)",
                          "/api/dashboard/token-stats",
                          "/api/dashboard/clients",
                          "/api/dashboard/timeline",
                          "/ws/dashboard?token=..."],
        })

    return app


def main():
    import argparse

    import uvicorn
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=32700)
    args = ap.parse_args()
    uvicorn.run(create_app(), host=args.host, port=args.port,
                log_level="warning")


if __name__ == "__main__":
    main()
