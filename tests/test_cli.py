"""CLI lifecycle: serve (lock), status, assistant guard, stop."""
import json
import os
import signal
import socket
import subprocess
import sys
import time
import urllib.request
from pathlib import Path

REPO = Path(__file__).resolve().parents[1]


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_serve_status_stop(tmp_path):
    port = _free_port()
    env = os.environ.copy()
    env.update({"LLMLB_RUN_DIR": str(tmp_path / "run"),
                "LLMLB_DATABASE_URL": str(tmp_path / "db.sqlite3")})
    proc = subprocess.Popen(
        [sys.executable, "-m", "llmlb_amd.cli", "serve", "--port",
         str(port), "--db", str(tmp_path / "db.sqlite3"),
         "--control-plane-port", str(_free_port())],
        cwd=str(REPO), env=env, stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT)
    try:
        deadline = time.time() + 30
        up = False
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/health", timeout=2):
                    up = True
                    break
            except Exception:
                time.sleep(0.3)
        assert up, "gateway did not come up under the CLI"
        # status
        res = subprocess.run(
            [sys.executable, "-m", "llmlb_amd.cli", "status", "--port",
             str(port)], cwd=str(REPO), env=env, capture_output=True,
            text=True, timeout=15)
        out = json.loads(res.stdout)
        assert out["running"] is True
        # second serve refuses (single-instance lock)
        res2 = subprocess.run(
            [sys.executable, "-m", "llmlb_amd.cli", "serve", "--port",
             str(port), "--db", str(tmp_path / "db2.sqlite3"),
             "--control-plane-port", str(_free_port())],
            cwd=str(REPO), env=env, capture_output=True, text=True,
            timeout=20)
        assert res2.returncode == 1
        assert "already running" in res2.stderr
        # stop
        res3 = subprocess.run(
            [sys.executable, "-m", "llmlb_amd.cli", "stop", "--port",
             str(port)], cwd=str(REPO), env=env, capture_output=True,
            text=True, timeout=15)
        assert res3.returncode == 0
        proc.wait(timeout=20)
    finally:
        if proc.poll() is None:
            proc.send_signal(signal.SIGTERM)
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()


def test_assistant_blocks_bad_hosts():
    res = subprocess.run(
        [sys.executable, "-m", "llmlb_amd.cli", "assistant",
         "http://evil.example.com/x"], cwd=str(REPO),
        capture_output=True, text=True, timeout=15)
    assert res.returncode == 2
    assert "whitelist" in res.stderr


def test_assistant_blocks_injection():
    res = subprocess.run(
        [sys.executable, "-m", "llmlb_amd.cli", "assistant",
         "http://127.0.0.1/x;rm -rf /"], cwd=str(REPO),
        capture_output=True, text=True, timeout=15)
    assert res.returncode == 2
    assert "blocked" in res.stderr


def test_serve_with_backend_chat_e2e(tmp_path):
    """`llmlb serve --backends llama-tiny:x1` brings up the full node
    (gateway + control plane + backend replica) and serves a chat
    round trip through the gateway."""
    port, cp, be = _free_port(), _free_port(), _free_port()
    env = os.environ.copy()
    env.update({"LLMLB_RUN_DIR": str(tmp_path / "run"),
                "LLMLB_DATABASE_URL": str(tmp_path / "db.sqlite3"),
                "LLMLB_ADMIN_TOKEN": "clitok",
                "LLMLB_HEALTH_CHECK_INTERVAL": "2"})
    proc = subprocess.Popen(
        [sys.executable, "-m", "llmlb_amd.cli", "serve", "--port",
         str(port), "--db", str(tmp_path / "db.sqlite3"),
         "--control-plane-port", str(cp),
         "--backends", "llama-tiny:x1",
         "--backend-base-port", str(be)],
        cwd=str(REPO), env=env, stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT)
    try:
        body = json.dumps({
            "model": "llama-tiny", "max_tokens": 4,
            "messages": [{"role": "user", "content": "hi"}]}).encode()
        deadline = time.time() + 90
        ok = False
        while time.time() < deadline:
            try:
                req = urllib.request.Request(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    data=body,
                    headers={"Content-Type": "application/json"})
                with urllib.request.urlopen(req, timeout=10) as r:
                    out = json.loads(r.read())
                    if out["choices"][0]["message"]["content"]:
                        ok = True
                        break
            except Exception:
                time.sleep(1.0)
        assert ok, "chat through the CLI-supervised node never succeeded"
        # assistant helper: allowed local URL round-trips
        res = subprocess.run(
            [sys.executable, "-m", "llmlb_amd.cli", "assistant",
             f"http://127.0.0.1:{port}/health"], cwd=str(REPO), env=env,
            capture_output=True, text=True, timeout=20)
        assert res.returncode == 0, res.stderr
        assert "ok" in res.stdout or "status" in res.stdout
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
