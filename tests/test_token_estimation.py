"""Token-estimation fallback: backends that omit the `usage` field get
their token counts BPE-estimated from the streamed text (reference
tiktoken cl100k_base fallback, llmlb/src/token/mod.rs:217-223) instead
of the per-event count."""
import json
import os
import socket
import subprocess
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


def _req(method, url, body=None, headers=None, timeout=10):
    data = json.dumps(body).encode() if body is not None else None
    r = urllib.request.Request(url, data=data, method=method,
                               headers=headers or {})
    with urllib.request.urlopen(r, timeout=timeout) as resp:
        return resp.status, resp.read().decode()


@pytest.fixture(scope="module")
def gw_nousage():
    mock_port, gw_port = _free_port(), _free_port()
    db = tempfile.mktemp(suffix=".sqlite3")
    mock = subprocess.Popen(
        [str(BIN / "llmlb-mock"), "--port", str(mock_port), "--workers",
         "1", "--models", "est", "--no-usage", "--stream-chunks", "40"],
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
    status, _ = _req("POST", base + "/api/endpoints",
                     {"name": "m", "base_url":
                      f"http://127.0.0.1:{mock_port}"},
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


def _expected_bpe_count(text):
    """Count with the SAME table the gateway ships, via the HF
    tokenizer machinery (inverse of gateway/tools/train_bpe.py)."""
    import base64
    path = REPO / "gateway" / "data" / "llmlb.tiktoken"
    ranks = {}
    for line in path.read_text().splitlines():
        tok, rank = line.rsplit(" ", 1)
        ranks[base64.b64decode(tok)] = int(rank)

    def bpe_count_piece(piece: bytes) -> int:
        if len(piece) <= 1 or piece in ranks:
            return 1
        sym = [piece[i:i + 1] for i in range(len(piece))]
        while len(sym) > 1:
            best = None
            for i in range(len(sym) - 1):
                cand = sym[i] + sym[i + 1]
                r = ranks.get(cand)
                if r is not None and (best is None or r < best[0]):
                    best = (r, i)
            if best is None:
                break
            i = best[1]
            sym[i:i + 2] = [sym[i] + sym[i + 1]]
        return len(sym)

    import regex
    pat = regex.compile(
        r"'s|'t|'re|'ve|'m|'ll|'d| ?\p{L}+| ?\p{N}+|"
        r" ?[^\s\p{L}\p{N}]+|\s+(?!\S)|\s+")
    return sum(bpe_count_piece(p.encode()) for p in pat.findall(text))


def test_streaming_estimate_close_to_bpe(gw_nousage):
    body = {"model": "est", "stream": True,
            "messages": [{"role": "user", "content": "hi"}]}
    status, _text = _req("POST", gw_nousage + "/v1/chat/completions",
                         body)
    assert status == 200
    time.sleep(0.5)  # history flush (200 ms loop)
    status, hist = _req("GET", gw_nousage + "/api/requests?limit=5",
                        headers={"Authorization": "Bearer t"})
    assert status == 200
    recs = json.loads(hist)
    rec = max(recs, key=lambda r: r["ts_ms"])   # ring is oldest-first
    got = rec["completion_tokens"]
    # streamed text is "tok0 tok1 ... tok39 " (40 deltas)
    text = "".join(f"tok{i} " for i in range(40))
    want = _expected_bpe_count(text)
    assert want > 40  # BPE count differs from the event count
    assert abs(got - want) / want <= 0.02, (got, want)


def test_nonstream_estimate(gw_nousage):
    body = {"model": "est",
            "messages": [{"role": "user", "content": "hi"}]}
    status, _ = _req("POST", gw_nousage + "/v1/chat/completions", body)
    assert status == 200
    time.sleep(0.5)
    status, hist = _req("GET", gw_nousage + "/api/requests?limit=5",
                        headers={"Authorization": "Bearer t"})
    recs = json.loads(hist)
    rec = max(recs, key=lambda r: r["ts_ms"])   # ring is oldest-first
    want = _expected_bpe_count("Hello from the mock backend.")
    assert abs(rec["completion_tokens"] - want) <= max(1, want * 0.02)
