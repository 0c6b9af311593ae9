"""Anthropic /v1/messages tool-use conversion against a scripted
OpenAI-SSE backend (tool_calls deltas) — covers the re-encoder's
tool_use / input_json_delta paths (reference anthropic.rs:728-1298)."""
import json
import socket
import subprocess
import tempfile
import threading
import time
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
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


class ScriptedBackend(BaseHTTPRequestHandler):
    """OpenAI-compatible backend that answers chat with tool_calls."""
    last_request = {}

    def log_message(self, *a):
        pass

    def do_GET(self):
        if self.path == "/v1/models":
            body = json.dumps({"object": "list", "data": [
                {"id": "tool-model", "object": "model"}]}).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
        else:
            self.send_response(404)
            self.end_headers()

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        req = json.loads(self.rfile.read(n) or b"{}")
        type(self).last_request = req
        if req.get("stream"):
            self.send_response(200)
            self.send_header("Content-Type", "text/event-stream")
            self.send_header("Transfer-Encoding", "chunked")
            self.end_headers()

            def chunk(obj):
                line = ("data: " + json.dumps(obj) + "\n\n").encode()
                self.wfile.write(f"{len(line):x}\r\n".encode() + line +
                                 b"\r\n")
            base = {"id": "x", "object": "chat.completion.chunk",
                    "model": "tool-model"}
            chunk({**base, "choices": [{"index": 0, "delta":
                  {"role": "assistant"}, "finish_reason": None}]})
            chunk({**base, "choices": [{"index": 0, "delta": {
                  "tool_calls": [{"index": 0, "id": "call_1",
                                  "type": "function",
                                  "function": {"name": "get_weather",
                                               "arguments": ""}}]},
                  "finish_reason": None}]})
            chunk({**base, "choices": [{"index": 0, "delta": {
                  "tool_calls": [{"index": 0, "function":
                                  {"arguments":
                                   "{\"city\": \"Tokyo\"}"}}]},
                  "finish_reason": None}]})
            chunk({**base, "choices": [{"index": 0, "delta": {},
                  "finish_reason": "tool_calls"}],
                  "usage": {"prompt_tokens": 9, "completion_tokens": 5}})
            line = b"data: [DONE]\n\n"
            self.wfile.write(f"{len(line):x}\r\n".encode() + line +
                             b"\r\n0\r\n\r\n")
        else:
            resp = {"id": "x", "object": "chat.completion",
                    "model": "tool-model",
                    "choices": [{"index": 0, "message": {
                        "role": "assistant", "content": "",
                        "tool_calls": [{"id": "call_1",
                                        "type": "function",
                                        "function": {
                                            "name": "get_weather",
                                            "arguments":
                                                "{\"city\": \"Tokyo\"}"
                                        }}]},
                        "finish_reason": "tool_calls"}],
                    "usage": {"prompt_tokens": 9, "completion_tokens": 5}}
            body = json.dumps(resp).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)


@pytest.fixture(scope="module")
def stack():
    backend_port, gw_port = _free_port(), _free_port()
    httpd = ThreadingHTTPServer(("127.0.0.1", backend_port),
                                ScriptedBackend)
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
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
    time.sleep(0.5)
    req = urllib.request.Request(
        base + "/api/endpoints",
        data=json.dumps({"name": "tools",
                         "base_url":
                             f"http://127.0.0.1:{backend_port}"}).encode(),
        headers={"Authorization": "Bearer t"})
    for _ in range(30):
        try:
            urllib.request.urlopen(req, timeout=10)
            break
        except Exception:
            time.sleep(0.3)
    yield {"base": base}
    gw.terminate()
    httpd.shutdown()


ANTHROPIC_REQ = {
    "model": "tool-model",
    "max_tokens": 64,
    "messages": [{"role": "user", "content": "weather in Tokyo?"}],
    "tools": [{"name": "get_weather",
               "description": "Get the weather",
               "input_schema": {"type": "object", "properties": {
                   "city": {"type": "string"}}}}],
    "tool_choice": {"type": "auto"},
}


def _post(base, body, stream=False):
    b = dict(body)
    if stream:
        b["stream"] = True
    req = urllib.request.Request(
        base + "/v1/messages", data=json.dumps(b).encode(),
        headers={"anthropic-version": "2023-06-01",
                 "Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=15) as resp:
        return resp.status, resp.read().decode()


def test_tools_forwarded_in_openai_shape(stack):
    _post(stack["base"], ANTHROPIC_REQ)
    sent = ScriptedBackend.last_request
    assert sent["tools"][0]["type"] == "function"
    fn = sent["tools"][0]["function"]
    assert fn["name"] == "get_weather"
    assert fn["parameters"]["properties"]["city"]["type"] == "string"
    assert sent["tool_choice"] == "auto"


def test_tool_use_nonstream_response(stack):
    status, body = _post(stack["base"], ANTHROPIC_REQ)
    assert status == 200
    obj = json.loads(body)
    assert obj["stop_reason"] == "tool_use"
    blocks = [b for b in obj["content"] if b["type"] == "tool_use"]
    assert blocks and blocks[0]["name"] == "get_weather"
    assert blocks[0]["input"] == {"city": "Tokyo"}
    assert obj["usage"]["output_tokens"] == 5


def test_tool_use_streaming_reencode(stack):
    status, raw = _post(stack["base"], ANTHROPIC_REQ, stream=True)
    assert status == 200
    assert "event: content_block_start" in raw
    assert '"type":"tool_use"' in raw.replace(" ", "")
    assert '"input_json_delta"' in raw
    assert '"partial_json"' in raw
    assert '"stop_reason":"tool_use"' in raw.replace(" ", "")
    assert "event: message_stop" in raw


def test_tool_result_roundtrip(stack):
    req = dict(ANTHROPIC_REQ)
    req["messages"] = [
        {"role": "user", "content": "weather in Tokyo?"},
        {"role": "assistant", "content": [
            {"type": "tool_use", "id": "call_1", "name": "get_weather",
             "input": {"city": "Tokyo"}}]},
        {"role": "user", "content": [
            {"type": "tool_result", "tool_use_id": "call_1",
             "content": "22C sunny"}]},
    ]
    status, _ = _post(stack["base"], req)
    assert status == 200
    sent = ScriptedBackend.last_request
    roles = [m["role"] for m in sent["messages"]]
    assert "tool" in roles
    tool_msg = next(m for m in sent["messages"] if m["role"] == "tool")
    assert tool_msg["tool_call_id"] == "call_1"
    assert tool_msg["content"] == "22C sunny"
    asst = next(m for m in sent["messages"] if m["role"] == "assistant")
    assert asst["tool_calls"][0]["function"]["name"] == "get_weather"
