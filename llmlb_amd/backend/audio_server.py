"""Synthetic media backends: ASR (transcription), TTS (speech) and
image generation.

BASELINE config 5 runs a heterogeneous fleet (8B text + TTS + ASR
backends).  There is no egress for real Whisper/TTS checkpoints, so
these serve deterministic synthetic results with realistic shapes:

  * ASR `/v1/audio/transcriptions` — accepts multipart or raw audio,
    returns a deterministic transcription derived from the payload hash
    (length proportional to payload size).
  * TTS `/v1/audio/speech` — returns a WAV (PCM16, 24 kHz) sine-sweep
    whose duration tracks the input text length.

Both advertise their models via GET /v1/models with names the gateway's
capability inference maps to audio_transcription / audio_speech
(health.cpp sync_endpoint_models).

  * image `/v1/images/generations` — returns a deterministic
    procedurally-generated PNG (b64_json) whose pattern derives from
    the prompt hash (reference api/images.rs proxies the same surface).

  python -m llmlb_amd.backend.audio_server --kind asr --port 9500
  python -m llmlb_amd.backend.audio_server --kind tts --port 9501
  python -m llmlb_amd.backend.audio_server --kind image --port 9502
"""
from __future__ import annotations

import argparse
import hashlib
import io
import json
import math
import struct
import time
import wave
import zlib

WORDS = ("the quick brown fox jumps over a lazy dog near river stone "
         "light sound model audio gateway stream token").split()


def synth_transcription(payload: bytes) -> str:
    h = hashlib.blake2b(payload, digest_size=16).digest()
    n_words = max(4, min(len(payload) // 2000, 120))
    out = []
    for i in range(n_words):
        out.append(WORDS[h[i % 16] % len(WORDS)])
    return " ".join(out)


def synth_wav(text: str) -> bytes:
    rate = 24000
    dur_s = min(0.25 + 0.06 * len(text.split()), 10.0)
    n = int(rate * dur_s)
    buf = io.BytesIO()
    with wave.open(buf, "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(rate)
        seed = int.from_bytes(
            hashlib.blake2b(text.encode(), digest_size=4).digest(),
            "little")
        base = 180 + (seed % 220)
        frames = bytearray()
        for i in range(n):
            t = i / rate
            f = base + 40 * math.sin(2 * math.pi * 1.5 * t)
            s = int(12000 * math.sin(2 * math.pi * f * t))
            frames += struct.pack("<h", s)
        w.writeframes(bytes(frames))
    return buf.getvalue()


def synth_png(prompt: str, size: int = 64) -> bytes:
    """Minimal deterministic PNG (RGB, no deps beyond zlib)."""
    seed = hashlib.blake2b(prompt.encode(), digest_size=32).digest()
    raw = bytearray()
    for y in range(size):
        raw.append(0)  # filter: none
        for x in range(size):
            raw.append((x * 4 + seed[y % 32]) & 0xFF)
            raw.append((y * 4 + seed[x % 32]) & 0xFF)
            raw.append((seed[(x ^ y) % 32]) & 0xFF)

    def chunk(tag: bytes, data: bytes) -> bytes:
        return (struct.pack(">I", len(data)) + tag + data
                + struct.pack(">I", zlib.crc32(tag + data) & 0xFFFFFFFF))

    ihdr = struct.pack(">IIBBBBB", size, size, 8, 2, 0, 0, 0)
    return (b"\x89PNG\r\n\x1a\n" + chunk(b"IHDR", ihdr)
            + chunk(b"IDAT", zlib.compress(bytes(raw)))
            + chunk(b"IEND", b""))


def build_app(kind: str, model_name: str):
    served = [model_name]
    start = time.time()

    async def read_body(receive) -> bytes:
        chunks = []
        while True:
            msg = await receive()
            if msg["type"] == "http.request":
                chunks.append(msg.get("body", b""))
                if not msg.get("more_body"):
                    break
            elif msg["type"] == "http.disconnect":
                break
        return b"".join(chunks)

    async def send_bytes(send, status, ctype, data):
        await send({"type": "http.response.start", "status": status,
                    "headers": [(b"content-type", ctype.encode()),
                                (b"content-length",
                                 str(len(data)).encode())]})
        await send({"type": "http.response.body", "body": data})

    async def app(scope, receive, send):
        if scope["type"] == "lifespan":
            while True:
                msg = await receive()
                if msg["type"] == "lifespan.startup":
                    await send({"type": "lifespan.startup.complete"})
                elif msg["type"] == "lifespan.shutdown":
                    await send({"type": "lifespan.shutdown.complete"})
                    return
        if scope["type"] != "http":
            return
        path = scope["path"]
        if path == "/v1/models":
            caps = {"asr": "audio_transcription", "tts": "audio_speech",
                    "image": "image_generation"}[kind]
            body = json.dumps({"object": "list", "data": [
                {"id": m, "object": "model", "owned_by": "llmlb_amd",
                 "capabilities": caps}
                for m in served]}).encode()
            await send_bytes(send, 200, "application/json", body)
            return
        if path in ("/health", "/api/health"):
            body = json.dumps({"status": "ok", "kind": kind,
                               "uptime_s": time.time() - start}).encode()
            await send_bytes(send, 200, "application/json", body)
            return
        if kind == "asr" and path == "/v1/audio/transcriptions":
            payload = await read_body(receive)
            text = synth_transcription(payload)
            await send_bytes(send, 200, "application/json",
                             json.dumps({"text": text}).encode())
            return
        if kind == "image" and path == "/v1/images/generations":
            import base64
            payload = await read_body(receive)
            try:
                req = json.loads(payload or b"{}")
            except json.JSONDecodeError:
                req = {}
            n = int(req.get("n", 1) or 1)
            prompt = str(req.get("prompt", ""))
            data = [{"b64_json": base64.b64encode(
                        synth_png(f"{prompt}#{i}")).decode()}
                    for i in range(min(n, 4))]
            body = json.dumps({"created": int(time.time()),
                               "data": data}).encode()
            await send_bytes(send, 200, "application/json", body)
            return
        if kind == "tts" and path == "/v1/audio/speech":
            payload = await read_body(receive)
            try:
                text = json.loads(payload or b"{}").get("input", "")
            except json.JSONDecodeError:
                text = ""
            await send_bytes(send, 200, "audio/wav", synth_wav(text))
            return
        await send_bytes(send, 404, "application/json",
                         b'{"error":{"message":"not found"}}')

    return app


def main(argv=None):
    import uvicorn
    ap = argparse.ArgumentParser()
    ap.add_argument("--kind", choices=("asr", "tts", "image"),
                    required=True)
    ap.add_argument("--model", default=None)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9500)
    args = ap.parse_args(argv)
    model = args.model or {
        "asr": "whisper-large-v3-synth",
        "tts": "kokoro-tts-synth",
        "image": "stable-diffusion-xl-synth"}[args.kind]
    uvicorn.run(build_app(args.kind, model), host=args.host,
                port=args.port, log_level="warning", access_log=False)


if __name__ == "__main__":
    main()
