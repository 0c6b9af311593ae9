"""llmlb CLI: serve / stop / status / bench / assistant.

Mirrors the reference's CLI surface (cli/mod.rs: serve, stop, status,
assistant) for the MI355X build:

  python -m llmlb_amd.cli serve [--port N] [--backends SPEC]
  python -m llmlb_amd.cli stop | status
  python -m llmlb_amd.cli assistant -- curl-like safe request helper
  python -m llmlb_amd.cli bench    -- config-1 proxy benchmark

``serve`` starts the native data plane (gateway/bin/llmlb-gw), the
Python control plane, and optionally local GPU backend replicas
(--backends "llama-3-8b:xN" = one replica per GPU for N GPUs, pinned
device_index for amdsmi-driven admission).  A single-instance lock
(pid file keyed by port, stale-PID detection — reference lock/mod.rs)
guards double starts.
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import time
import urllib.error
import urllib.request
from pathlib import Path

REPO = Path(__file__).resolve().parents[1]
RUN_DIR = Path(os.environ.get("LLMLB_RUN_DIR", "/tmp/llmlb"))


def lock_path(port: int) -> Path:
    return RUN_DIR / f"llmlb-{port}.lock.json"


def read_lock(port: int) -> dict | None:
    try:
        info = json.loads(lock_path(port).read_text())
    except (OSError, json.JSONDecodeError):
        return None
    # stale-PID detection
    try:
        os.kill(info["pid"], 0)
    except (OSError, KeyError):
        return None
    return info


def write_lock(port: int, extra: dict) -> None:
    RUN_DIR.mkdir(parents=True, exist_ok=True)
    lock_path(port).write_text(json.dumps(
        {"pid": os.getpid(), "started_at": time.time(), "port": port,
         **extra}))


def _gw_binary() -> Path:
    p = REPO / "gateway" / "bin" / "llmlb-gw"
    if not p.exists():
        subprocess.run(["make", "-C", str(REPO / "gateway"), "-j",
                        str(os.cpu_count() or 4)], check=True)
    return p


def cmd_serve(args) -> int:
    if read_lock(args.port):
        print(f"llmlb already running on port {args.port} "
              f"(pid {read_lock(args.port)['pid']})", file=sys.stderr)
        return 1
    procs: list[subprocess.Popen] = []
    env = os.environ.copy()
    admin_token = env.setdefault("LLMLB_ADMIN_TOKEN",
                                 os.urandom(12).hex())
    cp_port = args.control_plane_port
    env.setdefault("LLMLB_LOG_DIR", "logs")
    env.update({
        "LLMLB_PORT": str(args.port),
        "LLMLB_DATABASE_URL": args.db,
        "LLMLB_CONTROL_PLANE_URL": f"http://127.0.0.1:{cp_port}",
        "LLMLB_GATEWAY_URL": f"http://127.0.0.1:{args.port}",
    })

    def shutdown(*_):
        for p in reversed(procs):
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        lock_path(args.port).unlink(missing_ok=True)
        sys.exit(0)

    signal.signal(signal.SIGINT, shutdown)
    signal.signal(signal.SIGTERM, shutdown)

    # 1) data plane
    procs.append(subprocess.Popen([str(_gw_binary())], env=env))
    # 2) control plane
    procs.append(subprocess.Popen(
        [sys.executable, "-m", "llmlb_amd.api.control_plane", "--port",
         str(cp_port)], env=env, cwd=str(REPO)))
    # 3) local backend replicas (optional)
    if args.backends:
        spec = args.backends  # e.g. "llama-3-8b:x2" or "llama-tiny:x1"
        model, _, count = spec.partition(":x")
        n = int(count or "1")
        import torch
        n_gpu = torch.cuda.device_count()
        for i in range(n):
            device = f"cuda:{i}" if i < n_gpu else "cpu"
            benv = env.copy()
            if i < n_gpu:
                benv["HIP_VISIBLE_DEVICES"] = str(i)
                device = "cuda:0"
            port = args.backend_base_port + i
            procs.append(subprocess.Popen(
                [sys.executable, "-m", "llmlb_amd.backend.launcher",
                 "--model", model, "--device", device, "--port",
                 str(port)], env=benv, cwd=str(REPO)))
        # register the replicas once the gateway is up
        time.sleep(1.5)
        for i in range(n):
            body = json.dumps({
                "name": f"{model}-{i}",
                "base_url": f"http://127.0.0.1:{args.backend_base_port+i}",
                "device_index": i}).encode()
            req = urllib.request.Request(
                f"http://127.0.0.1:{args.port}/api/endpoints", data=body,
                headers={"Authorization": f"Bearer {admin_token}"})
            for _ in range(60):
                try:
                    urllib.request.urlopen(req, timeout=10)
                    break
                except urllib.error.URLError:
                    time.sleep(1.0)
    write_lock(args.port, {"admin_token": admin_token,
                           "control_plane_port": cp_port})
    print(f"llmlb serving on :{args.port} (control plane :{cp_port}); "
          f"admin token: {admin_token}")
    while all(p.poll() is None for p in procs[:2]):
        time.sleep(1)
    shutdown()
    return 0


def cmd_stop(args) -> int:
    info = read_lock(args.port)
    if not info:
        print("not running")
        return 1
    os.kill(info["pid"], signal.SIGTERM)
    print(f"sent SIGTERM to pid {info['pid']}")
    return 0


def cmd_status(args) -> int:
    info = read_lock(args.port)
    if not info:
        print(json.dumps({"running": False}))
        return 1
    try:
        with urllib.request.urlopen(
                f"http://127.0.0.1:{args.port}/health", timeout=5) as r:
            health = json.loads(r.read())
    except Exception as e:
        health = {"error": str(e)}
    print(json.dumps({"running": True, "pid": info["pid"],
                      "port": info["port"], "health": health}, indent=1))
    return 0


# safe request helper (reference cli/assistant.rs: host whitelist +
# injection blocklist)
_ALLOWED_HOSTS = ("127.0.0.1", "localhost", "0.0.0.0")
_BLOCKED_PATTERNS = (";", "&&", "|", "`", "$(", "\n")


def cmd_assistant(args) -> int:
    url = args.url
    for pat in _BLOCKED_PATTERNS:
        if pat in url:
            print(f"blocked: url contains '{pat}'", file=sys.stderr)
            return 2
    host = url.split("//")[-1].split("/")[0].split(":")[0]
    if host not in _ALLOWED_HOSTS:
        print(f"blocked: host '{host}' not in whitelist "
              f"{_ALLOWED_HOSTS}", file=sys.stderr)
        return 2
    data = args.data.encode() if args.data else None
    req = urllib.request.Request(url, data=data, method=args.method)
    if args.key:
        req.add_header("Authorization", f"Bearer {args.key}")
    if data:
        req.add_header("Content-Type", "application/json")
    try:
        with urllib.request.urlopen(req, timeout=args.timeout) as resp:
            print(resp.read().decode())
        return 0
    except urllib.error.HTTPError as e:
        print(e.read().decode(), file=sys.stderr)
        return 1


def cmd_bench(args) -> int:
    script = REPO / "benchmarks" / "run_local.sh"
    return subprocess.run(["bash", str(script)],
                          env={**os.environ, "DUR": str(args.duration)}
                          ).returncode


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="llmlb")
    ap.add_argument("--version", action="version",
                    version="llmlb-amd 0.1.0 (MI355X-native)")
    sub = ap.add_subparsers(dest="cmd", required=True)

    sp = sub.add_parser("serve", help="start gateway + control plane")
    sp.add_argument("--port", type=int,
                    default=int(os.environ.get("LLMLB_PORT", 32768)))
    sp.add_argument("--db",
                    default=os.environ.get("LLMLB_DATABASE_URL",
                                           "llmlb.sqlite3"))
    sp.add_argument("--control-plane-port", type=int, default=32700)
    sp.add_argument("--backends", default="",
                    help="local replicas, e.g. 'llama-3-8b:x8'")
    sp.add_argument("--backend-base-port", type=int, default=9300)
    sp.set_defaults(fn=cmd_serve)

    for name, fn in (("stop", cmd_stop), ("status", cmd_status)):
        p = sub.add_parser(name)
        p.add_argument("--port", type=int,
                       default=int(os.environ.get("LLMLB_PORT", 32768)))
        p.set_defaults(fn=fn)

    pa = sub.add_parser("assistant", help="safe local request helper")
    pa.add_argument("url")
    pa.add_argument("--method", default="GET")
    pa.add_argument("--data", default="")
    pa.add_argument("--key", default="")
    pa.add_argument("--timeout", type=int, default=30)
    pa.set_defaults(fn=cmd_assistant)

    pb = sub.add_parser("bench", help="run the config-1 proxy benchmark")
    pb.add_argument("--duration", type=int, default=10)
    pb.set_defaults(fn=cmd_bench)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
