"""Tensor-parallel backend replica: N ranks, one engine shard each.

Launch (inside an xGMI-adjacent quad chosen by parallel.topology):

  HIP_VISIBLE_DEVICES=0,1,2,3 torchrun --standalone --nnodes=1 \
      --nproc-per-node 4 --local-addr 127.0.0.1 \
      -m llmlb_amd.backend.tp_runner --model llama-3-70b --port 9400

Scheduling model: every rank runs an IDENTICAL engine replica of the
scheduler state; rank 0 owns the HTTP server and broadcasts newly
submitted requests before each step (control over gloo), so all ranks
admit the same sequences in the same order and the per-layer tensor
collectives (RCCL all-reduce in RowParallelLinear) line up.  Sampling
is deterministic given identical full logits (lm_head is replicated),
so every rank steps the same token stream without further sync.
"""
from __future__ import annotations

import argparse
import os
import threading
import time

import torch
import torch.distributed as dist

from ..parallel.tp import TPContext
from .engine import Engine, EngineConfig
from .launcher import PRESETS
from .sampler import SamplingParams


class TPEngineCoordinator:
    """Rank-0 facade: queues submissions, broadcasts them each step."""

    def __init__(self, engine: Engine, group_cpu):
        self.engine = engine
        self.group_cpu = group_cpu
        self.rank = dist.get_rank()
        self._pending: list[tuple] = []
        self._pending_aborts: list[int] = []
        self._sinks: dict[int, object] = {}
        self._rid_map: dict[int, int] = {}   # coord rid -> engine rid
        self._lock = threading.Lock()
        self._next_rid = 0
        self._stop = False

    # HTTP-facing submit (rank 0 only)
    def submit(self, prompt_ids, params: SamplingParams | None = None,
               sink=None):
        from dataclasses import asdict
        params = params or SamplingParams()
        with self._lock:
            rid = self._next_rid
            self._next_rid += 1
            # broadcast the FULL params dict so every sampler feature
            # (top_k/min_p/penalties/logprobs/stops) works under TP
            self._pending.append((rid, list(prompt_ids), asdict(params)))
            if sink is not None:
                self._sinks[rid] = sink
        return rid, sink

    def abort(self, rid: int) -> None:
        """Cancellation must reach every rank in the same step, or the
        TP group's collectives desynchronise."""
        with self._lock:
            self._pending_aborts.append(rid)

    def stop(self) -> None:
        """Rank 0: broadcast a shutdown sentinel so EVERY rank leaves
        the lockstep loop together."""
        with self._lock:
            self._pending_aborts.append(-1)

    def _drain_pending(self) -> list[tuple]:
        with self._lock:
            out = self._pending
            self._pending = []
            return out

    def loop(self):
        """Lockstep engine loop on every rank."""
        while not self._stop:
            if self.rank == 0:
                with self._lock:
                    aborts = self._pending_aborts
                    self._pending_aborts = []
                batch = self._drain_pending()
                obj = [(batch, aborts)]
            else:
                obj = [None]
            dist.broadcast_object_list(obj, src=0, group=self.group_cpu)
            batch, aborts = obj[0]
            for rid, prompt_ids, pdict in batch:
                params = SamplingParams(**pdict)
                if params.seed is None:
                    params.seed = 12345 + rid
                params.stop_token_ids = tuple(params.stop_token_ids)
                sink = self._sinks.get(rid) if self.rank == 0 else None
                erid, q = self.engine.submit(prompt_ids, params,
                                             sink=sink)
                self._rid_map[rid] = erid
            for rid in aborts:
                if rid == -1:          # group shutdown sentinel
                    self._stop = True
                    continue
                erid = self._rid_map.get(rid)
                if erid is not None:
                    self.engine.abort(erid)
            busy = self.engine.step()
            if not busy and not batch:
                # idle: cheap sleep, staying in lockstep (all ranks see
                # the same empty batch + empty running set)
                time.sleep(0.002)


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-70b")
    ap.add_argument("--served-name", default=None)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9400)
    ap.add_argument("--kv-blocks", type=int, default=None)
    ap.add_argument("--max-batch", type=int, default=128)
    args = ap.parse_args(argv)

    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    have_gpu = torch.cuda.is_available()
    backend = "nccl" if have_gpu else "gloo"  # nccl IS RCCL on ROCm
    dist.init_process_group(backend)
    group_cpu = dist.new_group(backend="gloo")  # control-plane casts
    ndev = torch.cuda.device_count() if have_gpu else 0
    device = f"cuda:{local_rank % ndev}" if have_gpu else "cpu"
    if have_gpu:
        torch.cuda.set_device(device)

    tp = TPContext()
    mcfg = PRESETS[args.model]()
    ecfg = EngineConfig(
        model=mcfg, device=device,
        dtype="bfloat16" if have_gpu else "float32",
        kv_cache_blocks=args.kv_blocks, max_batch=args.max_batch,
        tp_ctx=tp)
    engine = Engine(ecfg)
    coord = TPEngineCoordinator(engine, group_cpu)

    if rank == 0:
        from .server import BackendServer, build_asgi_app
        import uvicorn

        served = args.served_name or mcfg.name
        server = BackendServer(engine, served)
        # route submissions through the coordinator so they broadcast
        server.engine = _SubmitProxy(engine, coord)  # type: ignore
        app = build_asgi_app(server)
        cfg = uvicorn.Config(app, host=args.host, port=args.port,
                             log_level="warning", access_log=False)
        usrv = uvicorn.Server(cfg)
        threading.Thread(target=usrv.run, daemon=True).start()
        print(f"llmlb_amd TP backend: model={served} tp={tp.world_size} "
              f"port={args.port}", flush=True)
    if rank == 0:
        # SIGTERM drains the whole group via the lockstep sentinel so
        # every rank leaves its collective loop together
        import signal as _sig
        _sig.signal(_sig.SIGTERM, lambda *_: coord.stop())
        _sig.signal(_sig.SIGINT, lambda *_: coord.stop())
    coord.loop()


class _SubmitProxy:
    """Engine facade handed to the HTTP server on rank 0: submissions
    go through the coordinator; stats attrs proxy to the engine."""

    def __init__(self, engine: Engine, coord: TPEngineCoordinator):
        self._engine = engine
        self._coord = coord

    def submit(self, prompt_ids, params=None, sink=None):
        return self._coord.submit(prompt_ids, params, sink)

    def abort(self, rid):
        self._coord.abort(rid)

    def __getattr__(self, name):
        return getattr(self._engine, name)


if __name__ == "__main__":
    main()
