"""Backend replica launcher: one engine+server per GPU.

CLI (also used by bench.py and the gateway's node tooling):

  python -m llmlb_amd.backend.launcher --model llama-3-8b \
      --device cuda:0 --port 9200 [--served-name NAME] [--kv-blocks N]

Model configs resolve through LlamaConfig presets; TP replicas use
torchrun with --tp N (each rank binds cuda:LOCAL_RANK; rank 0 serves
HTTP and broadcasts per-step schedules — see parallel/tp.py).
"""
from __future__ import annotations

import argparse

import torch

from ..models.llama import LlamaConfig
from .engine import Engine, EngineConfig

PRESETS = {
    "llama-3-8b": LlamaConfig.llama3_8b,
    "llama-3-70b": LlamaConfig.llama3_70b,
    "qwen2-7b": LlamaConfig.qwen2_7b,
    "qwen2-0.5b": LlamaConfig.qwen2_0_5b,
    "llama-tiny": LlamaConfig.tiny,
    "llama-small": LlamaConfig.small,
}


def build_engine(model: str, device: str, kv_blocks: int | None = None,
                 max_batch: int = 256, tp_ctx=None,
                 hbm_fraction: float = 0.85,
                 weights: str | None = None,
                 max_prefill_tokens: int | None = None) -> Engine:
    """tp_ctx=None means 'infer from torch.distributed' — pass
    TPContext.solo() for DP replicas inside a larger world."""
    cfg_fn = PRESETS.get(model)
    if cfg_fn is None:
        raise SystemExit(f"unknown model preset '{model}' "
                         f"(have: {', '.join(PRESETS)})")
    mcfg = cfg_fn()
    dtype = "bfloat16" if device.startswith("cuda") else "float32"
    ecfg = EngineConfig(model=mcfg, device=device, dtype=dtype,
                        kv_cache_blocks=kv_blocks, max_batch=max_batch,
                        tp_ctx=tp_ctx, hbm_fraction=hbm_fraction,
                        weights_path=weights)
    if max_prefill_tokens:
        ecfg.max_prefill_tokens = max_prefill_tokens
    return Engine(ecfg)


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description="llmlb_amd backend replica")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--served-name", default=None,
                    help="model id to serve (defaults to preset name)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9200)
    ap.add_argument("--kv-blocks", type=int, default=None)
    ap.add_argument("--weights", default=None,
                    help="HF safetensors checkpoint (file or dir)")
    ap.add_argument("--tokenizer", default=None,
                    help="local tokenizer.json (tokenizers format)")
    ap.add_argument("--max-batch", type=int, default=256)
    args = ap.parse_args(argv)

    device = args.device
    if device is None:
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
    if device.startswith("cuda"):
        torch.cuda.set_device(device)
    engine = build_engine(args.model, device, args.kv_blocks,
                          args.max_batch, weights=args.weights)
    served = args.served_name or engine.cfg.model.name
    from .server import serve
    print(f"llmlb_amd backend: model={served} device={device} "
          f"port={args.port} kv_blocks={engine.kv.num_blocks}", flush=True)
    serve(engine, served, host=args.host, port=args.port,
          tokenizer_path=args.tokenizer)


if __name__ == "__main__":
    main()
