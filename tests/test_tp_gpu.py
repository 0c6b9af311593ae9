"""TP over RCCL on a real GPU: two ranks sharing cuda:0 (the 1-GPU CI
box) run the TP=2 sharded tiny model and must reproduce the TP=1
logits — exercising the actual RCCL all-reduce path the xGMI quads use
(dmabuf IPC; HSA_ENABLE_IPC_MODE_LEGACY=0 required on this pool)."""
import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = [pytest.mark.gpu,
              pytest.mark.timeout(240, method="thread")]


def _worker(rank, world, port, result_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    import torch.distributed as dist
    dist.init_process_group("nccl", rank=rank, world_size=world)
    torch.cuda.set_device(0)  # both ranks share the single CI GPU
    try:
        from tests.test_tp_cpu import _load_sharded, _master_weights
        from llmlb_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from llmlb_amd.parallel.tp import TPContext

        cfg = LlamaConfig.tiny()
        master = _master_weights(cfg)
        prompt = [3, 14, 15, 92, 6]
        tp = TPContext()
        model = LlamaForCausalLM(cfg, tp=tp, dtype=torch.float32,
                                 device="cuda:0")
        _load_sharded(model, master, cfg, rank, world)
        logits_tp = _forward_logits_dev(model, cfg, prompt, "cuda:0")
        if rank == 0:
            ref = LlamaForCausalLM(cfg, tp=TPContext.solo(),
                                   dtype=torch.float32, device="cuda:0")
            _load_sharded(ref, master, cfg, 0, 1)
            logits_ref = _forward_logits_dev(ref, cfg, prompt, "cuda:0")
            err = (logits_tp - logits_ref).abs().max().item()
            torch.save({"err": err}, result_path)
    finally:
        dist.destroy_process_group()


def _forward_logits_dev(model, cfg, prompt, dev):
    ids = torch.tensor(prompt, dtype=torch.long, device=dev)
    pos = torch.arange(len(prompt), dtype=torch.int32, device=dev)
    h = model.embed_tokens(ids)
    residual = None
    T = len(prompt)
    for li in range(cfg.layers):
        q, k, v, residual = model.layer_attn_prelude(li, h, residual, pos)
        qh = q.transpose(0, 1).unsqueeze(0)
        kh = k.transpose(0, 1).unsqueeze(0)
        vh = v.transpose(0, 1).unsqueeze(0)
        o = torch.nn.functional.scaled_dot_product_attention(
            qh, kh, vh, is_causal=T > 1, enable_gqa=True)
        attn = o.squeeze(0).transpose(0, 1).contiguous()
        h, residual = model.layer_attn_epilogue(li, attn, residual)
    return model.final(h[-1:], residual[-1:]).float().cpu()


def test_tp2_rccl_matches_tp1(tmp_path):
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    result = str(tmp_path / "tp_gpu.pt")
    try:
        mp.spawn(_worker, args=(2, port, result), nprocs=2, join=True)
    except Exception as e:  # RCCL may refuse 2 ranks on ONE device
        msg = str(e)
        if ("Duplicate GPU" in msg or "duplicate GPU" in msg
                or "invalid device" in msg or "unhandled system error"
                in msg):
            pytest.skip(f"RCCL cannot share one GPU across ranks: "
                        f"{msg[:120]}")
        raise
    out = torch.load(result)
    assert out["err"] < 1e-3, f"TP=2 RCCL logits diverge: {out['err']}"
