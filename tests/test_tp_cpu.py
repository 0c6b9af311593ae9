"""Tensor-parallel math correctness on CPU (gloo, world_size 2).

Builds a TP=2 tiny Llama whose shards are slices of a master weight
set, runs a prefill forward, and compares logits against the TP=1
model holding the full master weights — verifying the column/row
splits and the all-reduce seams (the same math RCCL runs on xGMI).
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from pathlib import Path

from llmlb_amd.models.llama import LlamaConfig

REPO = Path(__file__).resolve().parents[1]


def _master_weights(cfg, seed=7):
    g = torch.Generator().manual_seed(seed)

    def w(*shape):
        return torch.randn(*shape, generator=g) * 0.02

    D = cfg.head_dim
    qkv_out = (cfg.q_heads + 2 * cfg.kv_heads) * D
    layers = []
    for _ in range(cfg.layers):
        layers.append({
            "qkv": w(qkv_out, cfg.hidden),
            "o": w(cfg.hidden, cfg.q_heads * D),
            "gate_up": w(2 * cfg.inter, cfg.hidden),
            "down": w(cfg.hidden, cfg.inter),
        })
    return {"embed": w(cfg.vocab, cfg.hidden),
            "lm_head": w(cfg.vocab, cfg.hidden), "layers": layers}


def _load_sharded(model, master, cfg, rank, ws):
    """Slice master weights into this rank's shards.

    Column-parallel splits must be PER-HEAD-GROUP for qkv (q|k|v each
    split across ranks) and per-half for gate_up (gate|up each split),
    matching how split_qkv/swiglu consume the local output."""
    D = cfg.head_dim
    model.embed.data.copy_(master["embed"].to(model.embed.dtype))
    model.lm_head.data.copy_(master["lm_head"].to(model.embed.dtype))
    for li, layer in enumerate(model.layers_):
        mw = master["layers"][li]
        qD, kvD = cfg.q_heads * D, cfg.kv_heads * D
        q_full = mw["qkv"][:qD]
        k_full = mw["qkv"][qD:qD + kvD]
        v_full = mw["qkv"][qD + kvD:]

        def shard_rows(t):
            n = t.shape[0] // ws
            return t[rank * n:(rank + 1) * n]

        qkv_local = torch.cat([shard_rows(q_full), shard_rows(k_full),
                               shard_rows(v_full)])
        layer.qkv.weight.data.copy_(qkv_local.to(layer.qkv.weight.dtype))
        # row-parallel o: input dim (q_heads*D) split
        n_in = qD // ws
        layer.o_proj.weight.data.copy_(
            mw["o"][:, rank * n_in:(rank + 1) * n_in]
            .to(layer.o_proj.weight.dtype))
        gate_full = mw["gate_up"][:cfg.inter]
        up_full = mw["gate_up"][cfg.inter:]
        gu_local = torch.cat([shard_rows(gate_full), shard_rows(up_full)])
        layer.gate_up.weight.data.copy_(
            gu_local.to(layer.gate_up.weight.dtype))
        n_in2 = cfg.inter // ws
        layer.down.weight.data.copy_(
            mw["down"][:, rank * n_in2:(rank + 1) * n_in2]
            .to(layer.down.weight.dtype))
        layer.input_norm_w.data.fill_(1.0)
        layer.post_norm_w.data.fill_(1.0)
    model.final_norm_w.data.fill_(1.0)


def _forward_logits(model, cfg, prompt):
    ids = torch.tensor(prompt, dtype=torch.long)
    pos = torch.arange(len(prompt), dtype=torch.int32)
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
    return model.final(h[-1:], residual[-1:])


def _worker(rank, world, port, result_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from llmlb_amd.models.llama import LlamaForCausalLM
        from llmlb_amd.parallel.tp import TPContext

        cfg = LlamaConfig.tiny()
        master = _master_weights(cfg)
        prompt = [3, 14, 15, 92, 6]

        tp = TPContext()
        model = LlamaForCausalLM(cfg, tp=tp, dtype=torch.float32,
                                 device="cpu")
        _load_sharded(model, master, cfg, rank, world)
        logits_tp = _forward_logits(model, cfg, prompt)

        if rank == 0:
            # TP=1 reference with full master weights
            class _One:
                world_size = 1
                rank = 0
                group = None

                def all_reduce(self, t):
                    return t

            ref = LlamaForCausalLM(cfg, tp=_One(), dtype=torch.float32,
                                   device="cpu")
            _load_sharded(ref, master, cfg, 0, 1)
            logits_ref = _forward_logits(ref, cfg, prompt)
            err = (logits_tp - logits_ref).abs().max().item()
            torch.save({"err": err}, result_path)
    finally:
        dist.destroy_process_group()


def test_tp2_matches_tp1(tmp_path):
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    result = str(tmp_path / "tp_result.pt")
    mp.spawn(_worker, args=(2, port, result), nprocs=2, join=True)
    out = torch.load(result)
    assert out["err"] < 1e-4, f"TP=2 logits diverge: {out['err']}"


def test_tp_coordinator_lockstep_2rank():
    """The lockstep coordinator (broadcast submissions + aborts +
    shutdown) over gloo world_size=2 matches a solo engine."""
    import subprocess
    import sys
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29761",
         "tests/helpers/tp_coord_check.py"],
        cwd=str(REPO), capture_output=True, text=True, timeout=240)
    assert res.returncode == 0, res.stdout[-1500:] + res.stderr[-1500:]
    assert "tp-coord-check OK" in res.stdout
