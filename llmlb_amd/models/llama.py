"""Llama-3 family for the MI355X backend (bf16, TP-aware, paged KV).

This is the flagship model family served by the backends behind the
gateway (BASELINE configs 2-4: Llama-3-8B per-GPU replicas and
Llama-3-70B TP=4 quads).  GEMMs go through torch.nn.functional.linear
(hipBLASLt on ROCm); the fused hot ops (rmsnorm+residual, rope, swiglu,
paged decode attention, kv append) are the hand-written gfx950 kernels
in llmlb_amd.ops.

Weights are random-initialised (no network for checkpoints; BASELINE
prescribes synthetic prompts + random-init weights).
"""
from __future__ import annotations

from dataclasses import dataclass

import os

import torch
import torch.nn as nn

from .. import ops
from ..ops import reference as op_ref
from ..parallel.tp import (ColumnParallelLinear, RowParallelLinear,
                           TPContext, get_tp_context)


_QKV_VIEWS = os.environ.get("LLMLB_QKV_VIEWS", "1").lower() not in (
    "0", "false", "off")


@dataclass
class LlamaConfig:
    hidden: int = 4096
    layers: int = 32
    q_heads: int = 32
    kv_heads: int = 8
    head_dim: int = 128
    inter: int = 14336
    vocab: int = 128256
    rope_theta: float = 500000.0
    max_pos: int = 8192
    norm_eps: float = 1e-5
    # Qwen-2-style variants: bias on the fused qkv projection and
    # tied input/output embeddings
    qkv_bias: bool = False
    tie_embeddings: bool = False
    name: str = "llama-3-8b"

    @classmethod
    def llama3_8b(cls) -> "LlamaConfig":
        return cls()

    @classmethod
    def llama3_70b(cls) -> "LlamaConfig":
        return cls(hidden=8192, layers=80, q_heads=64, kv_heads=8,
                   head_dim=128, inter=28672, vocab=128256,
                   name="llama-3-70b")

    @classmethod
    def tiny(cls) -> "LlamaConfig":
        """2-layer Llama-3-shaped model for smoke/tests."""
        return cls(hidden=256, layers=2, q_heads=8, kv_heads=2, head_dim=32,
                   inter=512, vocab=1024, max_pos=512, name="llama-tiny")

    @classmethod
    def qwen2_7b(cls) -> "LlamaConfig":
        return cls(hidden=3584, layers=28, q_heads=28, kv_heads=4,
                   head_dim=128, inter=18944, vocab=152064,
                   rope_theta=1e6, max_pos=32768, qkv_bias=True,
                   tie_embeddings=False, name="qwen2-7b")

    @classmethod
    def qwen2_0_5b(cls) -> "LlamaConfig":
        return cls(hidden=896, layers=24, q_heads=14, kv_heads=2,
                   head_dim=64, inter=4864, vocab=151936,
                   rope_theta=1e6, max_pos=4096, qkv_bias=True,
                   tie_embeddings=True, name="qwen2-0.5b")

    @classmethod
    def small(cls) -> "LlamaConfig":
        """~0.5B-shaped model for 1-GPU quick benches."""
        return cls(hidden=1024, layers=8, q_heads=16, kv_heads=4,
                   head_dim=64, inter=2816, vocab=32000, max_pos=4096,
                   name="llama-small")


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp: TPContext, dtype):
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        ws = tp.world_size
        assert cfg.q_heads % ws == 0 and cfg.kv_heads % ws == 0, \
            "TP degree must divide q_heads and kv_heads"
        self.q_local = cfg.q_heads // ws
        self.kv_local = cfg.kv_heads // ws
        D = cfg.head_dim
        qkv_out = (cfg.q_heads + 2 * cfg.kv_heads) * D
        self.input_norm_w = nn.Parameter(
            torch.empty(cfg.hidden, dtype=dtype), requires_grad=False)
        self.post_norm_w = nn.Parameter(
            torch.empty(cfg.hidden, dtype=dtype), requires_grad=False)
        self.qkv = ColumnParallelLinear(cfg.hidden, qkv_out, tp, dtype,
                                        bias=cfg.qkv_bias)
        self.o_proj = RowParallelLinear(cfg.q_heads * D, cfg.hidden, tp,
                                        dtype)
        self.gate_up = ColumnParallelLinear(cfg.hidden, 2 * cfg.inter, tp,
                                            dtype)
        self.down = RowParallelLinear(cfg.inter, cfg.hidden, tp, dtype)
        self.scale = D ** -0.5

    def split_qkv(self, qkv: torch.Tensor):
        D = self.cfg.head_dim
        T = qkv.shape[0]
        q, k, v = qkv.split(
            [self.q_local * D, self.kv_local * D, self.kv_local * D], dim=-1)
        # strided views into the fused qkv output — rope / kv_append /
        # paged attention consume them with an explicit row stride, so
        # no .contiguous() copies (3 kernels/layer saved on the decode
        # hot path; profiles/r01_decode_gemm_analysis.md "elementwise").
        # LLMLB_QKV_VIEWS=0 restores the copies (A/B baseline).
        if _QKV_VIEWS:
            return (q.view(T, self.q_local, D),
                    k.view(T, self.kv_local, D),
                    v.view(T, self.kv_local, D))
        return (q.view(T, self.q_local, D).contiguous(),
                k.view(T, self.kv_local, D).contiguous(),
                v.view(T, self.kv_local, D).contiguous())


class LlamaForCausalLM(nn.Module):
    """Functional-style Llama; attention dispatch is owned by the engine
    (prefill SDPA vs paged decode), the model exposes per-layer pieces."""

    def __init__(self, cfg: LlamaConfig, tp: TPContext | None = None,
                 dtype: torch.dtype = torch.bfloat16,
                 device: str | torch.device = "cpu", seed: int = 1234,
                 init_weights: bool = True):
        super().__init__()
        self.cfg = cfg
        self.tp = tp or get_tp_context()
        self.dtype = dtype
        self.embed = nn.Parameter(
            torch.empty(cfg.vocab, cfg.hidden, dtype=dtype),
            requires_grad=False)
        self.layers_ = nn.ModuleList(
            [LlamaDecoderLayer(cfg, self.tp, dtype)
             for _ in range(cfg.layers)])
        self.final_norm_w = nn.Parameter(
            torch.empty(cfg.hidden, dtype=dtype), requires_grad=False)
        self.lm_head = nn.Parameter(
            torch.empty(cfg.vocab, cfg.hidden, dtype=dtype),
            requires_grad=False)
        self.cos_sin = op_ref.build_cos_sin_table(
            cfg.max_pos, cfg.head_dim, cfg.rope_theta)
        self.to(device)
        self.cos_sin = self.cos_sin.to(device)
        self.device = torch.device(device)
        if init_weights:          # skipped when a checkpoint follows
            self._init_weights(seed)

    def _init_weights(self, seed: int) -> None:
        # init on the target device (randn of 8B params on CPU is slow)
        g = torch.Generator(device=self.device).manual_seed(seed)
        std = 0.02

        def init_(p: torch.Tensor):
            p.data.copy_(torch.randn(p.shape, generator=g,
                                     device=self.device,
                                     dtype=torch.float32).mul_(std)
                         .to(p.dtype))

        init_(self.embed)
        if self.cfg.tie_embeddings:
            self.lm_head.data.copy_(self.embed.data)
        else:
            init_(self.lm_head)
        self.final_norm_w.data.fill_(1.0)
        for layer in self.layers_:
            layer.input_norm_w.data.fill_(1.0)
            layer.post_norm_w.data.fill_(1.0)
            init_(layer.qkv.weight)
            if layer.qkv.bias is not None:
                layer.qkv.bias.data.normal_(0.0, 0.02, generator=g)
            init_(layer.o_proj.weight)
            init_(layer.gate_up.weight)
            init_(layer.down.weight)

    @torch.inference_mode()
    def forward_dense(self, input_ids: torch.Tensor) -> torch.Tensor:
        """Plain causal forward over a [B, L] batch with NO KV cache —
        used for embeddings (backend /v1/embeddings): returns the
        final-norm hidden states [B, L, hidden]."""
        B, L = input_ids.shape
        dev = input_ids.device
        ids = input_ids.reshape(-1)
        positions = torch.arange(L, dtype=torch.int32,
                                 device=dev).repeat(B)
        nq, nkv, D = (self.cfg.q_heads // self.tp.world_size,
                      self.cfg.kv_heads // self.tp.world_size,
                      self.cfg.head_dim)
        h = self.embed_tokens(ids)
        residual = None
        for li in range(self.cfg.layers):
            q, k, v, residual = self.layer_attn_prelude(
                li, h, residual, positions)
            qh = q.view(B, L, nq, D).transpose(1, 2)
            kh = k.view(B, L, nkv, D).transpose(1, 2)
            vh = v.view(B, L, nkv, D).transpose(1, 2)
            o = torch.nn.functional.scaled_dot_product_attention(
                qh, kh, vh, is_causal=L > 1, enable_gqa=True)
            attn = o.transpose(1, 2).reshape(B * L, nq, D).contiguous()
            h, residual = self.layer_attn_epilogue(li, attn, residual)
        final = ops.rmsnorm(h + residual if residual is not None else h,
                            self.final_norm_w, self.cfg.norm_eps, None)[0]
        return final.view(B, L, -1)

    # -- checkpoint I/O ---------------------------------------------------

    @staticmethod
    def _shard(t: torch.Tensor, dim: int, ws: int, rank: int
               ) -> torch.Tensor:
        if ws == 1:
            return t
        size = t.shape[dim] // ws
        return t.narrow(dim, rank * size, size)

    def load_safetensors(self, path) -> int:
        """Load HuggingFace-format Llama weights (*.safetensors file,
        or a directory of shards).  Maps HF names onto the fused
        qkv / gate_up layouts and applies this rank's TP shard (column
        split for qkv/gate_up, input split for o_proj/down — Megatron
        convention, parallel/tp.py).  Returns the number of tensors
        consumed.  HF q/k use the same rotate-half RoPE layout as
        ops.rope_apply, so no permutation is needed."""
        from pathlib import Path as _P

        from safetensors import safe_open

        p = _P(path)
        files = sorted(p.glob("*.safetensors")) if p.is_dir() else [p]
        if not files:
            raise FileNotFoundError(f"no .safetensors under {path}")
        ws, rank = self.tp.world_size, self.tp.rank
        D = self.cfg.head_dim
        tensors: dict = {}
        n = 0
        for f in files:
            with safe_open(str(f), framework="pt") as sf:
                for name in sf.keys():
                    tensors[name] = sf.get_tensor(name)
                    n += 1

        def take(name):
            t = tensors[name]
            return t.to(self.dtype)

        def assign(param, t):
            assert param.shape == t.shape, (param.shape, t.shape)
            param.data.copy_(t.to(self.device))

        assign(self.embed, take("model.embed_tokens.weight"))
        assign(self.final_norm_w, take("model.norm.weight"))
        lm = ("lm_head.weight" if "lm_head.weight" in tensors
              else "model.embed_tokens.weight")   # tied embeddings
        assign(self.lm_head, take(lm))
        for i, layer in enumerate(self.layers_):
            pre = f"model.layers.{i}."
            assign(layer.input_norm_w, take(pre + "input_layernorm.weight"))
            assign(layer.post_norm_w,
                   take(pre + "post_attention_layernorm.weight"))
            q = self._shard(take(pre + "self_attn.q_proj.weight"), 0,
                            ws, rank)
            k = self._shard(take(pre + "self_attn.k_proj.weight"), 0,
                            ws, rank)
            v = self._shard(take(pre + "self_attn.v_proj.weight"), 0,
                            ws, rank)
            assign(layer.qkv.weight, torch.cat([q, k, v], dim=0))
            if layer.qkv.bias is not None:
                qb = self._shard(take(pre + "self_attn.q_proj.bias"), 0,
                                 ws, rank)
                kb = self._shard(take(pre + "self_attn.k_proj.bias"), 0,
                                 ws, rank)
                vb = self._shard(take(pre + "self_attn.v_proj.bias"), 0,
                                 ws, rank)
                assign(layer.qkv.bias, torch.cat([qb, kb, vb], dim=0))
            assign(layer.o_proj.weight,
                   self._shard(take(pre + "self_attn.o_proj.weight"), 1,
                               ws, rank))
            gate = self._shard(take(pre + "mlp.gate_proj.weight"), 0,
                               ws, rank)
            up = self._shard(take(pre + "mlp.up_proj.weight"), 0,
                             ws, rank)
            assign(layer.gate_up.weight, torch.cat([gate, up], dim=0))
            assign(layer.down.weight,
                   self._shard(take(pre + "mlp.down_proj.weight"), 1,
                               ws, rank))
        return n

    def save_safetensors(self, path) -> None:
        """Write HF-format weights (inverse of load; world_size 1 only
        — TP ranks hold shards, gather before saving)."""
        from safetensors.torch import save_file

        assert self.tp.world_size == 1, "save requires an unsharded model"
        D = self.cfg.head_dim
        out = {"model.embed_tokens.weight": self.embed.data,
               "model.norm.weight": self.final_norm_w.data,
               "lm_head.weight": self.lm_head.data}
        for i, layer in enumerate(self.layers_):
            pre = f"model.layers.{i}."
            qn, kn = self.cfg.q_heads * D, self.cfg.kv_heads * D
            qkv = layer.qkv.weight.data
            out[pre + "self_attn.q_proj.weight"] = qkv[:qn]
            out[pre + "self_attn.k_proj.weight"] = qkv[qn:qn + kn]
            out[pre + "self_attn.v_proj.weight"] = qkv[qn + kn:]
            if layer.qkv.bias is not None:
                qkvb = layer.qkv.bias.data
                out[pre + "self_attn.q_proj.bias"] = qkvb[:qn]
                out[pre + "self_attn.k_proj.bias"] = qkvb[qn:qn + kn]
                out[pre + "self_attn.v_proj.bias"] = qkvb[qn + kn:]
            out[pre + "self_attn.o_proj.weight"] = layer.o_proj.weight.data
            gu = layer.gate_up.weight.data
            out[pre + "mlp.gate_proj.weight"] = gu[:self.cfg.inter]
            out[pre + "mlp.up_proj.weight"] = gu[self.cfg.inter:]
            out[pre + "mlp.down_proj.weight"] = layer.down.weight.data
            out[pre + "input_layernorm.weight"] = layer.input_norm_w.data
            out[pre + "post_attention_layernorm.weight"] =                 layer.post_norm_w.data
        save_file({k: v.contiguous().cpu() for k, v in out.items()},
                  str(path))

    # -- forward pieces (called by the engine) ---------------------------

    def embed_tokens(self, input_ids: torch.Tensor) -> torch.Tensor:
        return self.embed[input_ids]

    def layer_attn_prelude(self, layer_idx: int, h: torch.Tensor,
                           residual: torch.Tensor | None,
                           positions: torch.Tensor, kv_sink=None,
                           quant_ok: bool = False):
        """norm -> qkv -> rope.  Returns (q, k, v, residual).

        ``kv_sink=(kcache, vcache, slots)`` fuses the paged-cache
        append into the RoPE launch (decode hot path: one kernel
        instead of two); the caller must then skip its own append."""
        layer = self.layers_[layer_idx]
        qkv = None
        # quant_ok: only the DECODE phase opts in — prefill chunks and
        # spec-verify stacks that happen to have T <= 128 measured a
        # 4.3x prefill slowdown on llama-3-70b (per-sequence quantize +
        # split-K overhead on tiny batches; profiles/r02_w8a8.md)
        if (quant_ok and ops.w8a8_active() and h.is_cuda and h.dim() == 2
                and h.shape[0] <= 128 and h.dtype == torch.bfloat16):
            # W8A8 decode arm: the rmsnorm emits the quantized row +
            # per-token scale directly; the fp8 MFMA GEMM consumes it
            nq, xs_, residual = ops.rmsnorm_q(h, layer.input_norm_w,
                                              self.cfg.norm_eps,
                                              residual)
            if residual is None:
                residual = h
            qkv = layer.qkv.forward_w8a8(nq, xs_)
            if qkv is not None and layer.qkv.bias is not None:
                qkv = qkv + layer.qkv.bias
        if qkv is None:
            normed, residual = ops.rmsnorm(h, layer.input_norm_w,
                                           self.cfg.norm_eps, residual)
            if residual is None:
                residual = h
            qkv = layer.qkv(normed)
        q, k, v = layer.split_qkv(qkv)
        if kv_sink is not None:
            kcache, vcache, slots = kv_sink
            ops.rope_kv_append(q, k, v, self.cos_sin, positions,
                               kcache, vcache, slots)
        else:
            q, k = ops.rope_apply(q, k, self.cos_sin, positions)
        return q, k, v, residual

    def layer_attn_epilogue(self, layer_idx: int, attn_out: torch.Tensor,
                            residual: torch.Tensor,
                            quant_ok: bool = False):
        """o_proj -> +residual -> norm -> mlp.  Returns (h, residual)."""
        layer = self.layers_[layer_idx]
        T = attn_out.shape[0]
        o = None
        if (quant_ok and ops.w8a8_active() and attn_out.is_cuda
                and T <= 128 and attn_out.dtype == torch.bfloat16):
            # o_proj joins the quantized path: one standalone row
            # quantize of the attention output (its producer has no
            # per-token amax), then the same W8A8/W4A8 GEMM
            a2 = attn_out.reshape(T, -1)
            if a2.is_contiguous():
                oq, osc = ops.rows_q(a2)
                o = layer.o_proj.forward_w8a8(oq, osc)
                if o is not None:
                    # RowParallelLinear: the mixin GEMM is shard-local
                    o = layer.o_proj.tp.all_reduce(o)
        if o is None:
            o = layer.o_proj(attn_out.reshape(T, -1))
        if (quant_ok and ops.w8a8_active() and o.is_cuda and T <= 128
                and o.dtype == torch.bfloat16):
            nq, xs_, residual = ops.rmsnorm_q(o, layer.post_norm_w,
                                              self.cfg.norm_eps,
                                              residual)
            gateup = layer.gate_up.forward_w8a8(nq, xs_)
            if gateup is not None:
                h = layer.down.forward_swiglu_w8a8(gateup)
                if h is not None:
                    return h, residual
                # fall back mid-path: finish with bf16 swiglu+down
                return layer.down.forward(ops.swiglu(gateup)), residual
        normed, residual = ops.rmsnorm(o, layer.post_norm_w,
                                       self.cfg.norm_eps, residual)
        gateup = layer.gate_up(normed)
        h = layer.down.forward_swiglu(gateup)
        return h, residual

    _lm8 = None
    _lm4 = None

    def final(self, h: torch.Tensor, residual: torch.Tensor,
              quant_ok: bool = False) -> torch.Tensor:
        # W8A8 lm_head on the decode path: the vocab projection is the
        # single largest per-step weight stream (1.05 GB bf16 on the
        # 128k llama-3 vocab) and the final rmsnorm is its natural
        # quantizing producer
        if (quant_ok and ops.w8a8_active() and h.is_cuda
                and h.dim() == 2 and h.shape[0] <= 128
                and h.dtype == torch.bfloat16):
            if (self._lm8 is None and self.lm_head.is_cuda
                    and not ops.w4_active()):
                self._lm8 = ops.quantize_fp8_weight(self.lm_head.data)
            if (ops.w4_active() and self._lm4 is None
                    and self.lm_head.is_cuda
                    and self.lm_head.shape[1] % 512 == 0):
                self._lm4 = ops.quantize_fp4_weight(self.lm_head.data)
            if self._lm8 is not None or self._lm4 is not None:
                nq, xs_, _ = ops.rmsnorm_q(h, self.final_norm_w,
                                           self.cfg.norm_eps, residual)
                if self._lm4 is not None:
                    try:
                        return ops.decode_gemm_w4a8(
                            nq, xs_, self._lm4[0], self._lm4[1],
                            self.lm_head.shape[0])
                    except ValueError:
                        pass
                if self._lm8 is not None:
                    try:
                        return ops.decode_gemm_w8a8(nq, xs_,
                                                    self._lm8[0],
                                                    self._lm8[1])
                    except ValueError:
                        pass
        normed, _ = ops.rmsnorm(h, self.final_norm_w, self.cfg.norm_eps,
                                residual)
        return torch.nn.functional.linear(normed, self.lm_head)
