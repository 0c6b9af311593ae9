"""Tensor parallelism over torch.distributed.

On ROCm the ``nccl`` backend IS RCCL and collectives ride the 7-link
xGMI mesh (point-to-point, ≈153 GB/s per link), so a TP group must be
placed on xGMI-adjacent GPUs — see ``llmlb_amd.parallel.topology`` for
the quad-placement used by the gateway (BASELINE config 4).  CPU tests
use ``gloo`` with world_size > 1.

Sharding follows the usual Megatron split: attention qkv and MLP
gate_up are column-parallel (no comm), o_proj and down_proj are
row-parallel (one all-reduce each) — two all-reduces per layer.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from ..ops import linear_decode


class TPContext:
    """Process-group wrapper; world_size==1 means no TP (no comm)."""

    @classmethod
    def solo(cls) -> "TPContext":
        """A no-comm context for DP replicas living inside a larger
        torch.distributed world (e.g. bench.py ranks): without it, a
        replica would mistake the global world for its TP group."""
        ctx = cls.__new__(cls)
        ctx.group = None
        ctx.world_size = 1
        ctx.rank = 0
        return ctx

    def __init__(self, group=None):
        if group is not None:
            self.group = group
            self.world_size = dist.get_world_size(group)
            self.rank = dist.get_rank(group)
        elif dist.is_available() and dist.is_initialized():
            self.group = None
            self.world_size = dist.get_world_size()
            self.rank = dist.get_rank()
        else:
            self.group = None
            self.world_size = 1
            self.rank = 0

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.world_size > 1:
            dist.all_reduce(t, group=self.group)
        return t


_default_ctx: TPContext | None = None


def get_tp_context() -> TPContext:
    global _default_ctx
    if _default_ctx is None:
        _default_ctx = TPContext()
    return _default_ctx


def set_tp_context(ctx: TPContext) -> None:
    global _default_ctx
    _default_ctx = ctx


class _Fp8WeightMixin:
    """Opt-in quantized weight storage for the DECODE path:
    LLMLB_FP8_WEIGHTS=1 -> per-row e4m3 bytes + scales (the W8A16 arm
    of the staged decode GEMM, and the W8A8 MFMA path when
    LLMLB_FP8_A8 is also set); LLMLB_FP4_WEIGHTS=1 -> MXFP4 blocks
    for the W4A8 path.  Prefill and large batches always keep the
    bf16 weights + hipBLASLt."""

    _w8 = None
    _w4 = None

    def _w4_pair(self):
        """MXFP4 weight copy (LLMLB_FP4_WEIGHTS=1): packed e2m1
        nibbles + E8M0 block scales for the W4A8 decode GEMM."""
        from .. import ops as _ops
        if not _ops.w4_active() or not self.weight.is_cuda \
                or self.weight.dtype != torch.bfloat16:
            return None
        if self._w4 is None:
            K = self.weight.shape[1]
            if K % 512 != 0:
                return None
            self._w4 = _ops.quantize_fp4_weight(self.weight.data)
        return self._w4

    def _w8_pair(self, force: bool = False):
        from .. import ops as _ops
        if not (force or _ops._FP8_WEIGHTS) or not self.weight.is_cuda \
                or self.weight.dtype != torch.bfloat16:
            return None
        if self._w8 is None:
            self._w8 = _ops.quantize_fp8_weight(self.weight.data)
        return self._w8

    def forward_w8a8(self, xq: torch.Tensor, x_scale: torch.Tensor):
        """W8A8 (or W4A8 under LLMLB_FP4_WEIGHTS=1) decode GEMM on
        pre-quantized activations (rmsnorm_q / swiglu_q outputs).
        Returns None when the path does not apply (caller recomputes
        via the bf16 path)."""
        from .. import ops as _ops
        if xq.shape[0] > 256:
            return None
        w4 = self._w4_pair()
        if w4 is not None:
            try:
                return _ops.decode_gemm_w4a8(xq, x_scale, w4[0], w4[1],
                                             self.weight.shape[0])
            except ValueError:
                pass
        pair = self._w8_pair(force=True)
        if pair is None:
            return None
        try:
            return _ops.decode_gemm_w8a8(xq, x_scale, pair[0], pair[1])
        except ValueError:
            return None


class ColumnParallelLinear(nn.Module, _Fp8WeightMixin):
    """Y = X W^T with W split along the output dim; no communication.

    Output is the local shard [*, out_features / tp]."""

    def __init__(self, in_features: int, out_features: int, tp: TPContext,
                 dtype=torch.bfloat16, bias: bool = False):
        super().__init__()
        assert out_features % tp.world_size == 0, \
            f"{out_features} not divisible by tp={tp.world_size}"
        self.tp = tp
        self.out_local = out_features // tp.world_size
        self.weight = nn.Parameter(
            torch.empty(self.out_local, in_features, dtype=dtype),
            requires_grad=False)
        self.bias = nn.Parameter(
            torch.zeros(self.out_local, dtype=dtype),
            requires_grad=False) if bias else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 2 and x.shape[0] <= 128 and x.is_cuda:
            y = linear_decode(x, self.weight, w8=self._w8_pair())
            return y + self.bias if self.bias is not None else y
        from .. import ops as _ops
        # prefill fp8: use only PRE-BUILT w8 copies (self._w8) — the
        # engine init decides whether they fit (70B + W4 does not);
        # building them lazily here would bypass that HBM guard
        if (x.dim() == 2 and x.is_cuda and x.dtype == torch.bfloat16
                and _ops.fp8_prefill_active() and self._w8 is not None):
            y = _ops.scaled_mm_prefill(x, self._w8, self.weight)
            if y is not None:
                return y + self.bias if self.bias is not None else y
        return torch.nn.functional.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module, _Fp8WeightMixin):
    """Y = X W^T with W split along the input dim; all-reduce on output."""

    def __init__(self, in_features: int, out_features: int, tp: TPContext,
                 dtype=torch.bfloat16):
        super().__init__()
        assert in_features % tp.world_size == 0
        self.tp = tp
        self.in_local = in_features // tp.world_size
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_local, dtype=dtype),
            requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 2 and x.shape[0] <= 128 and x.is_cuda:
            y = linear_decode(x, self.weight, w8=self._w8_pair())
        else:
            from .. import ops as _ops
            y = None
            # pre-built w8 only (see ColumnParallelLinear.forward)
            if (x.dim() == 2 and x.is_cuda
                    and x.dtype == torch.bfloat16
                    and _ops.fp8_prefill_active()
                    and self._w8 is not None):
                y = _ops.scaled_mm_prefill(x, self._w8,
                                           self.weight)
            if y is None:
                y = torch.nn.functional.linear(x, self.weight)
        return self.tp.all_reduce(y)

    def forward_swiglu_w8a8(self, gateup: torch.Tensor):
        """swiglu_q (quantizing activation kernel) + W8A8 down GEMM;
        None when inapplicable."""
        from .. import ops as _ops
        if gateup.shape[0] > 256 or not gateup.is_cuda:
            return None
        q, sc = _ops.swiglu_q(gateup)
        y = self.forward_w8a8(q, sc)
        return self.tp.all_reduce(y) if y is not None else None

    def forward_swiglu(self, gateup: torch.Tensor) -> torch.Tensor:
        """down(silu(gate)*up) with the SwiGLU fused into the decode
        GEMM's x staging when the measured policy applies (each rank
        holds its own [gate_local | up_local] halves, so the fusion is
        TP-transparent)."""
        from .. import ops as _ops
        if gateup.dim() == 2 and gateup.shape[0] <= 128 and gateup.is_cuda:
            y = _ops.swiglu_linear(gateup, self.weight)
            if y is not None:
                return self.tp.all_reduce(y)
        return self.forward(_ops.swiglu(gateup))
