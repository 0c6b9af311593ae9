"""Continuous-batching inference engine for the MI355X backend.

One engine per GPU (one process per GPU at node scale; TP replicas span
a process group over RCCL/xGMI).  Each ``step()``:

  1. admits queued prompts (prefill) while KV pages are available,
  2. runs one decode iteration for every running sequence through the
     paged-attention kernel (single batched launch per layer),
  3. samples, streams tokens to per-request queues, retires finished
     sequences and releases their pages.

Prefill attention uses torch SDPA (flash/math) on the prompt's own
K/V; the prompt K/V is simultaneously appended to the paged cache by
the gfx950 kv_append kernel.  Decode attention is the hand-written
paged kernel (llmlb_amd/ops/csrc/kernels.hip).
"""
from __future__ import annotations

import queue
import threading
import time
from dataclasses import dataclass, field

import torch

from .. import ops
from ..models.llama import LlamaConfig, LlamaForCausalLM
from .kv_cache import BLOCK_SIZE, KVCache
from .sampler import SamplingParams, sample


@dataclass
class EngineConfig:
    model: LlamaConfig
    device: str = "cuda:0"
    dtype: str = "bfloat16"
    kv_cache_blocks: int | None = None     # None = size from HBM budget
    hbm_fraction: float = 0.85             # of free HBM after weights
    max_batch: int = 256
    max_prefill_tokens: int = 8192
    tp_ctx: object = None
    seed: int = 1234


@dataclass
class GenerationResult:
    request_id: int
    token_ids: list[int] = field(default_factory=list)
    finish_reason: str = ""
    prompt_tokens: int = 0


class _Sequence:
    __slots__ = ("rid", "prompt_ids", "out_ids", "params", "block_table",
                 "length", "out_queue", "generator", "t_submit", "t_first")

    def __init__(self, rid, prompt_ids, params, out_queue, generator):
        self.rid = rid
        self.prompt_ids = prompt_ids
        self.out_ids: list[int] = []
        self.params = params
        self.block_table: list[int] = []
        self.length = 0                      # tokens with KV in cache
        self.out_queue = out_queue
        self.generator = generator
        self.t_submit = time.monotonic()
        self.t_first = 0.0


class Engine:
    def __init__(self, cfg: EngineConfig):
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        dtype = getattr(torch, cfg.dtype)
        self.model = LlamaForCausalLM(cfg.model, tp=cfg.tp_ctx, dtype=dtype,
                                      device=cfg.device, seed=cfg.seed)
        m = cfg.model
        n_kv_local = m.kv_heads // self.model.tp.world_size
        num_blocks = cfg.kv_cache_blocks or self._auto_blocks(n_kv_local)
        self.kv = KVCache(m.layers, num_blocks, n_kv_local, m.head_dim,
                          self.device, dtype)
        self.running: list[_Sequence] = []
        self.waiting: "queue.Queue[_Sequence]" = queue.Queue()
        self._next_rid = 0
        self._rid_lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None
        self.scale = m.head_dim ** -0.5
        self.n_q_local = m.q_heads // self.model.tp.world_size
        self.n_kv_local = n_kv_local
        # decode-attention workspace reused across steps
        self._ws = None
        self.steps_done = 0
        self.tokens_out = 0

    # -- sizing ----------------------------------------------------------
    def _auto_blocks(self, n_kv_local: int) -> int:
        m = self.cfg.model
        per_block = KVCache.bytes_per_block(m.layers, n_kv_local, m.head_dim)
        if self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            budget = int(free * self.cfg.hbm_fraction)
        else:
            budget = 64 * 1024 * 1024  # CPU tests: small
        return max(budget // per_block, 16)

    # -- public API ------------------------------------------------------
    def submit(self, prompt_ids: list[int],
               params: SamplingParams | None = None,
               sink=None) -> tuple[int, queue.Queue]:
        """Queue a request; returns (request_id, token sink).

        The sink receives .put((token_id, finish_reason|None)) calls from
        the engine thread; finish_reason is set on the final item.  Pass
        a custom ``sink`` (any object with put()) to bridge into asyncio
        (see server._AsyncSink)."""
        params = params or SamplingParams()
        with self._rid_lock:
            rid = self._next_rid
            self._next_rid += 1
        gen = None
        if params.seed is not None:
            gen = torch.Generator(device=self.device)
            gen.manual_seed(params.seed)
        out_q = sink if sink is not None else queue.Queue()
        seq = _Sequence(rid, list(prompt_ids), params, out_q, gen)
        self.waiting.put(seq)
        return rid, out_q

    def generate_sync(self, prompt_ids: list[int],
                      max_new_tokens: int = 16,
                      params: SamplingParams | None = None
                      ) -> GenerationResult:
        params = params or SamplingParams()
        params.max_new_tokens = max_new_tokens
        rid, out_q = self.submit(prompt_ids, params)
        res = GenerationResult(rid, prompt_tokens=len(prompt_ids))
        running_loop = self._thread is not None and self._thread.is_alive()
        while True:
            if not running_loop:
                self.step()
            try:
                tok, fin = out_q.get(block=running_loop, timeout=60.0)
            except queue.Empty:
                continue
            res.token_ids.append(tok)
            if fin:
                res.finish_reason = fin
                return res

    def start(self) -> None:
        """Run the engine loop in a background thread (server mode)."""
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="engine-loop")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=10)

    def _loop(self) -> None:
        while not self._stop.is_set():
            busy = self.step()
            if not busy:
                time.sleep(0.002)

    # -- scheduling ------------------------------------------------------
    def _admit(self) -> list[_Sequence]:
        admitted: list[_Sequence] = []
        budget = self.cfg.max_prefill_tokens
        while (len(self.running) + len(admitted) < self.cfg.max_batch
               and budget > 0):
            try:
                seq = self.waiting.get_nowait()
            except queue.Empty:
                break
            need = (len(seq.prompt_ids) + BLOCK_SIZE) // BLOCK_SIZE + 1
            if need > self.kv.free_blocks():
                self.waiting.put(seq)  # head-of-line wait for pages
                break
            if len(seq.prompt_ids) > budget and admitted:
                self.waiting.put(seq)
                break
            seq.block_table = self.kv.allocate(need)
            budget -= len(seq.prompt_ids)
            admitted.append(seq)
        return admitted

    def step(self) -> bool:
        """One scheduler iteration.  Returns True if any work was done."""
        prefills = self._admit()
        for seq in prefills:
            self._prefill(seq)
        if self.running:
            self._decode_step()
        self.steps_done += 1
        return bool(prefills or self.running)

    # -- forward passes --------------------------------------------------
    def _slots_for(self, seq: _Sequence, start: int, n: int) -> torch.Tensor:
        slots = [seq.block_table[(start + i) // BLOCK_SIZE] * BLOCK_SIZE
                 + (start + i) % BLOCK_SIZE for i in range(n)]
        return torch.tensor(slots, dtype=torch.int32, device=self.device)

    @torch.inference_mode()
    def _prefill(self, seq: _Sequence) -> None:
        T = len(seq.prompt_ids)
        ids = torch.tensor(seq.prompt_ids, dtype=torch.long,
                           device=self.device)
        positions = torch.arange(T, dtype=torch.int32, device=self.device)
        slots = self._slots_for(seq, 0, T)
        h = self.model.embed_tokens(ids)
        residual = None
        for li in range(self.cfg.model.layers):
            q, k, v, residual = self.model.layer_attn_prelude(
                li, h, residual, positions)
            ops.kv_append(self.kv.k[li], self.kv.v[li], k, v, slots)
            attn = self._sdpa_causal(q, k, v)
            h, residual = self.model.layer_attn_epilogue(li, attn, residual)
        logits = self.model.final(h[-1:], residual[-1:])
        seq.length = T
        self.running.append(seq)
        self._emit(seq, logits)

    def _sdpa_causal(self, q, k, v) -> torch.Tensor:
        """q: [T, n_q, D], k/v: [T, n_kv, D] -> [T, n_q, D]"""
        T = q.shape[0]
        qh = q.transpose(0, 1).unsqueeze(0)          # [1, n_q, T, D]
        kh = k.transpose(0, 1).unsqueeze(0)
        vh = v.transpose(0, 1).unsqueeze(0)
        o = torch.nn.functional.scaled_dot_product_attention(
            qh, kh, vh, is_causal=T > 1, enable_gqa=True)
        return o.squeeze(0).transpose(0, 1).contiguous()  # [T, n_q, D]

    @torch.inference_mode()
    def _decode_step(self) -> None:
        S = len(self.running)
        ids, positions, slots = [], [], []
        for seq in self.running:
            # grow block table when the next token crosses a page boundary
            if seq.length >= len(seq.block_table) * BLOCK_SIZE:
                seq.block_table.extend(self.kv.allocate(1))
            last = seq.out_ids[-1] if seq.out_ids else seq.prompt_ids[-1]
            ids.append(last)
            positions.append(seq.length)
            slots.append(seq.block_table[seq.length // BLOCK_SIZE]
                         * BLOCK_SIZE + seq.length % BLOCK_SIZE)
        dev = self.device
        ids_t = torch.tensor(ids, dtype=torch.long, device=dev)
        pos_t = torch.tensor(positions, dtype=torch.int32, device=dev)
        slots_t = torch.tensor(slots, dtype=torch.int32, device=dev)
        max_blocks = max(len(s.block_table) for s in self.running)
        bt = torch.zeros(S, max_blocks, dtype=torch.int32, device=dev)
        for i, s in enumerate(self.running):
            bt[i, :len(s.block_table)] = torch.tensor(
                s.block_table, dtype=torch.int32)
        seq_lens = torch.tensor([s.length + 1 for s in self.running],
                                dtype=torch.int32, device=dev)

        h = self.model.embed_tokens(ids_t)
        residual = None
        for li in range(self.cfg.model.layers):
            q, k, v, residual = self.model.layer_attn_prelude(
                li, h, residual, pos_t)
            ops.kv_append(self.kv.k[li], self.kv.v[li], k, v, slots_t)
            attn = ops.paged_attn_decode(q, self.kv.k[li], self.kv.v[li],
                                         bt, seq_lens, self.scale)
            h, residual = self.model.layer_attn_epilogue(li, attn, residual)
        logits = self.model.final(h, residual)
        for seq in self.running:
            seq.length += 1
        self._emit_batch(logits)

    # -- emission / retirement -------------------------------------------
    def _emit(self, seq: _Sequence, logits: torch.Tensor) -> None:
        tok = sample(logits, [seq.params], [seq.generator])
        if not self._push_token(seq, int(tok[0])):
            # finished on the very first token (max_new_tokens == 1)
            self.running.remove(seq)
            self.kv.release(seq.block_table)
            seq.block_table = []

    def _emit_batch(self, logits: torch.Tensor) -> None:
        toks = sample(logits, [s.params for s in self.running],
                      [s.generator for s in self.running])
        toks = toks.tolist()
        finished = []
        for seq, tok in zip(self.running, toks):
            if not self._push_token(seq, tok):
                finished.append(seq)
        for seq in finished:
            self.running.remove(seq)
            self.kv.release(seq.block_table)
            seq.block_table = []

    def _push_token(self, seq: _Sequence, tok: int) -> bool:
        """Returns False when the sequence is finished."""
        if not seq.out_ids:
            seq.t_first = time.monotonic()
        seq.out_ids.append(tok)
        self.tokens_out += 1
        fin = None
        if tok in seq.params.stop_token_ids:
            fin = "stop"
        elif len(seq.out_ids) >= seq.params.max_new_tokens:
            fin = "length"
        seq.out_queue.put((tok, fin))
        return fin is None

    @property
    def num_running(self) -> int:
        return len(self.running)
