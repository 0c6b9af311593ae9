"""Continuous-batching inference engine for the MI355X backend.

One engine per GPU (one process per GPU at node scale; TP replicas span
a process group over RCCL/xGMI).  Each ``step()``:

  1. admits queued prompts while KV pages are available and runs their
     prefill BATCHED (grouped by prompt length; one SDPA + one set of
     GEMMs per group instead of per sequence),
  2. runs one decode iteration for every running sequence through the
     paged-attention kernel — decode state (block tables, lengths, last
     tokens) lives in persistent device tensors so the per-step CPU
     work is one small gather, not a Python loop,
  3. samples, streams tokens to per-request sinks, retires finished
     sequences and releases their pages.

Prefill attention uses torch SDPA on the prompt's own K/V (flash path
on ROCm); decode attention is the hand-written gfx950 paged kernel
(llmlb_amd/ops/csrc/kernels.hip).

Optional serving features (all measured in profiles/):
  * automatic prefix caching (kv_cache.PrefixCache) — default on,
  * chunked prefill for long prompts (prefill_chunk),
  * prompt-lookup speculative decoding (spec_ngram / LLMLB_SPEC_NGRAM),
  * FP8 e4m3 KV cache (kv_dtype="fp8_e4m3"),
  * HF-safetensors checkpoint loading (weights_path),
  * request cancellation (abort()).
"""
from __future__ import annotations

import logging
import queue
import threading
import time
from collections import defaultdict
from dataclasses import dataclass, field

log = logging.getLogger("llmlb.engine")

import torch

from .. import ops
from ..models.llama import LlamaConfig, LlamaForCausalLM
from .kv_cache import BLOCK_SIZE, KVCache, PrefixCache
from .sampler import SamplingParams, sample


def os_env_flag(name: str, default: bool) -> bool:
    import os
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() not in ("0", "false", "no", "off")


@dataclass
class EngineConfig:
    model: LlamaConfig
    device: str = "cuda:0"
    dtype: str = "bfloat16"
    kv_cache_blocks: int | None = None     # None = size from HBM budget
    hbm_fraction: float = 0.85             # of free HBM after weights
    max_batch: int = 256
    max_prefill_tokens: int = 8192
    # automatic prefix caching: page-aligned prompt KV reuse
    # (kv_cache.PrefixCache); LLMLB_PREFIX_CACHE=0 disables
    enable_prefix_cache: bool = True
    # prompts longer than this prefill in chunks (activation-memory
    # bound for long prompts; keeps decode latency smooth)
    prefill_chunk: int = 4096
    # cap on prompt tokens ADMITTED per step (fairness: a decode step
    # runs between admission waves, bounding time-between-tokens for
    # running streams).  None = max_prefill_tokens.  Env
    # LLMLB_PREFILL_STEP_BUDGET overrides.
    prefill_step_budget: int | None = None
    # HF-format safetensors checkpoint (file or shard dir); None =
    # deterministic random init (synthetic serving)
    weights_path: str | None = None
    # KV cache storage: "bfloat16" (default) or "fp8_e4m3" (OCP e4m3
    # bytes + per-token-head scales: half the attention bandwidth,
    # double the resident context; needs D=128, GQA group <= 16 on
    # GPU).  Env LLMLB_KV_DTYPE overrides.
    kv_dtype: str = "bfloat16"
    # prompt-lookup speculative decoding: draft up to this many tokens
    # per step from the sequence's own n-gram history and verify them
    # in one batched multi-token pass (greedy sequences only).
    # 0 = off.  Env LLMLB_SPEC_NGRAM overrides.
    spec_ngram: int = 0
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
                 "length", "out_queue", "generator", "t_submit", "t_first",
                 "row", "cached_len", "hashes", "bigram_pos",
                 "trigram_pos", "spec_k", "max_new")

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
        self.row = -1                        # persistent decode-state row
        self.cached_len = 0                  # tokens reused from PrefixCache
        self.hashes: list[bytes] | None = None
        self.bigram_pos: dict | None = None  # spec: bigram -> last pos
        self.trigram_pos: dict | None = None  # spec: trigram -> last pos
        self.spec_k = 0                       # adaptive draft window
        self.max_new = params.max_new_tokens # clamped to max_pos at submit


class _DecodeGraph:
    """One hipGraph-captured decode forward at a fixed (padded) batch.

    The decode inner loop is launch-bound (~450 kernel launches per
    step at 32 layers); capturing it in a hipGraph replays the whole
    step at graph-replay cost (MI355X_MICROARCH.md row
    graph-replay-floor).  Padded rows point at the reserved sacrificial
    KV block 0 (their K/V writes and attention reads are discarded)."""

    BUCKETS = (8, 16, 32, 64, 128, 256)

    def __init__(self, engine: "Engine", S: int):
        self.S = S
        dev = engine.device
        MB = engine.max_blocks_per_seq
        self.ids = torch.zeros(S, dtype=torch.long, device=dev)
        self.lens = torch.zeros(S, dtype=torch.int32, device=dev)
        self.bt = torch.zeros(S, MB, dtype=torch.int32, device=dev)
        self.slots = torch.zeros(S, dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(S, dtype=torch.int32, device=dev)
        # fixed partition count per bucket (fills the 256-CU chip)
        base = S * engine.n_kv_local
        n_parts = 1
        while base * n_parts < 512 and n_parts < 16:
            n_parts *= 2
        self.n_parts = n_parts
        nq, D = engine.n_q_local, engine.cfg.model.head_dim
        group = nq // engine.n_kv_local
        use_v2 = ops._PAGED_ATTN_V2 and D == 128 and group <= 16
        slices = n_parts * 4 if use_v2 else n_parts
        if n_parts > 1:
            self.ws = (torch.empty(S, nq, slices, D, dtype=torch.float32,
                                   device=dev),
                       torch.empty(S, nq, slices, 2, dtype=torch.float32,
                                   device=dev))
        else:
            self.ws = None
        self.graph: torch.cuda.CUDAGraph | None = None
        self.logits: torch.Tensor | None = None
        self._engine = engine

    def _forward(self) -> torch.Tensor:
        eng = self._engine
        # captured: bakes the decode arms into the graph; reset in the
        # finally below so an exception during capture cannot leave
        # the flag set for a later prefill
        ops.set_decode_phase(True)
        try:
            h = eng.model.embed_tokens(self.ids)
            residual = None
            for li in range(eng.cfg.model.layers):
                q, k, v, residual = eng.model.layer_attn_prelude(
                    li, h, residual, self.lens,
                    kv_sink=(None if eng.kv.fp8 else
                             (eng.kv.k[li], eng.kv.v[li], self.slots)),
                    quant_ok=True)
                if eng.kv.fp8:
                    eng._kv_append(li, k, v, self.slots)
                attn = ops.paged_attn_decode(
                    q, eng.kv.k[li], eng.kv.v[li], self.bt,
                    self.seq_lens, eng.scale, n_parts=self.n_parts,
                    workspace=self.ws,
                    k_scale=(eng.kv.k_scale[li] if eng.kv.fp8
                             else None),
                    v_scale=(eng.kv.v_scale[li] if eng.kv.fp8
                             else None))
                h, residual = eng.model.layer_attn_epilogue(
                    li, attn, residual, quant_ok=True)
            return eng.model.final(h, residual, quant_ok=True)
        finally:
            ops.set_decode_phase(False)

    def capture(self) -> None:
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):  # warmup (allocator, hipBLASLt algo pick)
                self._forward()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.logits = self._forward()

    def run(self, ids, lens, bt, slots, seq_lens) -> torch.Tensor:
        n = ids.shape[0]
        self.ids[:n] = ids
        self.lens[:n] = lens
        self.bt[:n] = bt
        self.slots[:n] = slots
        self.seq_lens[:n] = seq_lens
        if n < self.S:
            # padded rows: sacrificial block 0, length 0
            self.lens[n:] = 0
            self.bt[n:] = 0
            self.slots[n:] = 0
            self.seq_lens[n:] = 1
        self.graph.replay()
        return self.logits[:n]


class Engine:
    def __init__(self, cfg: EngineConfig):
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        dtype = getattr(torch, cfg.dtype)
        self.model = LlamaForCausalLM(cfg.model, tp=cfg.tp_ctx, dtype=dtype,
                                      device=cfg.device, seed=cfg.seed,
                                      init_weights=not cfg.weights_path)
        if cfg.weights_path:
            self.model.load_safetensors(cfg.weights_path)
        if ((ops.w8a8_active() or ops._FP8_WEIGHTS)
                and self.device.type == "cuda"):
            # pre-quantize the fp8 weight copies NOW: lazily they are
            # built inside the first decode's graph capture, which the
            # first step (and any prefill-latency measurement around
            # it) absorbs — 0.3 s on llama-3-70b
            # under W4 the fp8 weight copies are dead weight (the
            # decode GEMMs take the fp4 path; only o_proj's W8A16 arm
            # still wants w8 and builds it lazily) — on llama-3-70b
            # pre-building both (140 GB bf16 + 70 GB fp8 + 17.5 GB
            # fp4) left no room for the KV cache
            w4_mode = ops.w4_active()
            # fp8 prefill reuses the per-row w8 copies; under W4 they
            # are otherwise unused, so only build them when the extra
            # half-weight footprint clearly fits (llama-3-70b + W4 +
            # w8 copies left no room for the KV cache)
            want_w8 = not w4_mode
            if w4_mode and ops.fp8_prefill_active():
                try:
                    free_b = torch.cuda.mem_get_info(self.device)[0]
                    w8_bytes = sum(
                        m.weight.numel() for m in self.model.modules()
                        if hasattr(m, "_w8_pair"))
                    want_w8 = w8_bytes < 0.3 * free_b
                except Exception:
                    want_w8 = False
            if w4_mode and not want_w8:
                # make the guard authoritative for the whole run: with
                # LLMLB_FP8_WEIGHTS also set, the decode branch's
                # non-force _w8_pair() would lazily rebuild the full
                # w8 set (~70 GB on llama-3-70b) during short prefill
                # fragments, bypassing this decision
                ops._FP8_WEIGHTS = False
            for mod in self.model.modules():
                if w4_mode and hasattr(mod, "_w4_pair"):
                    mod._w4_pair()
                if want_w8 and hasattr(mod, "_w8_pair"):
                    mod._w8_pair(force=True)
            if ops.fp8_prefill_active():
                # fill the fp8-vs-bf16 prefill table now (one-time,
                # ~1.5 s incl. hipBLASLt algo search); measuring
                # inline during a live prefill drains the launch-ahead
                # queue and was measured to cost ~2 s per 49k-token
                # prefill
                pairs = [(m.weight, m._w8)
                         for m in self.model.modules()
                         if getattr(m, "_w8", None) is not None]
                if pairs:
                    ops.tune_prefill_shapes(pairs, self.device)
            if ops.w8a8_active():
                try:
                    if w4_mode:
                        if (self.model._lm4 is None
                                and self.model.lm_head.shape[1] % 512
                                == 0):
                            self.model._lm4 = ops.quantize_fp4_weight(
                                self.model.lm_head.data)
                    elif self.model._lm8 is None:
                        self.model._lm8 = ops.quantize_fp8_weight(
                            self.model.lm_head.data)
                except Exception:
                    pass
        m = cfg.model
        n_kv_local = m.kv_heads // self.model.tp.world_size
        import os as _os0
        self.kv_dtype = _os0.environ.get("LLMLB_KV_DTYPE",
                                         cfg.kv_dtype) or "bfloat16"
        if (self.kv_dtype == "fp8_e4m3" and self.device.type == "cuda"
                and (m.head_dim != 128
                     or m.q_heads // m.kv_heads > 16)):
            raise ValueError(
                "fp8_e4m3 KV on GPU requires head_dim=128 and GQA "
                f"group <= 16 (got D={m.head_dim}, "
                f"group={m.q_heads // m.kv_heads})")
        num_blocks = cfg.kv_cache_blocks or self._auto_blocks(n_kv_local)
        self.kv = KVCache(m.layers, num_blocks, n_kv_local, m.head_dim,
                          self.device, dtype, kv_dtype=self.kv_dtype)
        import os as _os
        self.spec_ngram = int(_os.environ.get("LLMLB_SPEC_NGRAM",
                                              str(cfg.spec_ngram)) or 0)
        # decode piggyback at prefill-chunk boundaries (A/B knob)
        self._piggyback = _os.environ.get(
            "LLMLB_PREFILL_PIGGYBACK", "1").lower() not in (
                "0", "false", "off")
        self.spec_proposed = 0
        self.spec_accepted = 0
        self.prefix = (PrefixCache(self.kv)
                       if cfg.enable_prefix_cache
                       and os_env_flag("LLMLB_PREFIX_CACHE", True)
                       else None)
        self.running: list[_Sequence] = []
        self.waiting: "queue.Queue[_Sequence]" = queue.Queue()
        self._next_rid = 0
        self._rid_lock = threading.Lock()
        self._abort_rids: set[int] = set()
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None
        self.scale = m.head_dim ** -0.5
        self.n_q_local = m.q_heads // self.model.tp.world_size
        self.n_kv_local = n_kv_local
        self.steps_done = 0
        self.tokens_out = 0
        # observability: hipEvent-tagged decode-step GPU time (sampled
        # every 32 steps) and a tokens/sec EMA — both surfaced through
        # /api/health so the gateway can seed its TPS estimate
        # (BASELINE: hipEvent queue-depth tagging + backend-fed TPS)
        self.gpu_step_ms_ema = 0.0
        self.tokens_per_sec_ema = 0.0
        self._tps_window_t0 = time.monotonic()
        self._tps_window_tokens = 0
        self._ev_start = None
        self._ev_end = None

        # persistent decode state (device-resident)
        self.max_rows = cfg.max_batch
        self.max_blocks_per_seq = min(
            (m.max_pos + BLOCK_SIZE - 1) // BLOCK_SIZE, num_blocks)
        dev = self.device
        self._bt_dev = torch.zeros(self.max_rows, self.max_blocks_per_seq,
                                   dtype=torch.int32, device=dev)
        self._len_dev = torch.zeros(self.max_rows, dtype=torch.int32,
                                    device=dev)
        self._last_tok = torch.zeros(self.max_rows, dtype=torch.long,
                                     device=dev)
        self._free_rows = list(range(self.max_rows - 1, -1, -1))
        self._rows_cache: torch.Tensor | None = None  # [S] long
        self._rows_dirty = True
        # hipGraph-captured decode steps, one per padded batch bucket
        self.use_graphs = (self.device.type == "cuda"
                           and self.model.tp.world_size == 1
                           and cfg.model.layers > 0
                           and os_env_flag("LLMLB_DECODE_GRAPHS", True))
        self._graphs: dict[int, _DecodeGraph] = {}
        self.timing = defaultdict(float) \
            if os_env_flag("LLMLB_ENGINE_TIMING", False) else None
        import os as _os2
        _sb = _os2.environ.get("LLMLB_PREFILL_STEP_BUDGET")
        self._step_budget = (int(_sb) if _sb
                             else (cfg.prefill_step_budget
                                   or cfg.max_prefill_tokens))

    # -- sizing ----------------------------------------------------------
    def _auto_blocks(self, n_kv_local: int) -> int:
        m = self.cfg.model
        per_block = KVCache.bytes_per_block(m.layers, n_kv_local,
                                            m.head_dim,
                                            kv_dtype=self.kv_dtype)
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

        The sink receives .put((token_id, finish_reason|None, extra))
        calls from the engine thread; finish_reason is set on the final
        item and ``extra`` carries per-token logprob info when
        params.logprobs > 0 (None otherwise).  Pass
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
        # a generation can never pass the model's context window: clamp
        # so page reservation stays within max_blocks_per_seq
        seq.max_new = max(1, min(params.max_new_tokens,
                                 self.cfg.model.max_pos
                                 - len(seq.prompt_ids) - 1))
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
                tok, fin, _extra = out_q.get(block=running_loop,
                                             timeout=60.0)
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
            try:
                busy = self.step()
            except Exception:  # noqa: BLE001 — engine thread must survive
                # A step() failure (OOM, kernel error, scheduler bug)
                # must not kill the daemon thread and hang every
                # in-flight and queued request: fail them all with an
                # "error" finish and keep serving.
                log.exception("engine step failed; failing in-flight work")
                self._fail_all_inflight("engine step error")
                busy = False
            if not busy:
                time.sleep(0.002)

    def _fail_all_inflight(self, msg: str) -> None:
        for seq in list(self.running):
            try:
                seq.out_queue.put((0, "error", {"error": msg}))
            except Exception:  # noqa: BLE001
                pass
            self._retire(seq)
        while True:
            try:
                seq = self.waiting.get_nowait()
            except queue.Empty:
                break
            try:
                seq.out_queue.put((0, "error", {"error": msg}))
            except Exception:  # noqa: BLE001
                pass

    # -- scheduling ------------------------------------------------------
    def _admit(self) -> list[_Sequence]:
        admitted: list[_Sequence] = []
        budget = self._step_budget
        while (len(self.running) + len(admitted) < self.cfg.max_batch
               and budget > 0 and self._free_rows):
            try:
                seq = self.waiting.get_nowait()
            except queue.Empty:
                break
            if len(seq.prompt_ids) > budget and admitted:
                self.waiting.put(seq)
                break
            # reserve worst-case pages (prompt + full generation) so a
            # mid-decode growth can never exhaust the pool and kill the
            # engine thread; with 288 GB of HBM the utilisation cost is
            # noise, and parked prefix pages remain evictable
            need = min((len(seq.prompt_ids) + seq.max_new
                        + BLOCK_SIZE) // BLOCK_SIZE + 1,
                       self.max_blocks_per_seq)
            if self.prefix is not None:
                if seq.hashes is None:
                    seq.hashes = PrefixCache.chain_hashes(seq.prompt_ids)
                # never reuse the whole prompt: at least one token must
                # prefill to produce the first logits
                usable = (len(seq.prompt_ids) - 1) // BLOCK_SIZE
                matched = self.prefix.match(seq.hashes[:usable])
                seq.cached_len = len(matched) * BLOCK_SIZE
                need_new = need - len(matched)
                if not self.prefix.ensure_free(need_new):
                    self.prefix.release_seq(matched)
                    seq.cached_len = 0
                    self.waiting.put(seq)  # head-of-line wait for pages
                    break
                seq.block_table = matched + self.kv.allocate(need_new)
            else:
                if need > self.kv.free_blocks():
                    self.waiting.put(seq)  # head-of-line wait for pages
                    break
                seq.block_table = self.kv.allocate(need)
            seq.row = self._free_rows.pop()
            budget -= len(seq.prompt_ids)
            admitted.append(seq)
        return admitted

    def abort(self, rid: int) -> None:
        """Cancel a request (client disconnect): the sequence is
        retired at the next step boundary and its pages are freed."""
        with self._rid_lock:
            self._abort_rids.add(rid)

    def _apply_aborts(self) -> None:
        with self._rid_lock:
            if not self._abort_rids:
                return
            rids = self._abort_rids
            self._abort_rids = set()
        for seq in [s for s in self.running if s.rid in rids]:
            self._retire(seq)
        # drop queued ones too (rebuild the waiting queue)
        pending = []
        while True:
            try:
                s = self.waiting.get_nowait()
            except queue.Empty:
                break
            if s.rid not in rids:
                pending.append(s)
        for s in pending:
            self.waiting.put(s)

    def step(self) -> bool:
        """One scheduler iteration.  Returns True if any work was done."""
        self._apply_aborts()
        prefills = self._admit()
        if prefills:
            self._prefill_batch(prefills)
        if self.running:
            if (self.spec_ngram > 0
                    and all(s.params.temperature < 1e-5
                            and not s.params.needs_slow_path()
                            for s in self.running)):
                self._spec_step()
            else:
                self._decode_step()
        self.steps_done += 1
        return bool(prefills or self.running)

    # -- forward passes --------------------------------------------------
    def _kv_append(self, li: int, k, v, slots) -> None:
        ops.kv_append(self.kv.k[li], self.kv.v[li], k, v, slots,
                      k_scale=(self.kv.k_scale[li] if self.kv.fp8
                               else None),
                      v_scale=(self.kv.v_scale[li] if self.kv.fp8
                               else None))

    def _slots_for(self, seq: _Sequence, start: int, n: int) -> list[int]:
        return [seq.block_table[(start + i) // BLOCK_SIZE] * BLOCK_SIZE
                + (start + i) % BLOCK_SIZE for i in range(n)]

    @torch.inference_mode()
    def _prefill_batch(self, seqs: list[_Sequence]) -> None:
        """Batched prefill, grouped by (prompt length, cached prefix):
        same-shape groups share one packed forward (GEMMs batched, one
        SDPA per group).  A group longer than ``prefill_chunk`` runs in
        chunks — every chunk after the first behaves exactly like a
        cached-prefix suffix (its keys are already in the paged cache),
        which bounds activation memory for long prompts and keeps
        prefill waves from monopolising the GPU."""
        groups: dict[tuple[int, int], list[_Sequence]] = defaultdict(list)
        for s in seqs:
            groups[(len(s.prompt_ids), s.cached_len)].append(s)
        dev = self.device
        C = max(self.cfg.prefill_chunk, BLOCK_SIZE)
        for (L, P), group in groups.items():
            B = len(group)
            c0 = P
            while True:
                c1 = L if L - c0 <= C else c0 + C
                h, residual = self._prefill_chunk(group, B, c0, c1)
                if c1 == L:
                    break
                c0 = c1
                # decode piggyback: one decode step for
                # the running streams between prefill chunks, so a long
                # prompt no longer stalls everyone's time-between-tokens
                # for its whole multi-chunk prefill.  Intermediate chunk
                # activations are discarded anyway (only the KV writes
                # matter), so interleaving here is state-safe.
                if self.running and self._piggyback:
                    self._decode_step()
            Ls = L - c0                      # tokens in the final chunk
            last_idx = torch.arange(1, B + 1, device=dev) * Ls - 1
            logits = self.model.final(h[last_idx], residual[last_idx])
            toks = sample(logits, [s.params for s in group],
                          [s.generator for s in group],
                          histories=[(s.prompt_ids, s.out_ids)
                                     for s in group])
            # install decode state
            rows = torch.tensor([s.row for s in group], dtype=torch.long,
                                device=dev)
            bt_rows = torch.zeros(B, self.max_blocks_per_seq,
                                  dtype=torch.int32)
            for i, s in enumerate(group):
                bt_rows[i, :len(s.block_table)] = torch.tensor(
                    s.block_table, dtype=torch.int32)
                s.length = L
            self._bt_dev[rows] = bt_rows.to(dev)
            self._len_dev[rows] = L
            self._last_tok[rows] = toks
            if self.prefix is not None:
                # publish the freshly-filled full prompt pages (they are
                # immutable from here on: decode writes start in the next
                # page slot) and account the hit rate
                for s in group:
                    for b in range(P // BLOCK_SIZE, L // BLOCK_SIZE):
                        self.prefix.register(s.hashes[b], s.block_table[b])
                    self.prefix.hit_tokens += P
                    self.prefix.prompt_tokens += L
            self.running.extend(group)
            self._rows_dirty = True
            extras = self._logprob_extras(logits, toks, group)
            toks_cpu = toks.tolist()
            finished = []
            for s, tok, ex in zip(group, toks_cpu, extras):
                if not self._push_token(s, tok, ex):
                    finished.append(s)
            for s in finished:
                self._retire(s)

    def _prefill_chunk(self, group: list[_Sequence], B: int, P: int,
                       L: int) -> tuple[torch.Tensor, torch.Tensor]:
        """One prefill pass over prompt positions [P, L) for a
        same-shape group whose positions [0, P) already have KV in the
        paged cache (cached prefix pages and/or earlier chunks).
        Returns the final (h, residual) of the chunk's tokens."""
        dev = self.device
        Ls = L - P
        # fp8 prefill GEMMs only on P==0 chunks: continuation chunks
        # run the mask-path SDPA (fp32 math fallback) whose ~13 GB
        # buffers lose 30%+ wall time with the fp8 path interleaved
        # (profiles/r02_fp8_prefill_ab.md, same-box 8x6144 A/B)
        ops.set_prefill_fp8_ok(P == 0)
        ids = torch.tensor([s.prompt_ids[P:L] for s in group],
                           dtype=torch.long, device=dev).reshape(-1)
        positions = torch.arange(P, L, dtype=torch.int32, device=dev) \
            .repeat(B)
        slots = torch.tensor(
            [slot for s in group for slot in self._slots_for(s, P, Ls)],
            dtype=torch.int32, device=dev)
        if P:
            pslots = torch.tensor(
                [sl for s in group for sl in self._slots_for(s, 0, P)],
                dtype=torch.long, device=dev)
            pblks, poffs = pslots // BLOCK_SIZE, pslots % BLOCK_SIZE
            # rectangular causal mask: chunk query i sees keys 0..P+i
            mask = (torch.arange(L, device=dev)[None, :]
                    <= torch.arange(P, L, device=dev)[:, None])
        h = self.model.embed_tokens(ids)
        residual = None
        nq, nkv, D = self.n_q_local, self.n_kv_local, \
            self.cfg.model.head_dim
        for li in range(self.cfg.model.layers):
            q, k, v, residual = self.model.layer_attn_prelude(
                li, h, residual, positions)
            self._kv_append(li, k, v, slots)
            qh = q.view(B, Ls, nq, D).transpose(1, 2)
            kh = k.view(B, Ls, nkv, D)
            vh = v.view(B, Ls, nkv, D)
            if P:
                # gather this layer's cached K/V for positions [0, P)
                kpre, vpre = self.kv.gather(li, pblks, poffs)
                kh = torch.cat([kpre.view(B, P, nkv, D).to(kh.dtype), kh],
                               dim=1)
                vh = torch.cat([vpre.view(B, P, nkv, D).to(vh.dtype), vh],
                               dim=1)
            kh, vh = kh.transpose(1, 2), vh.transpose(1, 2)
            if P:
                # masked SDPA + enable_gqa has no fused kernel on ROCm
                # (the head-count mismatch forces torch's fp32 math
                # path: 35.4 ms/layer and ~13 GB of transient fp32 at
                # the 8x6144 shape).  Repeating KV to the full head
                # count keeps the efficient kernel: 6.1 ms/layer
                # measured (benchmarks/gqa_mask_probe.py); the ~50 us
                # copy is noise against the 29 ms saved.
                G = nq // nkv
                if G > 1:
                    kh = kh.repeat_interleave(G, dim=1)
                    vh = vh.repeat_interleave(G, dim=1)
                o = torch.nn.functional.scaled_dot_product_attention(
                    qh, kh, vh, attn_mask=mask)
            else:
                o = torch.nn.functional.scaled_dot_product_attention(
                    qh, kh, vh, is_causal=Ls > 1, enable_gqa=True)
            attn = o.transpose(1, 2).reshape(B * Ls, nq, D).contiguous()
            h, residual = self.model.layer_attn_epilogue(li, attn,
                                                         residual)
        return h, residual

    def _rows_tensor(self) -> torch.Tensor:
        if self._rows_dirty or self._rows_cache is None:
            self._rows_cache = torch.tensor(
                [s.row for s in self.running], dtype=torch.long,
                device=self.device)
            self._rows_dirty = False
        return self._rows_cache

    @torch.inference_mode()
    def _decode_step(self) -> None:
        _t0 = time.monotonic()
        # grow block tables where the NEXT token crosses a page boundary.
        # Admission reserves worst-case pages, so exhaustion here means a
        # logic bug or prefix-cache pressure — fail just the starved
        # sequence rather than raising out of the engine thread.
        grow_rows, grow_cols, grow_vals = [], [], []
        starved = []
        for seq in self.running:
            if seq.length >= len(seq.block_table) * BLOCK_SIZE:
                if self.prefix is not None:
                    self.prefix.ensure_free(1)
                if self.kv.free_blocks() < 1:
                    starved.append(seq)
                    continue
                blk = self.kv.allocate(1)[0]
                seq.block_table.append(blk)
                grow_rows.append(seq.row)
                grow_cols.append(len(seq.block_table) - 1)
                grow_vals.append(blk)
        for seq in starved:
            log.error("KV pool exhausted for rid=%s; failing sequence",
                      seq.rid)
            seq.out_queue.put((0, "error", {"error": "kv pages exhausted"}))
            self._retire(seq)
        if grow_rows:
            self._bt_dev[torch.tensor(grow_rows, device=self.device),
                         torch.tensor(grow_cols, device=self.device)] = \
                torch.tensor(grow_vals, dtype=torch.int32,
                             device=self.device)

        sample_gpu_time = (self.device.type == "cuda"
                           and self.steps_done % 32 == 0)
        if sample_gpu_time:
            if self._ev_start is None:
                self._ev_start = torch.cuda.Event(enable_timing=True)
                self._ev_end = torch.cuda.Event(enable_timing=True)
            self._ev_start.record()
        rows = self._rows_tensor()
        lens = self._len_dev[rows]                      # [S] int32
        ids = self._last_tok[rows]                      # [S] long
        bt = self._bt_dev[rows]                         # [S, MB] int32
        blk_idx = (lens // BLOCK_SIZE).long()
        slots = bt.gather(1, blk_idx.unsqueeze(1)).squeeze(1) \
            * BLOCK_SIZE + (lens % BLOCK_SIZE)
        seq_lens = lens + 1

        S = len(self.running)
        if self.timing is not None:
            torch.cuda.synchronize() if self.device.type == "cuda" else None
            self.timing["prep"] += time.monotonic() - _t0
            _t1 = time.monotonic()
        graph = self._graph_for(S)
        if graph is not None:
            logits = graph.run(ids, lens, bt, slots, seq_lens)
        else:
            ops.set_decode_phase(True)
            try:
                h = self.model.embed_tokens(ids)
                residual = None
                for li in range(self.cfg.model.layers):
                    q, k, v, residual = self.model.layer_attn_prelude(
                        li, h, residual, lens,
                        kv_sink=(None if self.kv.fp8 else
                                 (self.kv.k[li], self.kv.v[li], slots)),
                        quant_ok=True)
                    if self.kv.fp8:
                        self._kv_append(li, k, v, slots)
                    attn = ops.paged_attn_decode(
                        q, self.kv.k[li], self.kv.v[li], bt, seq_lens,
                        self.scale,
                        k_scale=(self.kv.k_scale[li] if self.kv.fp8
                                 else None),
                        v_scale=(self.kv.v_scale[li] if self.kv.fp8
                                 else None))
                    h, residual = self.model.layer_attn_epilogue(
                        li, attn, residual, quant_ok=True)
                logits = self.model.final(h, residual, quant_ok=True)
            finally:
                ops.set_decode_phase(False)
        if self.timing is not None:
            torch.cuda.synchronize() if self.device.type == "cuda" else None
            self.timing["forward"] += time.monotonic() - _t1
            _t2 = time.monotonic()
        toks = sample(logits, [s.params for s in self.running],
                      [s.generator for s in self.running],
                      histories=[(s.prompt_ids, s.out_ids)
                                 for s in self.running])
        self._last_tok[rows] = toks
        self._len_dev[rows] = lens + 1
        if sample_gpu_time:
            self._ev_end.record()
        toks_cpu = toks.tolist()
        if sample_gpu_time:
            ms = self._ev_start.elapsed_time(self._ev_end)
            self.gpu_step_ms_ema = (ms if self.gpu_step_ms_ema == 0
                                    else 0.2 * ms
                                    + 0.8 * self.gpu_step_ms_ema)
        if self.timing is not None:
            self.timing["sample_sync"] += time.monotonic() - _t2
            _t3 = time.monotonic()
        extras = self._logprob_extras(logits, toks, self.running)
        finished = []
        for seq, tok, ex in zip(self.running, toks_cpu, extras):
            seq.length += 1
            if not self._push_token(seq, tok, ex):
                finished.append(seq)
        for seq in finished:
            self._retire(seq)
        if self.timing is not None:
            self.timing["emit"] += time.monotonic() - _t3
            self.timing["steps"] += 1

    # -- prompt-lookup speculative decoding ------------------------------
    def _draft_for(self, seq: _Sequence) -> list[int]:
        """Prompt-lookup draft v2: match the trailing TRIGRAM first
        (higher precision), fall back to the bigram, and cap the draft
        by the sequence's ADAPTIVE window — grown on full acceptance,
        shrunk to the measured match length on a miss — so sequences
        in non-repetitive phases stop paying for long rejected drafts
        (profiles/r01_spec_decode.md follow-up)."""
        ctx = seq.prompt_ids + seq.out_ids
        if len(ctx) < 3:
            return []
        if seq.bigram_pos is None:
            seq.bigram_pos = {}
            seq.trigram_pos = {}
            for i in range(len(ctx) - 2):      # exclude trailing bigram
                seq.bigram_pos[(ctx[i], ctx[i + 1])] = i
            for i in range(len(ctx) - 3):
                seq.trigram_pos[(ctx[i], ctx[i + 1], ctx[i + 2])] = i
            seq.spec_k = min(4, self.spec_ngram) or self.spec_ngram
        pos = None
        if len(ctx) >= 4:
            tpos = seq.trigram_pos.get((ctx[-3], ctx[-2], ctx[-1]))
            if tpos is not None:
                pos = tpos + 3
        if pos is None:
            bpos = seq.bigram_pos.get((ctx[-2], ctx[-1]))
            if bpos is not None:
                pos = bpos + 2
        if pos is None:
            return []
        k = max(1, min(seq.spec_k or self.spec_ngram, self.spec_ngram))
        return ctx[pos:pos + k]

    def _adapt_spec_k(self, seq: _Sequence, proposed: int,
                      accepted: int) -> None:
        if proposed == 0:
            return
        if accepted == proposed:
            seq.spec_k = min(max(seq.spec_k * 2, seq.spec_k + 1),
                             self.spec_ngram)
        else:
            seq.spec_k = max(2, accepted + 1)

    def _index_token(self, seq: _Sequence) -> None:
        """Advance the n-gram indexes by the token just emitted."""
        if seq.bigram_pos is None:
            return
        ctx = seq.prompt_ids + seq.out_ids
        if len(ctx) >= 3:
            # the bigram ENDING at the previous token becomes eligible
            seq.bigram_pos[(ctx[-3], ctx[-2])] = len(ctx) - 3
        if len(ctx) >= 4 and seq.trigram_pos is not None:
            seq.trigram_pos[(ctx[-4], ctx[-3], ctx[-2])] = len(ctx) - 4

    @torch.inference_mode()
    def _spec_step(self) -> None:
        """One speculative step for ALL running (greedy) sequences:
        feed [last_token, draft...] padded to a uniform length, verify
        with one batched multi-token forward through the paged cache
        (rectangular causal masks), accept the longest matching prefix,
        emit accepted+1 tokens.  Rejected positions leave stale KV that
        the very next write at that slot overwrites (seq.length is the
        source of truth)."""
        dev = self.device
        seqs = self.running
        B = len(seqs)
        drafts = [self._draft_for(s) for s in seqs]
        # never let the fed window (last token + drafts) write KV past
        # the context window / per-seq page capacity
        cap = self.max_blocks_per_seq * BLOCK_SIZE
        for i, s in enumerate(seqs):
            room = min(self.cfg.model.max_pos, cap) - s.length - 1
            # also bound by what the KV pool can actually grow to right
            # now: pages beyond the current table must be allocatable,
            # else truncate the draft instead of raising mid-step
            # 
            free_room = (len(s.block_table) * BLOCK_SIZE
                         + self.kv.free_blocks() * BLOCK_SIZE
                         - s.length - 1)
            room = min(room, free_room)
            if len(drafts[i]) > max(room, 0):
                drafts[i] = drafts[i][:max(room, 0)]
        F = 1 + max((len(d) for d in drafts), default=0)
        if F == 1:
            self._decode_step()
            return
        # whole-batch page pre-check: per-seq free_room above cannot see
        # the OTHER sequences' concurrent growth, so verify the total
        # demand fits before touching the pool; plain decode needs only
        # the admission-time reservation and always proceeds
        total_need = sum(
            max(0, (s.length + F + BLOCK_SIZE - 1) // BLOCK_SIZE
                - len(s.block_table)) for s in seqs)
        if self.prefix is not None and total_need:
            self.prefix.ensure_free(total_need)
        if total_need > self.kv.free_blocks():
            self._decode_step()
            return
        last = self._last_tok[self._rows_tensor()].tolist()
        feed_rows, pos_rows, slot_rows = [], [], []
        for s, d, t0 in zip(seqs, drafts, last):
            # grow the block table to cover length+F new slots
            need_blocks = (s.length + F + BLOCK_SIZE - 1) // BLOCK_SIZE
            while len(s.block_table) < need_blocks:
                s.block_table.append(self.kv.allocate(1)[0])
            feed = [t0] + d + [t0] * (F - 1 - len(d))
            feed_rows.append(feed)
            pos_rows.append(list(range(s.length, s.length + F)))
            slot_rows.append(self._slots_for(s, s.length, F))
        ids = torch.tensor(feed_rows, dtype=torch.long,
                           device=dev).reshape(-1)
        positions = torch.tensor(pos_rows, dtype=torch.int32,
                                 device=dev).reshape(-1)
        slots = torch.tensor(slot_rows, dtype=torch.int32,
                             device=dev).reshape(-1)
        Lmax = max(s.length for s in seqs) + F
        # gather slots for key positions [0, Lmax) per seq (pad: slot 0,
        # masked out below)
        gslots = torch.zeros(B, Lmax, dtype=torch.long)
        for i, s in enumerate(seqs):
            sl = self._slots_for(s, 0, s.length)
            gslots[i, :s.length] = torch.tensor(sl, dtype=torch.long)
            gslots[i, s.length:s.length + F] = torch.tensor(
                slot_rows[i], dtype=torch.long)
        gslots = gslots.to(dev)
        gblks, goffs = gslots // BLOCK_SIZE, gslots % BLOCK_SIZE
        lens = torch.tensor([s.length for s in seqs], device=dev)
        qpos = (lens[:, None] + torch.arange(F, device=dev)[None, :])
        mask = (torch.arange(Lmax, device=dev)[None, None, :]
                <= qpos[:, :, None])              # [B, F, Lmax]
        mask = mask[:, None, :, :]                # broadcast over heads

        nq, nkv, D = self.n_q_local, self.n_kv_local, self.cfg.model.head_dim
        h = self.model.embed_tokens(ids)
        residual = None
        for li in range(self.cfg.model.layers):
            q, k, v, residual = self.model.layer_attn_prelude(
                li, h, residual, positions)
            self._kv_append(li, k, v, slots)
            kg, vg = self.kv.gather(li, gblks, goffs)
            qh = q.view(B, F, nq, D).transpose(1, 2)
            kh = kg.view(B, Lmax, nkv, D).to(q.dtype).transpose(1, 2)
            vh = vg.view(B, Lmax, nkv, D).to(q.dtype).transpose(1, 2)
            # masked SDPA + enable_gqa has no fused ROCm kernel (falls
            # to the fp32 math path); repeat KV to full heads instead
            # — same fix as the chunked-prefill attention above
            if nq != nkv:
                kh = kh.repeat_interleave(nq // nkv, dim=1)
                vh = vh.repeat_interleave(nq // nkv, dim=1)
            o = torch.nn.functional.scaled_dot_product_attention(
                qh, kh, vh, attn_mask=mask)
            attn = o.transpose(1, 2).reshape(B * F, nq, D).contiguous()
            h, residual = self.model.layer_attn_epilogue(li, attn,
                                                         residual)
        logits = self.model.final(h, residual).view(B, F, -1)
        preds = logits.argmax(dim=-1).tolist()     # [B][F]

        rows_idx, new_last, new_len = [], [], []
        finished = []
        for i, (s, d) in enumerate(zip(seqs, drafts)):
            m = 0
            while m < len(d) and preds[i][m] == d[m]:
                m += 1
            emit = preds[i][:m + 1]
            self.spec_proposed += len(d)
            self.spec_accepted += m
            self._adapt_spec_k(s, len(d), m)
            s.length += 1 + m
            rows_idx.append(s.row)
            new_last.append(emit[-1])
            new_len.append(s.length)
            done = False
            for tok in emit:
                if done:
                    break
                if not self._push_token(s, tok):
                    done = True
                self._index_token(s)
            if done:
                finished.append(s)
        rows_t = torch.tensor(rows_idx, dtype=torch.long, device=dev)
        self._last_tok[rows_t] = torch.tensor(new_last, dtype=torch.long,
                                              device=dev)
        self._len_dev[rows_t] = torch.tensor(new_len, dtype=torch.int32,
                                             device=dev)
        bt_rows = torch.zeros(len(seqs), self.max_blocks_per_seq,
                              dtype=torch.int32)
        for i, s in enumerate(seqs):
            bt_rows[i, :len(s.block_table)] = torch.tensor(
                s.block_table, dtype=torch.int32)
        self._bt_dev[rows_t] = bt_rows.to(dev)
        for s in finished:
            self._retire(s)

    def _graph_for(self, S: int) -> "_DecodeGraph | None":
        if not self.use_graphs:
            return None
        bucket = None
        for b in _DecodeGraph.BUCKETS:
            if S <= b <= self.max_rows:
                bucket = b
                break
        if bucket is None:
            return None
        g = self._graphs.get(bucket)
        if g is None:
            g = _DecodeGraph(self, bucket)
            try:
                g.capture()
            except Exception as e:  # graphs are an optimization only
                import logging
                logging.getLogger("llmlb.engine").warning(
                    "decode graph capture failed (bucket %d): %s", bucket,
                    e)
                self.use_graphs = False
                return None
            self._graphs[bucket] = g
        return g

    # -- emission / retirement -------------------------------------------
    def _retire(self, seq: _Sequence) -> None:
        if seq in self.running:
            self.running.remove(seq)
        if self.prefix is not None:
            self.prefix.release_seq(seq.block_table)
        else:
            self.kv.release(seq.block_table)
        seq.block_table = []
        if seq.row >= 0:
            self._free_rows.append(seq.row)
            seq.row = -1
        self._rows_dirty = True

    def _logprob_extras(self, logits: torch.Tensor, toks: torch.Tensor,
                        seqs: list[_Sequence]) -> list[dict | None]:
        """Per-token logprob payloads for sequences that asked for them
        (params.logprobs > 0).  Slow path: full-vocab log_softmax over
        just the requesting rows."""
        idxs = [i for i, s in enumerate(seqs) if s.params.logprobs > 0]
        extras: list[dict | None] = [None] * len(seqs)
        if not idxs:
            return extras
        sub = logits[idxs].float().log_softmax(dim=-1)
        toks_cpu = toks.tolist()
        for j, i in enumerate(idxs):
            k = min(seqs[i].params.logprobs, 20)
            row = sub[j]
            topv, topi = row.topk(k)
            extras[i] = {
                "logprob": row[toks_cpu[i]].item(),
                "top": list(zip(topi.tolist(), topv.tolist())),
            }
        return extras

    def _push_token(self, seq: _Sequence, tok: int,
                    extra: dict | None = None) -> bool:
        """Returns False when the sequence is finished."""
        if not seq.out_ids:
            seq.t_first = time.monotonic()
        seq.out_ids.append(tok)
        self.tokens_out += 1
        self._tps_window_tokens += 1
        now = time.monotonic()
        dt = now - self._tps_window_t0
        if dt >= 2.0:
            tps = self._tps_window_tokens / dt
            self.tokens_per_sec_ema = (tps if self.tokens_per_sec_ema == 0
                                       else 0.2 * tps
                                       + 0.8 * self.tokens_per_sec_ema)
            self._tps_window_t0 = now
            self._tps_window_tokens = 0
        fin = None
        if tok in seq.params.stop_token_ids:
            fin = "stop"
        elif len(seq.out_ids) >= seq.max_new:
            fin = "length"
        seq.out_queue.put((tok, fin, extra))
        return fin is None

    @property
    def num_running(self) -> int:
        return len(self.running)
