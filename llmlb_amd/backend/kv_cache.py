"""Paged KV cache for the MI355X backend.

Layout per layer: [num_blocks, n_kv_local, block_size, head_dim] bf16 —
the layout the gfx950 paged-attention kernel consumes (one 4 KiB page
per (block, kv_head) at head_dim=128, read 16 B/lane).

Sized for 288 GB HBM3E per GPU: the engine asks
``blocks_for_memory_fraction`` how many pages fit the configured HBM
budget after weights.

``PrefixCache`` adds automatic prefix caching on top: full prompt
pages are content-addressed by a chained BLAKE2 hash of their token
ids, so a request whose prompt shares a page-aligned prefix with an
earlier one (shared system prompt, few-shot header, chat history)
reuses those KV pages and prefills only the suffix.  Retired pages
with no active users park in an LRU pool and are evicted back to the
free list on demand — with 288 GB of HBM3E the pool is effectively a
multi-gigabyte prompt cache.
"""
from __future__ import annotations

import hashlib
from collections import OrderedDict

import torch

BLOCK_SIZE = 16  # tokens per page (matches PA_BLOCK in kernels.hip)


class KVCache:
    """``kv_dtype="fp8_e4m3"`` stores K/V as OCP e4m3 bytes (gfx950's
    native FP8) with one fp32 scale per (page, head, token) — half the
    attention bandwidth and double the resident context of bf16."""

    def __init__(self, layers: int, num_blocks: int, n_kv: int,
                 head_dim: int, device, dtype=torch.bfloat16,
                 kv_dtype: str = "bfloat16"):
        self.block_size = BLOCK_SIZE
        self.num_blocks = num_blocks
        self.fp8 = kv_dtype == "fp8_e4m3"
        store = torch.uint8 if self.fp8 else dtype
        self.k = [torch.zeros(num_blocks, n_kv, BLOCK_SIZE, head_dim,
                              dtype=store, device=device)
                  for _ in range(layers)]
        self.v = [torch.zeros(num_blocks, n_kv, BLOCK_SIZE, head_dim,
                              dtype=store, device=device)
                  for _ in range(layers)]
        if self.fp8:
            self.k_scale = [torch.zeros(num_blocks, n_kv, BLOCK_SIZE,
                                        dtype=torch.float32,
                                        device=device)
                            for _ in range(layers)]
            self.v_scale = [torch.zeros(num_blocks, n_kv, BLOCK_SIZE,
                                        dtype=torch.float32,
                                        device=device)
                            for _ in range(layers)]
        else:
            self.k_scale = self.v_scale = None
        self.out_dtype = dtype
        # block 0 is reserved as the sacrificial target of padded rows
        # in hipGraph-captured decode steps (engine._DecodeGraph)
        self._free: list[int] = list(range(num_blocks - 1, 0, -1))

    def gather(self, li: int, blks: torch.Tensor, offs: torch.Tensor):
        """K/V rows at (block, offset) as compute-dtype tensors
        [N, n_kv, D] — dequantizing when the cache is fp8.  Used by the
        suffix-prefill / speculative verify gather paths."""
        k = self.k[li][blks, :, offs]
        v = self.v[li][blks, :, offs]
        if not self.fp8:
            return k, v
        ks = self.k_scale[li][blks, :, offs].unsqueeze(-1)
        vs = self.v_scale[li][blks, :, offs].unsqueeze(-1)
        kf = k.view(torch.float8_e4m3fn).to(torch.float32) * ks
        vf = v.view(torch.float8_e4m3fn).to(torch.float32) * vs
        return kf.to(self.out_dtype), vf.to(self.out_dtype)

    @staticmethod
    def bytes_per_block(layers: int, n_kv: int, head_dim: int,
                        dtype=torch.bfloat16, kv_dtype="bfloat16") -> int:
        if kv_dtype == "fp8_e4m3":
            return 2 * layers * n_kv * BLOCK_SIZE * (head_dim + 4)
        esize = torch.tensor([], dtype=dtype).element_size()
        return 2 * layers * n_kv * BLOCK_SIZE * head_dim * esize

    def free_blocks(self) -> int:
        return len(self._free)

    def allocate(self, n: int) -> list[int]:
        if n > len(self._free):
            raise MemoryError(f"KV cache exhausted: want {n} blocks, "
                              f"{len(self._free)} free")
        return [self._free.pop() for _ in range(n)]

    def release(self, blocks: list[int]) -> None:
        self._free.extend(blocks)


class PrefixCache:
    """Content-addressed reuse of full prompt KV pages.

    Equivalent in purpose to vLLM's automatic prefix caching; the
    reference gateway has no engine of its own, so this is part of the
    new MI355X serving backend (BASELINE.json GPU north star).  All
    bookkeeping is host-side and O(pages); the GPU never sees it.
    """

    def __init__(self, kv: KVCache):
        self.kv = kv
        self.by_hash: dict[bytes, int] = {}
        self.block_hash: dict[int, bytes] = {}
        self.refs: dict[int, int] = {}
        # zero-ref cached pages, oldest first (eviction order)
        self.lru: "OrderedDict[int, None]" = OrderedDict()
        self.hit_tokens = 0
        self.prompt_tokens = 0

    @staticmethod
    def chain_hashes(prompt_ids: list[int]) -> list[bytes]:
        """One digest per FULL page, chained on the previous page so a
        hash identifies the whole prefix up to and including its page."""
        out: list[bytes] = []
        h = b"genesis"
        for b in range(len(prompt_ids) // BLOCK_SIZE):
            chunk = prompt_ids[b * BLOCK_SIZE:(b + 1) * BLOCK_SIZE]
            d = hashlib.blake2b(digest_size=16)
            d.update(h)
            d.update(b" ".join(str(t).encode() for t in chunk))
            h = d.digest()
            out.append(h)
        return out

    def match(self, hashes: list[bytes]) -> list[int]:
        """Longest cached prefix; matched pages gain a reference."""
        blocks: list[int] = []
        for h in hashes:
            blk = self.by_hash.get(h)
            if blk is None:
                break
            blocks.append(blk)
        for blk in blocks:
            self.refs[blk] = self.refs.get(blk, 0) + 1
            self.lru.pop(blk, None)
        return blocks

    def register(self, h: bytes, blk: int) -> None:
        """Publish a page the owning sequence just filled.  First
        writer wins; a duplicate page stays private to its sequence."""
        if h in self.by_hash:
            return
        self.by_hash[h] = blk
        self.block_hash[blk] = h
        self.refs[blk] = self.refs.get(blk, 0) + 1

    def release_seq(self, blocks: list[int]) -> None:
        """Drop a retiring sequence's references.  Cached pages park in
        the LRU pool; private pages return to the free list."""
        plain: list[int] = []
        for blk in blocks:
            if blk in self.block_hash:
                n = self.refs.get(blk, 1) - 1
                if n <= 0:
                    self.refs.pop(blk, None)
                    self.lru[blk] = None
                    self.lru.move_to_end(blk)
                else:
                    self.refs[blk] = n
            else:
                plain.append(blk)
        if plain:
            self.kv.release(plain)

    def ensure_free(self, n: int) -> bool:
        """Evict parked pages until ``n`` blocks are allocatable."""
        while self.kv.free_blocks() < n and self.lru:
            blk, _ = self.lru.popitem(last=False)
            h = self.block_hash.pop(blk, None)
            if h is not None:
                self.by_hash.pop(h, None)
            self.refs.pop(blk, None)
            self.kv.release([blk])
        return self.kv.free_blocks() >= n
