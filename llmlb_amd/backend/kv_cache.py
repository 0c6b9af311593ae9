"""Paged KV cache for the MI355X backend.

Layout per layer: [num_blocks, n_kv_local, block_size, head_dim] bf16 —
the layout the gfx950 paged-attention kernel consumes (one 4 KiB page
per (block, kv_head) at head_dim=128, read 16 B/lane).

Sized for 288 GB HBM3E per GPU: the engine asks
``blocks_for_memory_fraction`` how many pages fit the configured HBM
budget after weights.
"""
from __future__ import annotations

import torch

BLOCK_SIZE = 16  # tokens per page (matches PA_BLOCK in kernels.hip)


class KVCache:
    def __init__(self, layers: int, num_blocks: int, n_kv: int,
                 head_dim: int, device, dtype=torch.bfloat16):
        self.block_size = BLOCK_SIZE
        self.num_blocks = num_blocks
        self.k = [torch.zeros(num_blocks, n_kv, BLOCK_SIZE, head_dim,
                              dtype=dtype, device=device)
                  for _ in range(layers)]
        self.v = [torch.zeros(num_blocks, n_kv, BLOCK_SIZE, head_dim,
                              dtype=dtype, device=device)
                  for _ in range(layers)]
        # block 0 is reserved as the sacrificial target of padded rows
        # in hipGraph-captured decode steps (engine._DecodeGraph)
        self._free: list[int] = list(range(num_blocks - 1, 0, -1))

    @staticmethod
    def bytes_per_block(layers: int, n_kv: int, head_dim: int,
                        dtype=torch.bfloat16) -> int:
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
