"""Paged KV cache pool + block allocator, sized for 288 GB HBM3E.

Replaces the paged-KV machinery the reference inherits from vLLM
(paged_attention, reshape_and_cache, block tables).
Blocks are reference-counted so the n-candidate fan-out of one prompt
shares the prompt's full KV blocks (prefill once per prompt instead of
n times — the reference pays vLLM's prefix sharing here).
"""

from __future__ import annotations

from typing import List, Optional

import torch


class BlockAllocator:
    """Free-list allocator with reference counts (for shared prompt blocks)."""

    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        self._ref = [0] * num_blocks

    @property
    def num_free(self) -> int:
        return len(self._free)

    def alloc(self) -> int:
        if not self._free:
            raise MemoryError("KV cache pool exhausted")
        b = self._free.pop()
        assert self._ref[b] == 0
        self._ref[b] = 1
        return b

    def incref(self, block: int) -> None:
        assert self._ref[block] > 0
        self._ref[block] += 1

    def free(self, block: int) -> None:
        self._ref[block] -= 1
        assert self._ref[block] >= 0
        if self._ref[block] == 0:
            self._free.append(block)

    def refcount(self, block: int) -> int:
        return self._ref[block]

    def reset(self) -> None:
        """Return every block to the free list, dropping all refcounts.
        Only valid when no live sequence remains (error recovery after a
        failed generation — generate calls are serialized, so a failure
        strands every in-flight sequence of that call and nothing else)."""
        self._free = list(range(self.num_blocks - 1, -1, -1))
        self._ref = [0] * self.num_blocks


class KVCachePool:
    """Per-layer paged K/V tensors.

    Layout: (num_layers, num_blocks, block_size, n_kv_heads, head_dim) for
    K and V separately — layer slice [l] is a contiguous 4-D view passed to
    the HIP kernels.
    """

    def __init__(self, num_layers: int, num_blocks: int, block_size: int,
                 n_kv_heads: int, head_dim: int, dtype: torch.dtype, device):
        self.num_layers = num_layers
        self.num_blocks = num_blocks
        self.block_size = block_size
        shape = (num_layers, num_blocks, block_size, n_kv_heads, head_dim)
        self.key = torch.zeros(shape, dtype=dtype, device=device)
        self.value = torch.zeros(shape, dtype=dtype, device=device)
        self.allocator = BlockAllocator(num_blocks)

    @staticmethod
    def blocks_for(num_tokens: int, block_size: int) -> int:
        return (num_tokens + block_size - 1) // block_size

    def copy_block(self, src: int, dst: int) -> None:
        """Copy the full contents of one block across all layers (used when
        forking a sequence whose last prompt block is partial)."""
        self.key[:, dst].copy_(self.key[:, src])
        self.value[:, dst].copy_(self.value[:, src])

    @staticmethod
    def pool_size_bytes(num_blocks: int, num_layers: int, block_size: int,
                        n_kv_heads: int, head_dim: int, dtype: torch.dtype) -> int:
        return (2 * num_layers * num_blocks * block_size * n_kv_heads * head_dim
                * torch.tensor([], dtype=dtype).element_size())


class Sequence:
    """One decoding stream (one candidate of one prompt)."""

    __slots__ = ("seq_id", "prompt_ids", "output_ids", "block_table",
                 "context_len", "finished", "parent_prompt",
                 "last_logits_idx", "cand_index", "max_tokens")

    def __init__(self, seq_id: int, prompt_ids: List[int], parent_prompt: int):
        self.seq_id = seq_id
        self.prompt_ids = prompt_ids
        self.output_ids: List[int] = []
        self.block_table: List[int] = []
        self.context_len = 0  # number of KV tokens materialized
        self.finished = False
        self.parent_prompt = parent_prompt
        self.last_logits_idx: Optional[int] = None
        self.cand_index = 0  # which of the n candidates of its prompt
        self.max_tokens: Optional[int] = None  # per-seq cap (None -> sp)

    @property
    def total_len(self) -> int:
        return len(self.prompt_ids) + len(self.output_ids)
