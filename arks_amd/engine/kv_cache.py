"""Paged KV-cache block allocator.

Blocks are BLOCK_SIZE(=16)-token pages of the [num_blocks, num_kv_heads, 16,
head_dim] cache tensors (one pair per layer). The allocator is pure
bookkeeping — tensors live in the model runner. Free list is LIFO for L2/L3
locality of recently-freed pages.
"""

from __future__ import annotations


class BlockAllocator:
    def __init__(self, num_blocks: int, block_size: int = 16):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: list[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def can_allocate(self, n: int) -> bool:
        return len(self._free) >= n

    def allocate(self, n: int) -> list[int]:
        if n > len(self._free):
            raise RuntimeError(f"KV cache exhausted: want {n}, free {len(self._free)}")
        out = [self._free.pop() for _ in range(n)]
        return out

    def free(self, blocks: list[int]) -> None:
        self._free.extend(reversed(blocks))

    @staticmethod
    def blocks_needed(num_tokens: int, block_size: int = 16) -> int:
        return (num_tokens + block_size - 1) // block_size


def kv_cache_block_bytes(num_layers: int, num_kv_heads: int, head_dim: int,
                         block_size: int = 16, dtype_bytes: int = 2) -> int:
    """Bytes per block across all layers (K and V)."""
    return 2 * num_layers * num_kv_heads * block_size * head_dim * dtype_bytes
