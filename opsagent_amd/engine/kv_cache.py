"""Paged KV cache manager with block-level prefix caching.

SURVEY.md §2b: "hipGraph-captured decode step, re-used across multi-turn
ReAct iterations — the loop at simple.go:394-615 re-sends full history each
turn; persistent KV across tool-call turns = the agent-turn-latency lever."

Mechanism: the cache is a pool of fixed-size token blocks
([num_blocks, block_size, Hk_local, D] bf16 per layer, sized against the
288 GB of HBM3E per MI355X). Each FULL block of a sequence is content-hashed
by its chain hash (hash of all token ids up to and including the block);
completed blocks are published to a hash table. A new request reuses the
longest prefix of published blocks — so when the ReAct loop re-sends the
growing conversation, only the new suffix is prefilled. Freed blocks stay
cached (refcount 0) in LRU order and are evicted on allocation pressure.
"""

from __future__ import annotations

import collections
from typing import Dict, List, Optional, Sequence, Tuple

import torch


class BlockAllocatorError(RuntimeError):
    pass


class PagedKVCache:
    def __init__(
        self,
        num_layers: int,
        num_kv_heads_local: int,
        head_dim: int,
        block_size: int,
        num_blocks: int,
        device: str,
        dtype: torch.dtype = torch.bfloat16,
    ):
        assert block_size & (block_size - 1) == 0, "block_size must be a power of 2"
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.device = device
        self.layers: List[Tuple[torch.Tensor, torch.Tensor]] = []
        for _ in range(num_layers):
            k = torch.zeros(
                num_blocks, block_size, num_kv_heads_local, head_dim, dtype=dtype, device=device
            )
            v = torch.zeros_like(k)
            self.layers.append((k, v))
        # allocator state
        self._free: collections.deque[int] = collections.deque(range(num_blocks))
        self._ref: List[int] = [0] * num_blocks
        # prefix cache: chain_hash -> block_id; and reverse for eviction.
        # _block_tokens keeps each published block's OWN token slice so a
        # 64-bit chain-hash collision cannot silently alias wrong KV: the
        # chain hash binds the ancestry, the stored slice verifies the
        # block's content (O(block_size) per block, not O(prefix)).
        self._hash_to_block: Dict[int, int] = {}
        self._block_to_hash: List[Optional[int]] = [None] * num_blocks
        self._block_tokens: List[Optional[Tuple[int, ...]]] = [None] * num_blocks
        # LRU of refcount-0 cached blocks (evictable)
        self._evictable: "collections.OrderedDict[int, None]" = collections.OrderedDict()
        self.stats = {"reused_blocks": 0, "allocated_blocks": 0, "evictions": 0}

    # -- low-level allocator ----------------------------------------------
    def num_free(self) -> int:
        return len(self._free) + len(self._evictable)

    def _pop_free_block(self) -> int:
        if self._free:
            return self._free.popleft()
        if self._evictable:
            bid, _ = self._evictable.popitem(last=False)  # LRU
            h = self._block_to_hash[bid]
            if h is not None:
                del self._hash_to_block[h]
                self._block_to_hash[bid] = None
                self._block_tokens[bid] = None
            self.stats["evictions"] += 1
            return bid
        raise BlockAllocatorError("KV cache out of blocks")

    def alloc_block(self) -> int:
        bid = self._pop_free_block()
        self._ref[bid] = 1
        self.stats["allocated_blocks"] += 1
        return bid

    def retain(self, bid: int) -> None:
        if self._ref[bid] == 0:
            self._evictable.pop(bid, None)
        self._ref[bid] += 1

    def release(self, bid: int) -> None:
        assert self._ref[bid] > 0
        self._ref[bid] -= 1
        if self._ref[bid] == 0:
            if self._block_to_hash[bid] is not None:
                self._evictable[bid] = None  # keep content for prefix reuse
            else:
                self._free.append(bid)

    # -- prefix cache -------------------------------------------------------
    @staticmethod
    def chain_hash(prev: int, ids: Sequence[int]) -> int:
        return hash((prev, tuple(ids)))

    def lookup_prefix(self, token_ids: Sequence[int]) -> Tuple[List[int], int]:
        """Longest reusable prefix. Returns (block_ids retained, n_tokens).

        Never reuses the entire prompt: at least one token must be recomputed
        so the engine has a last-token hidden state to sample from.
        """
        bs = self.block_size
        blocks: List[int] = []
        h = 0
        max_full = (len(token_ids) - 1) // bs  # leave ≥1 token to prefill
        for i in range(max_full):
            h = self.chain_hash(h, token_ids[i * bs : (i + 1) * bs])
            bid = self._hash_to_block.get(h)
            if bid is None:
                break
            # collision guard: the hash indexes, token equality decides
            if self._block_tokens[bid] != tuple(token_ids[i * bs : (i + 1) * bs]):
                break
            blocks.append(bid)
        for bid in blocks:
            self.retain(bid)
        self.stats["reused_blocks"] += len(blocks)
        return blocks, len(blocks) * bs

    def publish_block(
        self, bid: int, chain_h: int, prefix_tokens: Tuple[int, ...]
    ) -> int:
        """Publish a full block under its chain hash. If an identical block
        (hash AND token-verified) is already cached, switch to it (dedup) and
        release ours. Returns the canonical block id."""
        existing = self._hash_to_block.get(chain_h)
        if existing is not None and existing != bid:
            if self._block_tokens[existing] == prefix_tokens:
                self.retain(existing)
                self.release(bid)
                return existing
            # hash collision with DIFFERENT content: evict the stale entry
            self._block_to_hash[existing] = None
            self._block_tokens[existing] = None
        self._hash_to_block[chain_h] = bid
        self._block_to_hash[bid] = chain_h
        self._block_tokens[bid] = prefix_tokens
        return bid


class SequenceState:
    """Per-sequence cache bookkeeping."""

    def __init__(self, cache: PagedKVCache, token_ids: List[int]):
        self.cache = cache
        self.token_ids = list(token_ids)  # prompt + generated
        self.blocks: List[int] = []
        self.num_cached = 0     # tokens whose KV is in cache
        self.chain_h = 0        # chain hash over published full blocks
        self.published = 0      # number of full blocks published

    def reuse_prefix(self) -> int:
        blocks, n = self.cache.lookup_prefix(self.token_ids)
        self.blocks = blocks
        self.num_cached = n
        self.published = len(blocks)
        h = 0
        bs = self.cache.block_size
        for i in range(len(blocks)):
            h = self.cache.chain_hash(h, self.token_ids[i * bs : (i + 1) * bs])
        self.chain_h = h
        return n

    def ensure_capacity(self, n_tokens: int) -> None:
        bs = self.cache.block_size
        need = (n_tokens + bs - 1) // bs
        while len(self.blocks) < need:
            self.blocks.append(self.cache.alloc_block())

    def slots_for(self, start: int, count: int) -> torch.Tensor:
        """Flat slot ids for token positions [start, start+count)."""
        bs = self.cache.block_size
        out = torch.empty(count, dtype=torch.int32)
        for i in range(count):
            pos = start + i
            out[i] = self.blocks[pos // bs] * bs + pos % bs
        return out

    def all_slots(self, upto: int) -> torch.Tensor:
        return self.slots_for(0, upto)

    def publish_full_blocks(self) -> None:
        """Publish newly-completed full blocks to the prefix cache."""
        bs = self.cache.block_size
        full = self.num_cached // bs
        while self.published < full:
            i = self.published
            self.chain_h = self.cache.chain_hash(
                self.chain_h, self.token_ids[i * bs : (i + 1) * bs]
            )
            self.blocks[i] = self.cache.publish_block(
                self.blocks[i], self.chain_h,
                tuple(self.token_ids[i * bs : (i + 1) * bs]),
            )
            self.published += 1

    def free(self) -> None:
        for bid in self.blocks:
            self.cache.release(bid)
        self.blocks = []
