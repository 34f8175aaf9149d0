"""Paged-KV block manager: free list, allocation, preemption.

Reference behavior: csrc/gpu/step.cu (step_paddle) — free blocks
of stopped sequences, allocate to growing sequences, preempt the longest
running sequence when the free list empties, recover preempted sequences.
Host-side here (the per-step bookkeeping is tiny next to a decode step);
a device-side scheduler is a later optimization.
"""
from __future__ import annotations

from typing import List, Optional, Set

import torch


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int, max_blocks_per_seq: int,
                 max_batch: int, device="cpu"):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.max_blocks_per_seq = max_blocks_per_seq
        self.max_batch = max_batch
        self.free_list: List[int] = list(range(num_blocks - 1, -1, -1))
        # -1 = unallocated
        self.block_table = torch.full((max_batch, max_blocks_per_seq), -1, dtype=torch.int32)
        self.seq_lens = torch.zeros(max_batch, dtype=torch.int32)
        self.active: Set[int] = set()

    def free_blocks_available(self) -> int:
        return len(self.free_list)

    def blocks_needed(self, n_tokens: int) -> int:
        return (n_tokens + self.block_size - 1) // self.block_size

    def allocate_slot(self, prompt_len: int) -> Optional[int]:
        """Find a free batch slot and allocate blocks for the prompt.
        Returns the slot index or None if no capacity."""
        need = self.blocks_needed(prompt_len)
        if need > len(self.free_list):
            return None
        slot = None
        for i in range(self.max_batch):
            if i not in self.active:
                slot = i
                break
        if slot is None:
            return None
        for j in range(need):
            self.block_table[slot, j] = self.free_list.pop()
        self.seq_lens[slot] = prompt_len
        self.active.add(slot)
        return slot

    def extend(self, slot: int, n_new: int = 1) -> bool:
        """Grow a sequence by n_new tokens, allocating blocks as needed.
        Returns False when out of blocks (caller must preempt or wait)."""
        cur = int(self.seq_lens[slot])
        have = self.blocks_needed(cur) if cur else 0
        need = self.blocks_needed(cur + n_new)
        for j in range(have, need):
            if not self.free_list:
                return False
            self.block_table[slot, j] = self.free_list.pop()
        self.seq_lens[slot] = cur + n_new
        return True

    def release(self, slot: int):
        for j in range(self.max_blocks_per_seq):
            blk = int(self.block_table[slot, j])
            if blk >= 0:
                self.free_list.append(blk)
                self.block_table[slot, j] = -1
        self.seq_lens[slot] = 0
        self.active.discard(slot)

    def preempt_longest(self) -> Optional[int]:
        """Free the longest active sequence (reference step.cu picks argmax
        used length).  Returns the preempted slot."""
        if not self.active:
            return None
        slot = max(self.active, key=lambda s: int(self.seq_lens[s]))
        self.release(slot)
        return slot
