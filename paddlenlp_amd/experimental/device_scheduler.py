"""Device-resident decode loop: fused sampling + in-kernel block scheduler.

Reference behavior: the reference's block-decode runtime keeps all loop
state on device and advances it with fused ops per step —
`get_token_penalty_multi_scores_v2` -> `top_p_sampling_reject` ->
`set_stop_value_multi_ends_v2` -> `update_inputs` -> `step_paddle`
(csrc/gpu/step.cu:316) — with only `not_need_stop` visible to
the host loop condition.  This module is the MI355X equivalent: the decode
loop makes zero host round-trips per step; block allocation, preemption,
eos handling and sampling all run in HIP kernels
(paddlenlp_amd/ops/csrc/sampling.hip).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..ops.functional import _load_extension


class DeviceDecodeLoop:
    """Owns the device-resident decode state for a FusedMultiTransformer
    engine: block table, sequence lengths, free list, stop flags, token
    history and per-sequence sampling parameters."""

    def __init__(self, engine, max_batch: int, num_blocks: int,
                 max_blocks_per_seq: int, device,
                 eos_ids: List[int], pad_id: int = 0,
                 max_gen_len: int = 2048):
        self.C = _load_extension()
        self.engine = engine
        self.block_size = engine.config.block_size
        self.device = device
        self.max_batch = max_batch
        self.max_gen_len = max_gen_len
        self.pad_id = pad_id

        dev = device
        self.block_table = torch.full((max_batch, max_blocks_per_seq), -1,
                                      dtype=torch.int32, device=dev)
        self.seq_lens = torch.zeros(max_batch, dtype=torch.int32, device=dev)
        self.stop_flags = torch.ones(max_batch, dtype=torch.int8, device=dev)
        self.active = torch.zeros(max_batch, dtype=torch.int8, device=dev)
        self.is_block_step = torch.zeros(max_batch, dtype=torch.int8, device=dev)
        self.free_list = torch.arange(num_blocks - 1, -1, -1,
                                      dtype=torch.int32, device=dev)
        self.free_count = torch.tensor([num_blocks], dtype=torch.int32, device=dev)
        self.pre_ids = torch.zeros(max_batch, max_gen_len, dtype=torch.int64, device=dev)
        self.pre_lens = torch.zeros(max_batch, dtype=torch.int32, device=dev)
        self.not_need_stop = torch.zeros(1, dtype=torch.int32, device=dev)
        self.eos_ids = torch.tensor(eos_ids or [-1], dtype=torch.int64, device=dev)
        # per-sequence sampling params
        self.temperature = torch.ones(max_batch, dtype=torch.float32, device=dev)
        self.top_p = torch.zeros(max_batch, dtype=torch.float32, device=dev)  # 0 = greedy
        self.rep_pen = torch.ones(max_batch, dtype=torch.float32, device=dev)
        self.max_new = torch.full((max_batch,), max_gen_len, dtype=torch.int32, device=dev)
        self.tokens = torch.zeros(max_batch, 1, dtype=torch.int64, device=dev)

    # ------------------------------------------------------------------
    # admission (host side; rare relative to decode steps)
    # ------------------------------------------------------------------
    def _host_alloc_blocks(self, n: int) -> Optional[torch.Tensor]:
        have = int(self.free_count.item())
        if n > have:
            return None
        blocks = self.free_list[have - n:have].clone()
        self.free_count.fill_(have - n)
        return blocks

    def add_request(self, slot: int, prompt_len: int, first_token: int,
                    temperature: float = 1.0, top_p: float = 0.0,
                    rep_pen: float = 1.0, max_new: int = None,
                    blocks: torch.Tensor = None) -> bool:
        """Activate a slot whose prompt KV was just prefilled.  `blocks`
        must be the block ids used during prefill (from allocate_for_prefill)."""
        if blocks is None:
            return False
        self.block_table[slot].fill_(-1)
        self.block_table[slot, :blocks.numel()] = blocks
        self.seq_lens[slot] = prompt_len
        self.stop_flags[slot] = 0
        self.active[slot] = 1
        self.is_block_step[slot] = 0
        self.pre_lens[slot] = 0
        self.temperature[slot] = temperature
        self.top_p[slot] = top_p
        self.rep_pen[slot] = rep_pen
        self.max_new[slot] = max_new or self.max_gen_len
        self.tokens[slot, 0] = first_token
        self.not_need_stop.fill_(1)
        return True

    def allocate_for_prefill(self, prompt_len: int) -> Optional[torch.Tensor]:
        # +1: the first decode step appends a token before block_step runs
        need = (prompt_len + 1 + self.block_size - 1) // self.block_size
        return self._host_alloc_blocks(need)

    # ------------------------------------------------------------------
    # the zero-host-sync loop
    # ------------------------------------------------------------------
    def decode_steps(self, n_steps: int) -> torch.Tensor:
        """Run up to n_steps decode iterations entirely on device.
        Returns the token history tensor (pre_ids)."""
        C = self.C
        for _ in range(n_steps):
            logits = self.engine.decode_step(self.tokens, self.block_table,
                                             self.seq_lens)
            if logits.dim() == 3:
                logits = logits[:, -1]
            logits = logits.contiguous().to(torch.bfloat16)
            C.apply_repetition_penalty(logits, self.pre_ids, self.pre_lens,
                                       self.rep_pen)
            u = torch.rand(self.max_batch, dtype=torch.float32, device=self.device)
            nxt = C.topp_sample(logits, self.temperature, self.top_p, u)
            C.decode_update(nxt, self.stop_flags, self.active, self.pre_ids,
                            self.pre_lens, self.seq_lens, self.eos_ids,
                            self.not_need_stop, self.max_new, self.pad_id)
            C.block_step(self.block_table, self.seq_lens, self.stop_flags,
                         self.active, self.free_list, self.free_count,
                         self.is_block_step, self.block_size)
            self.tokens.copy_(nxt.unsqueeze(1))
        return self.pre_ids

    def running(self) -> bool:
        return bool(self.not_need_stop.item())

    def harvest(self, slot: int) -> List[int]:
        n = int(self.pre_lens[slot].item())
        return self.pre_ids[slot, :n].tolist()
