"""hipGraph capture of the decode step.

Steady-state decode batches are launch-bound (dozens of small kernels per
layer); capturing the whole forward + logits computation per batch-size
bucket and replaying removes the per-kernel launch gaps (microarch
'launches-baseline': ≈1.2 µs per boundary).

Capture rules honored: static input/output buffers, no allocation inside the
captured region (the mempool is shared across buckets), padding rows carry
slot_mapping = -1 (the cache-write kernel skips them) and seq_len = 0 (the
decode kernel emits zeros).  Batches with LoRA tokens or prompt-logprob rows
fall back to eager.
"""

from __future__ import annotations

from typing import Optional

import torch

_BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256, 384, 512, 768, 1024]


class DecodeGraphRunner:
    def __init__(self, model, kv_caches, model_config, block_size: int,
                 max_num_seqs: int, max_model_len: int, device: str):
        self.model = model
        self.kv_caches = kv_caches
        self.device = device
        self.block_size = block_size
        self.max_blocks = (max_model_len + block_size - 1) // block_size
        self.buckets = [b for b in _BUCKETS if b <= max(max_num_seqs, 1)]
        if not self.buckets:
            self.buckets = [1]
        bmax = self.buckets[-1]

        dev = device
        self.in_ids = torch.zeros(bmax, dtype=torch.long, device=dev)
        self.in_pos = torch.zeros(bmax, dtype=torch.long, device=dev)
        self.in_slots = torch.full((bmax,), -1, dtype=torch.long, device=dev)
        self.in_seq_lens = torch.zeros(bmax, dtype=torch.int32, device=dev)
        self.in_block_tables = torch.zeros(
            (bmax, self.max_blocks), dtype=torch.int32, device=dev
        )
        self.out_logits: dict[int, torch.Tensor] = {}
        self.graphs: dict[int, torch.cuda.CUDAGraph] = {}
        self._pool = None
        self._empty_i32 = torch.empty(0, dtype=torch.int32, device=dev)
        self._empty_bt = torch.empty((0, 0), dtype=torch.int32, device=dev)

    def _meta(self, b: int):
        from .metadata import AttnMetadata

        return AttnMetadata(
            num_prefill_seqs=0, num_prefill_tokens=0, num_decode_seqs=b,
            slot_mapping=self.in_slots[:b],
            prefill_query_start_loc=self._empty_i32,
            prefill_seq_lens=self._empty_i32,
            prefill_block_tables=self._empty_bt,
            max_prefill_query_len=0, max_prefill_seq_len=0,
            decode_seq_lens=self.in_seq_lens[:b],
            decode_block_tables=self.in_block_tables[:b],
            max_decode_seq_len=0,
        )

    @torch.inference_mode()
    def capture(self) -> None:
        stream = torch.cuda.Stream()
        with torch.cuda.stream(stream):
            # one eager warmup pass (cuBLAS/hipBLASLt workspace init etc.)
            b = self.buckets[-1]
            hidden = self.model(self.in_ids[:b], self.in_pos[:b],
                                self.kv_caches, self._meta(b))
            self.model.compute_logits(hidden)
        torch.cuda.current_stream().wait_stream(stream)
        torch.cuda.synchronize()

        for b in reversed(self.buckets):
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=self._pool):
                hidden = self.model(self.in_ids[:b], self.in_pos[:b],
                                    self.kv_caches, self._meta(b))
                logits = self.model.compute_logits(hidden)
            if self._pool is None:
                self._pool = g.pool()
            self.graphs[b] = g
            self.out_logits[b] = logits
        torch.cuda.synchronize()

    def bucket_for(self, n: int) -> Optional[int]:
        for b in self.buckets:
            if b >= n:
                return b
        return None

    @torch.inference_mode()
    def run(self, ids, pos, slots, seq_lens, block_tables) -> torch.Tensor:
        """All tensors device-resident; returns logits rows [n, vocab]."""
        n = ids.shape[0]
        b = self.bucket_for(n)
        assert b is not None
        self.in_ids[:n].copy_(ids)
        self.in_pos[:n].copy_(pos)
        self.in_slots[:n].copy_(slots)
        self.in_slots[n:b].fill_(-1)
        self.in_seq_lens[:n].copy_(seq_lens)
        self.in_seq_lens[n:b].zero_()
        w = block_tables.shape[1]
        self.in_block_tables[:n, :w].copy_(block_tables)
        if w < self.max_blocks:
            self.in_block_tables[:n, w:].zero_()
        self.graphs[b].replay()
        return self.out_logits[b][:n]
