"""Per-step attention metadata shared by the model runner and the attention ops.

Token order within a step batch: all prefill-chunk tokens first (grouped per
sequence), then one token per decoding sequence.  This lets attention run the
MFMA prefill kernel over the first ``num_prefill_tokens`` rows and the
decode kernel over the rest without re-gathering.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class AttnMetadata:
    num_prefill_seqs: int
    num_prefill_tokens: int
    num_decode_seqs: int

    slot_mapping: torch.Tensor  # [T] int64 device — flat KV slot per token

    # Prefill part (empty tensors when no prefill)
    prefill_query_start_loc: torch.Tensor  # [np+1] int32 device
    prefill_seq_lens: torch.Tensor         # [np] int32 device (context incl. chunk)
    prefill_block_tables: torch.Tensor     # [np, max_blocks] int32 device
    max_prefill_query_len: int
    max_prefill_seq_len: int

    # Decode part
    decode_seq_lens: torch.Tensor      # [nd] int32 device
    decode_block_tables: torch.Tensor  # [nd, max_blocks] int32 device
    max_decode_seq_len: int

    @property
    def num_tokens(self) -> int:
        return self.num_prefill_tokens + self.num_decode_seqs

    @property
    def is_decode_only(self) -> bool:
        return self.num_prefill_tokens == 0
