"""Draft-MODEL speculative decoding (draft half).

``--speculator-name <model preset>`` loads a second, smaller model whose
greedy continuations are proposed as drafts; the target model verifies them
through the SAME multi-token chunk path the n-gram speculator uses
(engine/spec.py + llm_engine's accept loop), so the greedy-exactness
guarantee is unchanged — rejected drafts roll back and the output equals a
non-speculative run token for token (tested).

Design choices for MI355X serving:
  * The draft keeps its OWN paged KV cache but reuses the target's block
    tables / block ids one-to-one (same slot mapping), so there is no second
    block manager; the cache is sized into the same 288 GB pool at
    init_kv_cache time.
  * Draft state is one integer per request (``draft_computed``): proposals
    first catch the draft up on tokens it has not ingested (bonus/rejection
    suffix — a tiny chunked prefill), then run k-1 single-token decode
    rounds, batched across all eligible requests.
  * Rank 0 only: proposals happen before the step batch is broadcast, so TP
    workers never see the draft model (matches the n-gram speculator).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from .config import ModelConfig
from .metadata import AttnMetadata
from .request import Request


def is_draft_model_spec(name: Optional[str]) -> bool:
    from .config import _PRESETS

    return bool(name) and name.strip("[]").lower() != "ngram" and (
        name.strip("[]") in _PRESETS
    )


class DraftModel:
    def __init__(self, name: str, target_config, device: str, block_size: int):
        from ..models import get_model
        from .weights import synth_llama_weights

        self.config = ModelConfig.from_model_arg(name)
        self.config.dtype = target_config.dtype  # match the target's compute
        # the draft shares the target's context budget for positions
        self.config.max_model_len = min(self.config.max_model_len,
                                        target_config.max_model_len)
        self.device = device
        self.block_size = block_size
        with torch.device(device):
            self.model = get_model(self.config).eval()
        self.model.to(device)
        if self.config.weights_path:
            from .weights import load_safetensors_weights

            self.model.load_weights(
                load_safetensors_weights(self.config.weights_path))
        else:
            self.model.load_weights(synth_llama_weights(self.config, seed=1))
        self.kv_caches: list = []

    def cache_bytes_per_block(self) -> int:
        mc = self.config
        elt = torch.tensor([], dtype=mc.dtype).element_size()
        return 2 * mc.num_layers * self.block_size * mc.num_kv_heads * mc.head_dim * elt

    def alloc_cache(self, num_blocks: int) -> None:
        mc = self.config
        shape = (num_blocks, self.block_size, mc.num_kv_heads, mc.head_dim)
        self.kv_caches = [
            (torch.zeros(shape, dtype=mc.dtype, device=self.device),
             torch.zeros(shape, dtype=mc.dtype, device=self.device))
            for _ in range(mc.num_layers)
        ]

    # ------------------------------------------------------------------
    def _slots(self, req: Request, positions: list[int]) -> list[int]:
        bs = self.block_size
        return [req.block_ids[p // bs] * bs + p % bs for p in positions]

    @torch.inference_mode()
    def propose(self, reqs: list[Request], k: int) -> None:
        """Fill ``req.spec_draft`` (and draft bookkeeping) for each request."""
        dev = self.device
        bs = self.block_size
        work = []
        for req in reqs:
            n = req.num_tokens
            budget = min(k, self.config.max_model_len - n - 1)
            # draft KV writes for proposed tokens must stay inside the
            # blocks the target has already allocated
            budget = min(budget, len(req.block_ids) * bs - n)
            if budget <= 0:
                req.spec_draft = []
                continue
            work.append((req, n, budget))
        if not work:
            return

        # ---- catch-up chunk: tokens the draft has not ingested ------------
        ids, pos, slots, qsl, seq_lens, tables = [], [], [], [0], [], []
        for req, n, _ in work:
            start = min(req.draft_computed, n - 1)
            chunk = req.all_token_ids[start:n]
            p = list(range(start, n))
            ids.extend(chunk)
            pos.extend(p)
            slots.extend(self._slots(req, p))
            qsl.append(qsl[-1] + len(chunk))
            seq_lens.append(n)
            tables.append(req.block_ids)
        logits = self._forward_prefill(ids, pos, slots, qsl, seq_lens, tables)
        preds = torch.argmax(logits, dim=-1).tolist()

        proposals = [[preds[i]] for i in range(len(work))]
        # ---- k-1 single-token decode rounds -------------------------------
        for j in range(1, max(w[2] for w in work)):
            live = [i for i, w in enumerate(work) if w[2] > j]
            if not live:
                break
            ids = [proposals[i][-1] for i in live]
            pos = [work[i][1] + j - 1 for i in live]
            slots = []
            seq_lens = []
            tables = []
            for idx, i in enumerate(live):
                req, n, _ = work[i]
                slots.extend(self._slots(req, [pos[idx]]))
                seq_lens.append(n + j)
                tables.append(req.block_ids)
            logits = self._forward_decode(ids, pos, slots, seq_lens, tables)
            preds = torch.argmax(logits, dim=-1).tolist()
            for idx, i in enumerate(live):
                proposals[i].append(preds[idx])

        for i, (req, n, budget) in enumerate(work):
            req.spec_draft = proposals[i][:budget]
            # the draft ingested the suffix through n-1 plus the proposed
            # tokens it decoded (positions n .. n+len-2)
            req.draft_computed = n + max(len(req.spec_draft) - 1, 0)
            req._draft_base = n

    # ------------------------------------------------------------------
    def _meta(self, *, prefill_qsl=None, prefill_seq_lens=None,
              decode_seq_lens=None, slot_mapping=None, tables=None,
              num_prefill_tokens=0):
        dev = self.device
        int32 = torch.int32
        maxb = max((len(t) for t in tables), default=1)
        bt = np.zeros((len(tables), max(1, maxb)), dtype=np.int32)
        for i, t in enumerate(tables):
            bt[i, :len(t)] = t
        bt_t = torch.from_numpy(bt).to(dev)
        slot_t = torch.tensor(slot_mapping, dtype=torch.long, device=dev)
        empty = torch.empty(0, dtype=int32, device=dev)
        if prefill_qsl is not None:
            return AttnMetadata(
                num_prefill_seqs=len(prefill_seq_lens),
                num_prefill_tokens=num_prefill_tokens,
                num_decode_seqs=0,
                slot_mapping=slot_t,
                prefill_query_start_loc=torch.tensor(prefill_qsl, dtype=int32, device=dev),
                prefill_seq_lens=torch.tensor(prefill_seq_lens, dtype=int32, device=dev),
                prefill_block_tables=bt_t,
                max_prefill_query_len=max(
                    b - a for a, b in zip(prefill_qsl, prefill_qsl[1:])),
                max_prefill_seq_len=max(prefill_seq_lens),
                decode_seq_lens=empty,
                decode_block_tables=torch.empty((0, 0), dtype=int32, device=dev),
                max_decode_seq_len=0,
            )
        return AttnMetadata(
            num_prefill_seqs=0, num_prefill_tokens=0,
            num_decode_seqs=len(decode_seq_lens),
            slot_mapping=slot_t,
            prefill_query_start_loc=empty,
            prefill_seq_lens=empty,
            prefill_block_tables=torch.empty((0, 0), dtype=int32, device=dev),
            max_prefill_query_len=0, max_prefill_seq_len=0,
            decode_seq_lens=torch.tensor(decode_seq_lens, dtype=int32, device=dev),
            decode_block_tables=bt_t,
            max_decode_seq_len=max(decode_seq_lens),
        )

    def _forward_prefill(self, ids, pos, slots, qsl, seq_lens, tables):
        dev = self.device
        meta = self._meta(prefill_qsl=qsl, prefill_seq_lens=seq_lens,
                          slot_mapping=slots, tables=tables,
                          num_prefill_tokens=len(ids))
        ids_t = torch.tensor(ids, dtype=torch.long, device=dev)
        pos_t = torch.tensor(pos, dtype=torch.long, device=dev)
        hidden = self.model(ids_t, pos_t, self.kv_caches, meta)
        rows = torch.tensor([q - 1 for q in qsl[1:]], dtype=torch.long, device=dev)
        return self.model.compute_logits(hidden[rows])

    def _forward_decode(self, ids, pos, slots, seq_lens, tables):
        dev = self.device
        meta = self._meta(decode_seq_lens=seq_lens, slot_mapping=slots,
                          tables=tables)
        ids_t = torch.tensor(ids, dtype=torch.long, device=dev)
        pos_t = torch.tensor(pos, dtype=torch.long, device=dev)
        hidden = self.model(ids_t, pos_t, self.kv_caches, meta)
        return self.model.compute_logits(hidden)
