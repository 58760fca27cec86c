"""GPU worker: owns the model, the paged KV cache, and step execution.

One worker per GPU (one process per GPU for TP>1).  Rank 0
drives: it turns a SchedulerOutput into a flat token batch, broadcasts it to
the other TP ranks (torch.distributed over RCCL/gloo), runs the forward, and
samples.  Non-zero ranks sit in ``worker_loop`` executing broadcast commands.
"""

from __future__ import annotations

import gc
from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from .. import ops
from ..models import get_model
from ..parallel import get_tp_rank, get_tp_world_size, tp_broadcast_object
from . import roctx
from .config import EngineConfig
from .metadata import AttnMetadata
from .sampler import Sampler, SamplerOutput, prompt_logprob_dicts
from .scheduler import ScheduledItem, SchedulerOutput


@dataclass
class ExecuteResult:
    # one entry per sampling item (scheduler items with samples=True, in order)
    sampler_output: SamplerOutput
    # speculative items: (item, model predictions for draft rows + bonus)
    spec_results: list = None
    # item-aligned prompt logprob additions handled directly on Request objects


class _DoneStep:
    """Already-synchronous step result (slow sampling paths)."""

    __slots__ = ("_result",)

    def __init__(self, result: ExecuteResult):
        self._result = result

    def finish(self) -> ExecuteResult:
        return self._result


class _PendingStep:
    """Fused-sampler step in flight; finish() syncs the sampled token ids
    (and any deferred logprob tensors the sampler launched alongside)."""

    __slots__ = ("_out", "_sampler")

    def __init__(self, out: torch.Tensor, sampler: Sampler):
        self._out = out
        self._sampler = sampler

    def finish(self) -> ExecuteResult:
        return ExecuteResult(
            sampler_output=self._sampler.finish_fused(self._out.tolist()),
            spec_results=[],
        )


def _pad_np_tables(tables) -> np.ndarray:
    """Pad a (possibly ragged) block-table list to a dense int32 array."""
    if isinstance(tables, np.ndarray):
        if tables.size == 0:
            return np.zeros((0, 0), dtype=np.int32)
        return np.ascontiguousarray(tables, dtype=np.int32)
    if not tables:
        return np.zeros((0, 0), dtype=np.int32)
    maxb = max(1, max(len(t) for t in tables))
    arr = np.zeros((len(tables), maxb), dtype=np.int32)
    for i, t in enumerate(tables):
        if t:
            arr[i, : len(t)] = t
    return arr


def _pad_block_tables(tables, device) -> torch.Tensor:
    if isinstance(tables, np.ndarray):  # already padded (TP broadcast path)
        if tables.size == 0:
            return torch.empty((0, 0), dtype=torch.int32, device=device)
        return torch.from_numpy(np.ascontiguousarray(tables, dtype=np.int32)).to(
            device, non_blocking=True)
    if not tables:
        return torch.empty((0, 0), dtype=torch.int32, device=device)
    maxb = max(1, max(len(t) for t in tables))
    arr = np.zeros((len(tables), maxb), dtype=np.int32)
    for i, t in enumerate(tables):
        if t:
            arr[i, : len(t)] = t
    return torch.from_numpy(arr).to(device, non_blocking=True)


def _tables_to_array(tables: list[list[int]]) -> np.ndarray:
    if not tables:
        return np.zeros((0, 0), dtype=np.int64)
    maxb = max(1, max(len(t) for t in tables))
    arr = np.zeros((len(tables), maxb), dtype=np.int64)
    for i, t in enumerate(tables):
        if t:
            arr[i, : len(t)] = t
    return arr


def pack_batch(batch: dict) -> tuple[np.ndarray, np.ndarray]:
    """Flatten a step batch into (header, payload) int64 arrays for a pair of
    tensor broadcasts — ~10x cheaper than pickling the dict at batch 256."""
    pt = _tables_to_array(batch["prefill_tables"])
    dt = _tables_to_array(batch["decode_tables"])
    lora = batch["lora_ids"] if batch["lora_ids"] is not None else []
    parts = [
        np.asarray(batch["token_ids"], dtype=np.int64),
        np.asarray(batch["positions"], dtype=np.int64),
        np.asarray(batch["slot_mapping"], dtype=np.int64),
        np.asarray(batch["qsl"], dtype=np.int64),
        np.asarray(batch["prefill_seq_lens"], dtype=np.int64),
        pt.reshape(-1),
        np.asarray(batch["decode_seq_lens"], dtype=np.int64),
        dt.reshape(-1),
        np.asarray(batch["logit_rows"], dtype=np.int64),
        np.asarray(lora, dtype=np.int64),
    ]
    header = np.array(
        [batch["num_sample_rows"], len(batch["qsl"]),
         len(batch["prefill_seq_lens"]), pt.shape[1] if pt.size else 0,
         len(batch["decode_seq_lens"]), dt.shape[1] if dt.size else 0,
         1 if batch["lora_ids"] is not None else 0]
        + [len(p) for p in parts],
        dtype=np.int64,
    )
    return header, np.concatenate(parts) if parts else np.zeros(0, np.int64)


def unpack_batch(header: np.ndarray, payload: np.ndarray) -> dict:
    num_sample_rows, n_qsl, n_pf, pf_maxb, n_dec, dec_maxb, has_lora = header[:7]
    sizes = header[7:]
    arrs = []
    off = 0
    for n in sizes:
        arrs.append(payload[off:off + n])
        off += n
    (token_ids, positions, slot_mapping, qsl, pf_lens, pf_flat, dec_lens,
     dec_flat, logit_rows, lora) = arrs
    return dict(
        token_ids=token_ids,
        positions=positions,
        slot_mapping=slot_mapping,
        qsl=qsl.tolist(),
        prefill_seq_lens=pf_lens,
        prefill_tables=pf_flat.reshape(int(n_pf), int(pf_maxb)) if pf_maxb else
        np.zeros((0, 0), np.int64),
        decode_seq_lens=dec_lens,
        decode_tables=dec_flat.reshape(int(n_dec), int(dec_maxb)) if dec_maxb else
        np.zeros((0, 0), np.int64),
        logit_rows=logit_rows.tolist(),
        num_sample_rows=int(num_sample_rows),
        lora_ids=lora.tolist() if has_lora else None,
    )


class Worker:
    def __init__(self, config: EngineConfig):
        self.config = config
        self.model_config = config.model_config
        self.device = config.resolve_device()
        self.rank = get_tp_rank()
        self.tp = get_tp_world_size()
        if self.device == "cuda":
            torch.cuda.set_device(self.rank % max(1, torch.cuda.device_count()))
        torch.manual_seed(config.seed)

        with torch.device(self.device):
            self.model = get_model(self.model_config).eval()
        self.model.to(self.device)
        if self.model_config.weights_path:
            self._load_weights()
        elif config.load_synthetic_weights:
            from .weights import synth_llama_weights

            self.model.load_weights(synth_llama_weights(self.model_config, config.seed))

        if config.quantization:
            self._quantize_model(config.quantization)

        self.block_size = config.cache_config.block_size
        self.kv_caches: list[tuple[torch.Tensor, torch.Tensor]] = []
        self.num_blocks = 0
        self.sampler = Sampler(self.device)
        self.graph_runner = None  # set by capture_decode_graphs()
        self.loras: dict[int, object] = {}  # lora_int_id -> LoRAAdapter

        # draft-MODEL speculation (E17): rank 0 holds the whole draft model
        # (proposals happen before the step batch is broadcast)
        self.draft = None
        from .draft import DraftModel, is_draft_model_spec

        if is_draft_model_spec(config.speculative_model) and self.rank == 0:
            self.draft = DraftModel(
                config.speculative_model.strip("[]"), self.model_config,
                self.device, self.block_size,
            )

        # pinned H2D staging, double-buffered by step parity: .to(cuda,
        # non_blocking=True) from pageable numpy is silently SYNCHRONOUS
        # (it serializes with the stream, stalling the pipelined step's
        # launch phase behind the previous step's GPU work)
        self._pin: dict[str, torch.Tensor] = {}
        self._pin_flip = 0

        if self.tp > 1 and self.device == "cuda":
            from ..parallel import init_xgmi_allreduce

            init_xgmi_allreduce()  # E15: direct-xGMI AR, RCCL fallback

    # ------------------------------------------------------------------
    def _quantize_model(self, method: str) -> None:
        """Weight-only RTN quantization of the attention/MLP linears (E18).

        --quantize int8 -> per-channel int8; int4 / awq / gptq / squeezellm
        -> packed int4 with group-128 scales (the 4-bit storage profile of
        those methods; checkpoint-specific scales are not available offline,
        so RTN is applied to the loaded weights).  Embedding / lm_head stay
        bf16 (their accuracy sensitivity; standard practice).
        """
        from ..parallel.layers import (
            ColumnParallelLinear,
            MergedColumnParallelLinear,
            RowParallelLinear,
            quantize_module_,
        )

        qbits = 8 if method == "int8" else 4
        kinds = (ColumnParallelLinear, MergedColumnParallelLinear,
                 RowParallelLinear)
        count = 0
        for mod in self.model.modules():
            if isinstance(mod, kinds) and hasattr(mod, "weight"):
                # int4 scales group along K (128); int8 has no shape need —
                # the native-kernel dispatch checks its own constraints and
                # falls back to dequant+hipBLASLt otherwise
                if qbits == 4 and mod.weight.shape[1] % 128:
                    continue
                quantize_module_(mod, qbits)
                count += 1
        if self.device == "cuda":
            torch.cuda.empty_cache()
        print(f"[worker] quantized {count} linears to "
              f"{'int8' if qbits == 8 else 'int4-g128'} ({method})", flush=True)

    def add_lora(self, lora_path: str, lora_int_id: int) -> None:
        from .lora import load_lora_adapter

        if self.tp > 1 and self.rank == 0:
            tp_broadcast_object(("add_lora", lora_path, lora_int_id))
        if lora_int_id in self.loras:
            from .lora import invalidate_adapter

            invalidate_adapter(lora_int_id)
        self.loras[lora_int_id] = load_lora_adapter(
            lora_path, lora_int_id,
            device=self.device, dtype=self.model_config.dtype,
            max_lora_rank=self.config.max_lora_rank,
        )

    # ------------------------------------------------------------------
    def _load_weights(self) -> None:
        from .weights import load_safetensors_weights

        weights = load_safetensors_weights(self.model_config.weights_path)
        self.model.load_weights(weights)

    # ------------------------------------------------------------------
    def init_kv_cache(self) -> int:
        """Allocate the paged KV cache; returns the number of blocks."""
        cfg = self.config
        mc = self.model_config
        kv_heads_local = max(1, mc.num_kv_heads // self.tp)
        # fp8 KV cache: e4m3 storage (1 B/elem) — halves the decode HBM
        # stream and doubles block capacity in the 288 GB pool
        kv_dtype = (torch.uint8 if cfg.cache_config.kv_cache_dtype == "fp8"
                    else mc.dtype)
        self.kv_dtype = kv_dtype
        elt = torch.tensor([], dtype=kv_dtype).element_size()
        block_bytes = 2 * mc.num_layers * self.block_size * kv_heads_local * mc.head_dim * elt
        if self.draft is not None:
            # the draft's mirrored cache shares block ids: budget its bytes
            block_bytes += self.draft.cache_bytes_per_block()

        if cfg.cache_config.num_gpu_blocks is not None:
            self.num_blocks = cfg.cache_config.num_gpu_blocks
        elif self.device == "cuda":
            self._profile_peak_memory()
            free, total = torch.cuda.mem_get_info()
            usable = total * cfg.cache_config.gpu_memory_utilization - (total - free)
            self.num_blocks = max(64, int(usable // block_bytes))
        else:
            # CPU tests: enough for max_num_seqs full-length sequences, capped.
            per_seq = (mc.max_model_len + self.block_size - 1) // self.block_size
            self.num_blocks = min(8192, per_seq * cfg.scheduler_config.max_num_seqs + 8)

        if self.tp > 1:
            # ranks may profile slightly different free memory; the block
            # manager (rank 0) must never hand out a block id some rank
            # didn't allocate
            import torch.distributed as dist

            t = torch.tensor(
                [self.num_blocks], dtype=torch.int64,
                device=self.device if self.device == "cuda" else "cpu",
            )
            dist.all_reduce(t, op=dist.ReduceOp.MIN)
            self.num_blocks = int(t.item())

        shape = (self.num_blocks, self.block_size, kv_heads_local, mc.head_dim)
        self.kv_caches = [
            (
                torch.zeros(shape, dtype=kv_dtype, device=self.device),
                torch.zeros(shape, dtype=kv_dtype, device=self.device),
            )
            for _ in range(mc.num_layers)
        ]
        if self.draft is not None:
            self.draft.alloc_cache(self.num_blocks)
        self._maybe_capture_graphs()
        return self.num_blocks

    def _maybe_capture_graphs(self) -> None:
        """hipGraph-capture the decode step per batch bucket (E24).

        TP>1 capture (RCCL collectives inside the graph) is gated behind
        VTA_GRAPH_TP=1 until validated on a multi-GPU box.
        """
        import os

        if self.device != "cuda" or self.config.enforce_eager:
            return
        if os.environ.get("VTA_FORCE_REFERENCE", "0") == "1":
            return  # torch-reference ops sync to host; not capturable
        mc = self.model_config
        if getattr(mc, "expert_parallel", False) and self.tp > 1:
            return  # EP all-to-all host-syncs on split sizes: eager only
        if getattr(mc, "num_experts", 0):
            # the grouped-GEMM MoE path (argsort/bincount/index_add — all
            # shape-static) is capturable; the per-expert fallback loop uses
            # nonzero() and is not
            inter_local = mc.intermediate_size // max(1, self.tp)
            grouped_ok = (
                ops.has_native()
                and mc.dtype == torch.bfloat16
                and inter_local % 128 == 0
                and mc.hidden_size % 128 == 0
                and mc.hidden_size % 64 == 0
            )
            if not grouped_ok:
                return
        if self.tp > 1 and os.environ.get("VTA_GRAPH_TP", "1") != "1":
            return
        from .graph_runner import DecodeGraphRunner

        try:
            self.graph_runner = DecodeGraphRunner(
                self.model, self.kv_caches, self.model_config, self.block_size,
                self.config.scheduler_config.max_num_seqs,
                self.model_config.max_model_len, self.device,
            )
            self.graph_runner.capture()
            if self.tp > 1:
                # RCCL-inside-graph is the risky capture: verify replay
                # against an eager forward on identical inputs before trusting
                # it (all ranks run this concurrently so collectives align)
                self._validate_graph_replay()
        except Exception:
            import traceback

            traceback.print_exc()
            print("[worker] hipGraph capture failed; decode stays eager",
                  flush=True)
            self.graph_runner = None

    @torch.inference_mode()
    def _validate_graph_replay(self) -> None:
        gr = self.graph_runner
        vocab = self.model_config.vocab_size
        for b in {gr.buckets[0], gr.buckets[-1]}:
            ids = (torch.arange(b, device=self.device) * 7919 + 13) % vocab
            pos = torch.zeros(b, dtype=torch.long, device=self.device)
            slots = torch.full((b,), -1, dtype=torch.long, device=self.device)
            seq_lens = torch.zeros(b, dtype=torch.int32, device=self.device)
            bt = torch.zeros((b, gr.max_blocks), dtype=torch.int32,
                             device=self.device)
            gr.in_ids[:b].copy_(ids)
            gr.in_pos[:b].copy_(pos)
            gr.in_slots[:b].copy_(slots)
            gr.in_seq_lens[:b].copy_(seq_lens)
            gr.in_block_tables[:b].zero_()
            hidden = self.model(gr.in_ids[:b], gr.in_pos[:b], self.kv_caches,
                                gr._meta(b))
            eager = self.model.compute_logits(hidden).float()
            replay = gr.run(ids, pos, slots, seq_lens, bt).float()
            torch.cuda.synchronize()
            if not torch.allclose(eager, replay, atol=5e-2, rtol=5e-2):
                raise RuntimeError(
                    f"graph replay mismatch at bucket {b}: "
                    f"max|diff|={float((eager - replay).abs().max())}"
                )

    @torch.inference_mode()
    def _profile_peak_memory(self) -> None:
        """One forward at the max token budget to materialise activations."""
        t = min(self.config.scheduler_config.max_num_batched_tokens, 8192)
        mc = self.model_config
        ids = torch.zeros(t, dtype=torch.long, device=self.device)
        pos = torch.zeros(t, dtype=torch.long, device=self.device)
        meta = AttnMetadata(
            num_prefill_seqs=1, num_prefill_tokens=t, num_decode_seqs=0,
            slot_mapping=torch.arange(t, dtype=torch.long, device=self.device),
            prefill_query_start_loc=torch.tensor([0, t], dtype=torch.int32, device=self.device),
            prefill_seq_lens=torch.tensor([t], dtype=torch.int32, device=self.device),
            prefill_block_tables=torch.arange(
                (t + self.block_size - 1) // self.block_size, dtype=torch.int32,
                device=self.device).unsqueeze(0),
            max_prefill_query_len=t, max_prefill_seq_len=t,
            decode_seq_lens=torch.empty(0, dtype=torch.int32, device=self.device),
            decode_block_tables=torch.empty((0, 0), dtype=torch.int32, device=self.device),
            max_decode_seq_len=0,
        )
        kv_heads_local = max(1, mc.num_kv_heads // self.tp)
        nb = (t + self.block_size - 1) // self.block_size
        shape = (nb, self.block_size, kv_heads_local, mc.head_dim)
        tmp_cache = [
            (torch.zeros(shape, dtype=mc.dtype, device=self.device),
             torch.zeros(shape, dtype=mc.dtype, device=self.device))
            for _ in range(mc.num_layers)
        ]
        hidden = self.model(ids, pos, tmp_cache, meta)
        self.model.compute_logits(hidden[-1:])
        del tmp_cache, hidden
        gc.collect()
        torch.cuda.empty_cache()

    # ------------------------------------------------------------------
    def build_batch(self, sched: SchedulerOutput) -> dict:
        """Rank-0: flatten a SchedulerOutput into broadcastable CPU arrays."""
        prefills = [i for i in sched.items if i.num_new_tokens > 1]
        decodes = [i for i in sched.items if i.num_new_tokens == 1]
        ordered = prefills + decodes

        token_ids: list[int] = []
        positions: list[int] = []
        slot_mapping: list[int] = []
        lora_ids: list[int] = []
        qsl = [0]
        prefill_seq_lens = []
        prefill_tables = []
        decode_seq_lens = []
        decode_tables = []
        sample_rows = []   # row in batch producing the sampled logits, per sampling item
        extra_rows = []    # rows needed for prompt logprobs
        bs = self.block_size

        for it in ordered:
            req = it.request
            s = req.num_computed_tokens
            e = s + it.num_new_tokens
            toks = req.token_slice(s, e)
            token_ids.extend(toks)
            positions.extend(range(s, e))
            lid = req.lora_request.lora_int_id if req.lora_request else 0
            lora_ids.extend([lid] * (e - s))
            for pos in range(s, e):
                b = req.block_ids[pos // bs]
                slot_mapping.append(b * bs + pos % bs)
            if it.num_new_tokens > 1:
                qsl.append(qsl[-1] + it.num_new_tokens)
                prefill_seq_lens.append(e)
                prefill_tables.append(req.block_ids)
            else:
                decode_seq_lens.append(e)
                decode_tables.append(req.block_ids)

        row = 0
        sampling_items: list[ScheduledItem] = []
        spec_items: list[tuple[ScheduledItem, int, int]] = []  # (item, row0, D)
        spec_rows: list[int] = []
        # (item, rows-in-extra-space, prompt token indices, carry_out row or None)
        prompt_lp_specs = []
        for it in ordered:
            req = it.request
            nrow = row + it.num_new_tokens
            if it.samples and req.spec_draft:
                d = len(req.spec_draft)
                # logits rows for the real last token + every draft position
                spec_items.append((it, len(spec_rows), d))
                spec_rows.extend(range(nrow - 1 - d, nrow))
            elif it.samples:
                sample_rows.append(nrow - 1)
                sampling_items.append(it)
            if (
                req.sampling_params.prompt_logprobs is not None
                and req.num_computed_tokens < req.num_prompt_tokens
            ):
                s = req.num_computed_tokens
                e = min(s + it.num_new_tokens, req.num_prompt_tokens)
                # rows for logits at positions [s, e-2] predict tokens s+1..e-1
                lp_rows = list(range(row, row + (e - s) - 1))
                extra_rows.extend(lp_rows)
                carry_out = None
                if e < req.num_prompt_tokens:
                    # still prefilling after this chunk: also need logits at
                    # position e-1 to cover token e next chunk
                    carry_out = row + (e - s) - 1
                    extra_rows.append(carry_out)
                prompt_lp_specs.append((it, lp_rows, list(range(s + 1, e)), carry_out))
            row = nrow

        logit_rows_all = sample_rows + spec_rows + extra_rows
        batch = dict(
            token_ids=token_ids,
            positions=positions,
            slot_mapping=slot_mapping,
            qsl=qsl if len(qsl) > 1 else [0],
            prefill_seq_lens=prefill_seq_lens,
            prefill_tables=prefill_tables,
            decode_seq_lens=decode_seq_lens,
            decode_tables=decode_tables,
            logit_rows=logit_rows_all,
            num_sample_rows=len(sample_rows),
            lora_ids=lora_ids if any(lora_ids) else None,
        )
        # host-only bookkeeping (not broadcast)
        self._sampling_items = sampling_items
        self._spec_items = spec_items
        self._num_spec_rows = len(spec_rows)
        self._prompt_lp_specs = prompt_lp_specs
        self._extra_row_base = len(sample_rows) + len(spec_rows)
        return batch

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def _h2d(self, name: str, arr: np.ndarray, dtype: torch.dtype) -> torch.Tensor:
        """numpy -> device via a persistent pinned staging buffer (true async)."""
        if self.device != "cuda":
            return torch.from_numpy(np.ascontiguousarray(arr)).to(dtype)
        need = int(arr.size)
        key = f"{name}.{self._pin_flip}"
        buf = self._pin.get(key)
        if buf is None or buf.numel() < need:
            buf = torch.empty(max(need, 256), dtype=dtype, pin_memory=True)
            self._pin[key] = buf
        staging = buf[:need]
        staging.copy_(torch.from_numpy(np.ascontiguousarray(arr)).view(-1).to(dtype))
        return staging.to(self.device, non_blocking=True).view(*arr.shape)

    @torch.inference_mode()
    def execute_batch(self, batch: dict) -> Optional[torch.Tensor]:
        """All ranks: run the forward; returns logits rows on every rank."""
        dev = self.device
        self._pin_flip ^= 1
        ids = self._h2d("ids", np.asarray(batch["token_ids"], dtype=np.int64), torch.int64)
        pos = self._h2d("pos", np.asarray(batch["positions"], dtype=np.int64), torch.int64)
        np_seqs = len(batch["prefill_seq_lens"])

        # hipGraph replay path: pure-decode batch, every row samples, no LoRA
        n = len(batch["token_ids"])
        if (
            self.graph_runner is not None
            and np_seqs == 0
            and not batch.get("lora_ids")
            and batch["num_sample_rows"] == n
            and len(batch["logit_rows"]) == n
            and self.graph_runner.bucket_for(n) is not None
        ):
            slots = self._h2d("slots", np.asarray(batch["slot_mapping"], dtype=np.int64), torch.int64)
            seq_lens = self._h2d("dsl", np.asarray(batch["decode_seq_lens"], dtype=np.int32), torch.int32)
            bt = self._h2d("dbt", _pad_np_tables(batch["decode_tables"]), torch.int32)
            return self.graph_runner.run(ids, pos, slots, seq_lens, bt)

        npt = batch["qsl"][-1] if np_seqs else 0
        meta = AttnMetadata(
            num_prefill_seqs=np_seqs,
            num_prefill_tokens=npt,
            num_decode_seqs=len(batch["decode_seq_lens"]),
            slot_mapping=self._h2d("slots", np.asarray(batch["slot_mapping"], dtype=np.int64), torch.int64),
            prefill_query_start_loc=self._h2d("qsl", np.asarray(batch["qsl"], dtype=np.int32), torch.int32),
            prefill_seq_lens=self._h2d("psl", np.asarray(batch["prefill_seq_lens"], dtype=np.int32), torch.int32),
            prefill_block_tables=self._h2d("pbt", _pad_np_tables(batch["prefill_tables"]), torch.int32),
            max_prefill_query_len=max(
                (b - a for a, b in zip(batch["qsl"], batch["qsl"][1:])), default=0),
            max_prefill_seq_len=max(batch["prefill_seq_lens"], default=0),
            decode_seq_lens=self._h2d("dsl", np.asarray(batch["decode_seq_lens"], dtype=np.int32), torch.int32),
            decode_block_tables=self._h2d("dbt", _pad_np_tables(batch["decode_tables"]), torch.int32),
            max_decode_seq_len=max(batch["decode_seq_lens"], default=0),
        )
        from . import lora as lora_rt

        if batch.get("lora_ids"):
            lora_rt.set_context(
                torch.tensor(batch["lora_ids"], dtype=torch.int32, device=dev),
                self.loras,
            )
        try:
            hidden = self.model(ids, pos, self.kv_caches, meta)
        finally:
            lora_rt.clear_context()
        logit_rows = batch["logit_rows"]
        if not logit_rows:
            return None
        rows = torch.tensor(logit_rows, dtype=torch.long, device=dev)
        return self.model.compute_logits(hidden[rows])

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def execute_begin(self, sched: SchedulerOutput):
        """Launch one step WITHOUT host sync where possible.

        Returns an object with ``.finish() -> ExecuteResult``.  When the whole
        sampling batch takes the fused kernel, finish() only does the final
        device->host copy — the pipelined engine step calls it AFTER doing
        the previous step's host postprocessing, overlapping that work with
        this step's GPU execution.  Slow-path batches (logprobs, penalties,
        spec rows, prompt logprobs) complete synchronously inside this call.
        """
        import time as _time

        _tb = _time.perf_counter()
        batch = self.build_batch(sched)
        self.last_build_time = _time.perf_counter() - _tb
        if self.tp > 1:
            import torch.distributed as dist

            header, payload = pack_batch(batch)
            tp_broadcast_object(("execute_packed", header.tolist()))
            t = torch.from_numpy(payload).to(
                self.device if self.device == "cuda" else "cpu")
            dist.broadcast(t, src=0)
        import os as _os
        import time as _time

        _dbg = _os.environ.get("VTA_LAUNCH_TIMING", "0") == "1"
        if _dbg:
            _t1 = _time.perf_counter()
        with roctx.trace_range("worker.forward"):
            logits = self.execute_batch(batch)
        if _dbg:
            _t2 = _time.perf_counter()

        ns = batch["num_sample_rows"]
        if ns and not self._spec_items and not self._prompt_lp_specs:
            sampling_reqs = [it.request for it in self._sampling_items]
            with roctx.trace_range("worker.sample"):
                fused = self.sampler.try_launch_fused(logits[:ns], sampling_reqs)
            if fused is not None:
                if _dbg:
                    _t3 = _time.perf_counter()
                    d = self.__dict__.setdefault("_lt", [0.0, 0.0, 0.0, 0])
                    d[0] += _t1 - _tb
                    d[1] += _t2 - _t1
                    d[2] += _t3 - _t2
                    d[3] += 1
                    if d[3] >= 128:
                        import sys as _sys

                        print(f"[launch-timing] per-step ms build={d[0]/d[3]*1e3:.2f} "
                              f"forward-launch={d[1]/d[3]*1e3:.2f} "
                              f"sampler-launch={d[2]/d[3]*1e3:.2f}",
                              file=_sys.stderr, flush=True)
                        self._lt = [0.0, 0.0, 0.0, 0]
                return _PendingStep(fused, self.sampler)
        return _DoneStep(self._execute_finish(batch, logits))

    def execute(self, sched: SchedulerOutput) -> ExecuteResult:
        """Rank-0 entry point for one engine step (synchronous)."""
        return self.execute_begin(sched).finish()

    def _execute_finish(self, batch: dict, logits) -> ExecuteResult:
        ns = batch["num_sample_rows"]
        sampling_reqs = [it.request for it in self._sampling_items]
        if ns:
            sampler_out = self.sampler.sample(logits[:ns], sampling_reqs)
        else:
            sampler_out = SamplerOutput(token_ids=[], logprobs=[])

        # speculative verification (E17): greedy argmax over the draft rows;
        # the engine accepts the longest matching prefix + the bonus token
        spec_results: list[tuple[ScheduledItem, list[int]]] = []
        if self._spec_items:
            srows = logits[ns:ns + self._num_spec_rows]
            preds = torch.argmax(srows, dim=-1).tolist()
            for it, row0, d in self._spec_items:
                spec_results.append((it, preds[row0:row0 + d + 1]))

        # prompt logprobs (E8): extra logits rows follow the sampling rows,
        # in the order build_batch appended them to extra_rows.
        off = self._extra_row_base
        for it, lp_rows, token_idx, carry_out in self._prompt_lp_specs:
            req = it.request
            num_lp = req.sampling_params.prompt_logprobs or 1
            s = req.num_computed_tokens
            # token at chunk start: logits carried from the previous chunk
            if s > 0 and req._plp_carry is not None:
                req.prompt_logprobs.extend(
                    prompt_logprob_dicts(
                        req._plp_carry.unsqueeze(0), [req.prompt_token_ids[s]], num_lp
                    )
                )
            req._plp_carry = None
            n = len(lp_rows)
            if n:
                chunk_rows = logits[off: off + n].float().log_softmax(dim=-1)
                off += n
                actual = [req.prompt_token_ids[j] for j in token_idx]
                req.prompt_logprobs.extend(
                    prompt_logprob_dicts(chunk_rows, actual, num_lp)
                )
            if carry_out is not None:
                req._plp_carry = logits[off].float().log_softmax(dim=-1)
                off += 1

        return ExecuteResult(sampler_output=sampler_out, spec_results=spec_results)

    # ------------------------------------------------------------------
    def worker_loop(self) -> None:
        """Non-zero TP ranks: execute broadcast commands until stopped."""
        assert self.rank != 0
        while True:
            cmd = tp_broadcast_object(None)
            if cmd is None:
                continue
            kind = cmd[0]
            if kind == "execute":
                self.execute_batch(cmd[1])
            elif kind == "execute_packed":
                self.execute_batch(self.recv_packed_batch(cmd[1]))
            elif kind == "add_lora":
                self.add_lora(cmd[1], cmd[2])
            elif kind == "barrier":
                import torch.distributed as dist

                dist.barrier()
                if self.device == "cuda":
                    torch.cuda.synchronize()
            elif kind == "stop":
                return

    def recv_packed_batch(self, header_list: list[int]):
        """Non-zero ranks: receive the packed step batch broadcast."""
        import torch.distributed as dist

        header = np.asarray(header_list, dtype=np.int64)
        total = int(header[7:].sum())
        t = torch.empty(
            total, dtype=torch.int64,
            device=self.device if self.device == "cuda" else "cpu")
        dist.broadcast(t, src=0)
        return unpack_batch(header, t.cpu().numpy())

    def stop_workers(self) -> None:
        if self.tp > 1 and self.rank == 0:
            tp_broadcast_object(("stop",))
