"""Sampling pipeline.

Per-request temperature / top-k / top-p / seeded RNG / repetition penalty /
min-tokens EOS suppression / logits processors / logprob+rank+top-N
extraction, vectorised over the step's sampling rows.  All tensor math runs
on-device; the per-position logprob dicts the TGIS wire format needs
(grpc_server.py:701-756) are assembled host-side from one batched top-k.

Greedy and plain temperature-sampling rows take the fused HIP kernel
(kernels/sampling.hip, exponential-race identity) in one pass; top-k/top-p,
penalties, processors and logprob extraction run the vectorised torch path.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch

from .request import Request
from .types import Logprob, PosLogprobs

_NEG_INF = float("-inf")

import os as _os_mod

# deferred (async) logprob extraction on the fused path; VTA_DEFER_LP=0
# falls back to the synchronous slow-path extraction
_DEFER_LP = _os_mod.environ.get("VTA_DEFER_LP", "1") == "1"


@dataclass
class SamplerOutput:
    token_ids: list[int]
    # per sampling row: logprob info for the sampled token (or None)
    logprobs: list[Optional[PosLogprobs]]


def _apply_top_k_top_p(logits: torch.Tensor, top_k: torch.Tensor, top_p: torch.Tensor) -> torch.Tensor:
    """Mask logits outside top-k / nucleus top-p, rows vectorised."""
    vocab = logits.shape[-1]
    sorted_logits, sorted_idx = torch.sort(logits, dim=-1, descending=True)
    # top-k mask
    ranks = torch.arange(vocab, device=logits.device).unsqueeze(0)
    k = top_k.clamp(min=1, max=vocab).unsqueeze(1)
    mask = ranks >= k
    # top-p mask on the sorted probabilities
    probs = torch.softmax(sorted_logits, dim=-1)
    cum = probs.cumsum(dim=-1)
    pmask = (cum - probs) > top_p.unsqueeze(1)  # keep tokens whose prefix-mass < p
    mask |= pmask
    mask[:, 0] = False  # always keep the argmax token
    sorted_logits = sorted_logits.masked_fill(mask, _NEG_INF)
    return torch.full_like(logits, _NEG_INF).scatter_(-1, sorted_idx, sorted_logits)


class Sampler:
    def __init__(self, device: str, max_logprobs: int = 11):
        self.device = device
        self.max_logprobs = max_logprobs
        # pinned staging (double-buffered): pageable host->device uploads are
        # silently synchronous and would stall the pipelined step's launch
        self._pin: dict = {}
        self._pin_flip = 0
        self._pending_lp = None  # deferred logprob tensors (finish_fused)

    def _h2d_list(self, name: str, data: list, dtype: torch.dtype) -> torch.Tensor:
        if self.device != "cuda":
            return torch.tensor(data, dtype=dtype)
        n = len(data)
        key = f"{name}.{self._pin_flip}"
        buf = self._pin.get(key)
        if buf is None or buf.numel() < n:
            buf = torch.empty(max(n, 256), dtype=dtype, pin_memory=True)
            self._pin[key] = buf
        staging = buf[:n]
        staging.copy_(torch.tensor(data, dtype=dtype))
        return staging.to(self.device, non_blocking=True)

    def _generator_for(self, req: Request) -> Optional[torch.Generator]:
        if req.sampling_params.seed is None:
            return None
        if req.generator is None:
            g = torch.Generator(device=self.device)
            g.manual_seed(req.sampling_params.seed)
            req.generator = g
        return req.generator

    @torch.inference_mode()
    def try_launch_fused(
        self, logits: torch.Tensor, requests: list[Request]
    ) -> Optional[torch.Tensor]:
        """Launch the fused one-pass sampler WITHOUT host sync.

        Returns the device token-id tensor (caller syncs via .tolist()) when
        every row is greedy or plain temperature sampling — no top-k/top-p,
        penalties, processors or guided masks — else None.  Rows that
        request logprobs/top-N stay ON this path: their log-softmax / top-k
        / rank tensors are launched asynchronously here and synced later in
        ``finish_fused`` (the TGIS wire always wants top-N details, and
        dropping to the sync path would stall the pipelined step).  The
        pipelined engine step uses this to overlap host postprocessing with
        the next step's GPU work.
        """
        from .. import ops as _ops

        n, vocab = logits.shape
        simple = True
        any_sampling = False
        ban_rows: list[int] = []  # rows with min_tokens pending: EOS banned
        ban_eos: list[int] = []
        lp_rows: list[int] = []   # rows that want logprob details (deferred)
        for i, req in enumerate(requests):
            p = req.sampling_params
            if (
                p.repetition_penalty != 1.0
                or p.logits_processors
                or req.guided_state is not None
            ):
                simple = False
                break
            if p.logprobs is not None:
                if not _DEFER_LP:  # kill switch: old behavior = slow path
                    simple = False
                    break
                lp_rows.append(i)
            if (p.min_tokens and req.num_output_tokens < p.min_tokens
                    and req.eos_token_id is not None):
                ban_rows.append(i)
                ban_eos.append(req.eos_token_id)
            if p.temperature != 0.0:
                any_sampling = True
                if not (p.top_k <= 0 or p.top_k >= vocab) or p.top_p < 1.0:
                    simple = False
                    break
        if not simple:
            self._pending_lp = None
            return None
        import os as _os
        import time as _time

        _dbg = _os.environ.get("VTA_LAUNCH_TIMING", "0") == "1"
        if _dbg:
            _ta = _time.perf_counter()
        self._pin_flip ^= 1
        if ban_rows:
            # min_tokens EOS suppression as ONE batched index_fill_ so the
            # whole batch stays on the fused path.  NOTE: advanced-index
            # assignment (index_put_) host-synchronizes on ROCm — it cost a
            # full hidden GPU-step stall in the pipelined launch phase;
            # flat index_fill_ stays asynchronous.
            flat = [r * vocab + e for r, e in zip(ban_rows, ban_eos)]
            logits.view(-1).index_fill_(
                0, self._h2d_list("ban", flat, torch.long), _NEG_INF)
        if not _ops.native_enabled(logits):
            # CPU / force-reference: plain argmax-or-race, still deferred-sync
            if not any_sampling:
                out = torch.argmax(logits, dim=-1)
            else:
                probs = torch.softmax(
                    logits.float() / torch.tensor(
                        [max(r.sampling_params.temperature, 1e-6)
                         for r in requests]
                    ).unsqueeze(1),
                    dim=-1,
                )
                q = torch.empty_like(probs)
                for i, req in enumerate(requests):
                    g = self._generator_for(req)
                    if g is None:
                        q[i].exponential_()
                    else:
                        q[i].exponential_(generator=g)
                out = torch.argmax(probs / q, dim=-1)
            self._launch_deferred_logprobs(logits, out, lp_rows, requests)
            return out
        temps = self._h2d_list(
            "temps", [r.sampling_params.temperature for r in requests],
            torch.float32,
        )
        noise = None
        if any_sampling:
            noise = torch.empty((n, vocab), dtype=torch.float32, device=logits.device)
            for i, req in enumerate(requests):
                if req.sampling_params.temperature != 0.0:
                    g = self._generator_for(req)
                    if g is None:
                        noise[i].exponential_()
                    else:
                        noise[i].exponential_(generator=g)
        if _dbg:
            _tb2 = _time.perf_counter()
        out = torch.empty(n, dtype=torch.long, device=logits.device)
        if _dbg:
            _tc = _time.perf_counter()
        _ops.sample_argmax(out, logits, temps, noise)
        if _dbg:
            _td = _time.perf_counter()
            d = self.__dict__.setdefault("_lt2", [0.0, 0.0, 0.0, 0])
            d[0] += _tb2 - _ta
            d[1] += _tc - _tb2
            d[2] += _td - _tc
            d[3] += 1
            if d[3] >= 128:
                import sys as _sys

                print(f"[sampler-timing] per-step ms ban+temps={d[0]/d[3]*1e3:.2f} "
                      f"empty={d[1]/d[3]*1e3:.2f} kernel-launch={d[2]/d[3]*1e3:.2f}",
                      file=_sys.stderr, flush=True)
                self._lt2 = [0.0, 0.0, 0.0, 0]
        self._launch_deferred_logprobs(logits, out, lp_rows, requests)
        return out

    def _launch_deferred_logprobs(
        self, logits: torch.Tensor, out: torch.Tensor,
        lp_rows: list[int], requests: list[Request],
    ) -> None:
        """Launch log-softmax/top-k/rank work for logprob rows WITHOUT a
        host sync; ``finish_fused`` syncs and builds the wire dicts.  Uses
        the post-min_tokens-ban logits — same semantics as the sync path
        (``sample`` bans EOS before its log_softmax too)."""
        self._pending_lp = None
        if not lp_rows:
            return
        from .. import ops as _ops

        n, vocab = logits.shape
        k = min(self.max_logprobs, vocab)
        if len(lp_rows) == n:
            rows = logits
            chosen = out
        else:
            idx = self._h2d_list("lp_rows", lp_rows, torch.long)
            rows = logits.index_select(0, idx)
            chosen = out.index_select(0, idx)
        if _ops.logsoftmax_topk_usable(rows):
            # one-HBM-pass HIP kernel (E8): log-softmax + top-K + rank
            topv, topi, chosen_lp, ranks = _ops.logsoftmax_topk(
                rows, chosen, k)
        else:
            lp = torch.log_softmax(rows.float(), dim=-1)
            topv, topi = torch.topk(lp, k, dim=-1)
            chosen_lp = lp.gather(1, chosen.unsqueeze(1)).squeeze(1)
            ranks = (lp > chosen_lp.unsqueeze(1)).sum(dim=-1) + 1
        self._pending_lp = (
            topv, topi, chosen_lp, ranks, list(lp_rows),
            [requests[i].sampling_params.logprobs for i in lp_rows],
        )

    def finish_fused(self, token_ids: list[int]) -> SamplerOutput:
        """Sync the deferred logprob tensors (if any) and build the
        per-position dicts the wire format needs."""
        out_logprobs: list[Optional[PosLogprobs]] = [None] * len(token_ids)
        pend = getattr(self, "_pending_lp", None)
        self._pending_lp = None
        if pend is not None:
            topv, topi, chosen_lp, ranks, rows_idx, nums = pend
            topv = topv.tolist()
            topi = topi.tolist()
            chosen_lp = chosen_lp.tolist()
            ranks = ranks.tolist()
            for row, i in enumerate(rows_idx):
                d: PosLogprobs = {}
                for j in range(min(nums[row], len(topi[row]))):
                    d[topi[row][j]] = Logprob(logprob=topv[row][j], rank=j + 1)
                tok = token_ids[i]
                if tok not in d:
                    d[tok] = Logprob(logprob=chosen_lp[row], rank=ranks[row])
                out_logprobs[i] = d
        return SamplerOutput(token_ids=token_ids, logprobs=out_logprobs)

    @torch.inference_mode()
    def sample(self, logits: torch.Tensor, requests: list[Request]) -> SamplerOutput:
        """logits: [N, vocab] raw lm-head outputs for the N sampling rows.

        The all-greedy no-extras batch (the serving steady state) stays in
        the lm-head dtype end to end — one argmax pass, no [N, vocab] f32
        materialisation (2x 131 MB per step at batch 256 / 128k vocab).
        """
        n, vocab = logits.shape
        assert n == len(requests)

        fused = self.try_launch_fused(logits, requests)
        if fused is not None:
            return self.finish_fused(fused.tolist())

        logits = logits.float()

        # --- per-request host-side adjustments (rare paths) -------------
        for i, req in enumerate(requests):
            p = req.sampling_params
            # repetition penalty over prompt + generated tokens
            if p.repetition_penalty != 1.0:
                ids = torch.tensor(
                    sorted(set(req.all_token_ids)), device=logits.device, dtype=torch.long
                )
                row = logits[i]
                vals = row[ids]
                row[ids] = torch.where(
                    vals > 0, vals / p.repetition_penalty, vals * p.repetition_penalty
                )
            # custom logits processors (typical_p, length penalty, ...)
            if p.logits_processors:
                row = logits[i]
                for proc in p.logits_processors:
                    row = proc(req.output_token_ids, row)
                logits[i] = row
            # guided decoding token mask
            if req.guided_state is not None:
                allowed = req.guided_state.allowed_token_ids()
                if allowed is not None:
                    mask = torch.full((vocab,), _NEG_INF, device=logits.device)
                    idx = torch.tensor(allowed, device=logits.device, dtype=torch.long)
                    mask[idx] = 0.0
                    logits[i] += mask
            # min_tokens: suppress EOS until satisfied
            if p.min_tokens and req.num_output_tokens < p.min_tokens and req.eos_token_id is not None:
                logits[i, req.eos_token_id] = _NEG_INF

        # --- logprob extraction uses pre-temperature logits --------------
        need_lp = [i for i, r in enumerate(requests) if r.sampling_params.logprobs is not None]
        logprob_rows = torch.log_softmax(logits[need_lp], dim=-1) if need_lp else None

        # --- sampling -----------------------------------------------------
        temps = torch.tensor(
            [r.sampling_params.temperature for r in requests],
            device=logits.device, dtype=torch.float,
        )
        greedy_mask = temps == 0.0
        sampled = torch.empty(n, dtype=torch.long, device=logits.device)

        if bool(greedy_mask.any()):
            sampled[greedy_mask] = torch.argmax(logits[greedy_mask], dim=-1)

        sample_idx = (~greedy_mask).nonzero(as_tuple=True)[0]
        if len(sample_idx):
            sub = logits[sample_idx] / temps[sample_idx].unsqueeze(1)
            top_k = torch.tensor(
                [
                    requests[i].sampling_params.top_k
                    if requests[i].sampling_params.top_k > 0 else vocab
                    for i in sample_idx.tolist()
                ],
                device=logits.device,
            )
            top_p = torch.tensor(
                [requests[i].sampling_params.top_p for i in sample_idx.tolist()],
                device=logits.device, dtype=torch.float,
            )
            if bool((top_k < vocab).any()) or bool((top_p < 1.0).any()):
                sub = _apply_top_k_top_p(sub, top_k, top_p)
            probs = torch.softmax(sub, dim=-1)
            # exponential-race sampling; per-request generator when seeded
            q = torch.empty_like(probs)
            rows_with_seed = [
                (j, self._generator_for(requests[i]))
                for j, i in enumerate(sample_idx.tolist())
            ]
            if all(g is None for _, g in rows_with_seed):
                q.exponential_()
            else:
                for j, g in rows_with_seed:
                    q[j].exponential_(generator=g)
            sampled[sample_idx] = torch.argmax(probs / q, dim=-1)

        # --- assemble host results ---------------------------------------
        sampled_cpu = sampled.tolist()
        out_logprobs: list[Optional[PosLogprobs]] = [None] * n
        if need_lp:
            k = self.max_logprobs
            lp = logprob_rows
            chosen = sampled[need_lp]
            chosen_lp = lp.gather(1, chosen.unsqueeze(1)).squeeze(1)
            ranks = (lp > chosen_lp.unsqueeze(1)).sum(dim=-1) + 1
            topv, topi = torch.topk(lp, min(k, vocab), dim=-1)
            topv = topv.tolist()
            topi = topi.tolist()
            chosen_lp = chosen_lp.tolist()
            ranks = ranks.tolist()
            for row, i in enumerate(need_lp):
                req = requests[i]
                num = req.sampling_params.logprobs
                d: PosLogprobs = {}
                for j in range(min(num, len(topi[row]))):
                    d[topi[row][j]] = Logprob(logprob=topv[row][j], rank=j + 1)
                tok = sampled_cpu[i]
                if tok not in d:
                    d[tok] = Logprob(logprob=chosen_lp[row], rank=ranks[row])
                out_logprobs[i] = d
        return SamplerOutput(token_ids=sampled_cpu, logprobs=out_logprobs)


def prompt_logprob_dicts(
    logprob_rows: torch.Tensor,  # [M, vocab] log-softmax rows
    actual_ids: list[int],
    num_logprobs: int,
) -> list[PosLogprobs]:
    """Build per-position logprob dicts for prompt tokens (E8)."""
    m, vocab = logprob_rows.shape
    k = min(max(num_logprobs, 1), vocab)
    topv, topi = torch.topk(logprob_rows, k, dim=-1)
    ids_t = torch.tensor(actual_ids, device=logprob_rows.device, dtype=torch.long)
    actual_lp = logprob_rows.gather(1, ids_t.unsqueeze(1)).squeeze(1)
    ranks = (logprob_rows > actual_lp.unsqueeze(1)).sum(dim=-1) + 1
    topv = topv.tolist()
    topi = topi.tolist()
    actual_lp = actual_lp.tolist()
    ranks = ranks.tolist()
    out = []
    for r in range(m):
        d: PosLogprobs = {}
        for j in range(min(num_logprobs, len(topi[r]))):
            d[topi[r][j]] = Logprob(logprob=topv[r][j], rank=j + 1)
        tok = actual_ids[r]
        if tok not in d:
            d[tok] = Logprob(logprob=actual_lp[r], rank=ranks[r])
        out.append(d)
    return out
