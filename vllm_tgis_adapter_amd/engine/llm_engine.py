"""Synchronous engine: scheduler + worker + postprocessing per step.

This is the layer the reference delegates to vLLM for (EngineClient's
engine side); here it is built natively: continuous batching (E1), paged KV
(E5), incremental detokenization + stop handling (E10), per-step sampling
(E7/E8) and request metrics (E21).
"""

from __future__ import annotations

import time
from typing import Optional

from ..parallel import get_tp_rank, init_distributed
from .block_manager import BlockManager
from .config import EngineConfig
from .detokenizer import Detokenizer, StopChecker
from .request import Request
from .scheduler import Scheduler
from . import roctx
from .tokenizer import get_tokenizer
from .types import LoRARequest, RequestOutput, SamplingParams
from .worker import Worker


class LLMEngine:
    def __init__(self, config: EngineConfig):
        self.config = config
        self.model_config = config.model_config
        device = config.resolve_device()
        init_distributed(config.tensor_parallel_size, device=device)
        self.rank = get_tp_rank()

        self.tokenizer = get_tokenizer(self.model_config)
        self.worker = Worker(config)
        num_blocks = self.worker.init_kv_cache()
        self.block_manager = BlockManager(
            num_blocks, config.cache_config.block_size,
            enable_prefix_caching=config.cache_config.enable_prefix_caching,
        )
        self.scheduler = Scheduler(config.scheduler_config, self.block_manager)
        self.detokenizer = Detokenizer(self.tokenizer)
        self.stop_checker = StopChecker(self.model_config.max_model_len)
        self.eos_token_id = self.tokenizer.eos_token_id
        from . import spec as _spec
        from .metrics import EngineMetrics

        from .draft import is_draft_model_spec

        self.spec_draft_model = is_draft_model_spec(config.speculative_model)
        self.spec_enabled = (
            _spec.is_ngram_spec(config.speculative_model) or self.spec_draft_model
        )
        self.spec_k = config.speculative_num_tokens

        self.metrics = EngineMetrics(self.model_config.model)
        # pipelined stepping (overlap host postprocess with the next step's
        # GPU work): deferred (items, [(item, token_id, logprobs)]) of the
        # previous step, drained while the current step runs on-device
        self._deferred = None
        import os as _os2

        self._pipeline_enabled = _os2.environ.get("VTA_PIPELINE", "1") == "1"
        self._pipeline_min = int(_os2.environ.get("VTA_PIPELINE_MIN", "64"))
        # optional per-phase step timing (bench --timing): phase -> seconds
        self.phase_times: Optional[dict] = None
        import os as _os

        if _os.environ.get("VTA_STEP_TIMING", "0") == "1":
            from collections import defaultdict

            self.phase_times = defaultdict(float)
            self._timing_autoprint = True
        else:
            self._timing_autoprint = False

    # ------------------------------------------------------------------
    def add_request(
        self,
        request_id: str,
        prompt: Optional[str],
        prompt_token_ids: list[int],
        sampling_params: SamplingParams,
        arrival_time: Optional[float] = None,
        lora_request: Optional[LoRARequest] = None,
        trace_headers: Optional[dict] = None,
    ) -> Request:
        req = Request(
            request_id=request_id,
            prompt=prompt,
            prompt_token_ids=prompt_token_ids,
            sampling_params=sampling_params,
            arrival_time=arrival_time,
            lora_request=lora_request,
            trace_headers=trace_headers,
        )
        req.eos_token_id = self.eos_token_id
        if sampling_params.structured_outputs is not None:
            from .guided import build_guided_state

            req.guided_state = build_guided_state(
                sampling_params.structured_outputs, self.tokenizer
            )
        self.scheduler.add_request(req)
        return req

    def add_lora(self, lora_request: LoRARequest) -> None:
        self.worker.add_lora(lora_request.lora_path, lora_request.lora_int_id)

    def abort_request(self, request_id: str) -> Optional[RequestOutput]:
        req = self.scheduler.abort_request(request_id)
        if req is None:
            return None
        return req.make_output()

    def has_unfinished(self) -> bool:
        return self.scheduler.has_unfinished() or self._deferred is not None

    # ------------------------------------------------------------------
    # Benchmark step-window (driver contract): time EXACTLY `timed_steps`
    # engine steps, barrier+synchronize-bracketed on both sides, after
    # `warmup_steps` armed warmup steps.  Timestamps are CLOCK_MONOTONIC
    # (system-wide on Linux) so a client process can window its own
    # delivery timestamps against them.
    def arm_bench_window(self, warmup_steps: int, timed_steps: int) -> None:
        self._bench_win = {
            "warmup": int(warmup_steps), "steps": int(timed_steps),
            "count": 0, "t0": None, "t1": None, "produced": 0, "result": None,
        }

    def bench_window_result(self) -> Optional[dict]:
        bw = getattr(self, "_bench_win", None)
        return bw["result"] if bw else None

    def _bench_barrier(self) -> None:
        if self.config.tensor_parallel_size > 1:
            import torch.distributed as dist

            from ..parallel import tp_broadcast_object

            tp_broadcast_object(("barrier",))
            dist.barrier()
        if self.worker.device == "cuda":
            import torch

            torch.cuda.synchronize()

    def _bench_tick(self, produced_this_step: int) -> None:
        bw = self._bench_win
        if bw["result"] is not None:
            return
        if bw["t0"] is not None:
            bw["produced"] += produced_this_step
        bw["count"] += 1
        if bw["count"] == bw["warmup"]:
            self._bench_barrier()
            bw["t0"] = time.monotonic()
        elif bw["count"] == bw["warmup"] + bw["steps"]:
            self._bench_barrier()
            bw["t1"] = time.monotonic()
            bw["result"] = {
                "t0": bw["t0"], "t1": bw["t1"],
                "elapsed_s": bw["t1"] - bw["t0"],
                "steps": bw["steps"], "produced": bw["produced"],
            }

    # ------------------------------------------------------------------
    def step(self) -> list[RequestOutput]:
        with roctx.trace_range("engine.step"):
            if (
                self._pipeline_enabled
                and not self.spec_enabled
                and (self._deferred is not None
                     or len(self.scheduler.running) >= self._pipeline_min)
            ):
                return self._step_pipelined()
            return self._step_sync()

    # ------------------------------------------------------------------
    def _step_pipelined(self) -> list[RequestOutput]:
        """Overlapped step: launch this step's GPU work, drain the PREVIOUS
        step's host postprocessing (detok, stop strings, output building)
        while it runs, then sync this step's sampled tokens.

        Wire behavior is identical except that outputs are delivered one
        engine step later and stop-STRING finishes are decided one step
        later — always before the next token would be appended, so counts
        and text are exact (EOS and no-stop-length finishes stay immediate
        via StopChecker.check_cheap).  Engages only at >= VTA_PIPELINE_MIN
        running requests; spec decode and empty batches use the sync path.
        """
        pt = self.phase_times
        if pt is not None:
            t0 = time.perf_counter()
        sched = self.scheduler.schedule()
        if sched.is_empty:
            return self._drain_deferred()
        pending = self.worker.execute_begin(sched)
        for it in sched.items:
            it.request.num_computed_tokens += it.num_new_tokens
            self.block_manager.register_prefix(it.request)
        if pt is not None:
            t1 = time.perf_counter()
            pt["schedule+launch"] += t1 - t0

        outputs = self._drain_deferred()
        if pt is not None:
            t2 = time.perf_counter()
            pt["drain(overlapped)"] += t2 - t1

        result = pending.finish()
        now = time.time()
        sampler_out = result.sampler_output
        pairs = []
        for it, token_id, lp in zip(
            self.worker._sampling_items, sampler_out.token_ids,
            sampler_out.logprobs,
        ):
            req = it.request
            if req.status.is_finished:  # aborted / late-stop before append
                continue
            if req.metrics.first_token_time is None:
                req.metrics.first_token_time = now
            req.metrics.last_token_time = now
            req.output_token_ids.append(token_id)
            if req.logprobs is not None:
                req.logprobs.append(lp)
                if lp and token_id in lp:
                    req.cumulative_logprob += lp[token_id].logprob
            if req.guided_state is not None:
                req.guided_state.advance(token_id)
            self.stop_checker.check_cheap(req, token_id)
            if req.status.is_finished and req in self.scheduler.running:
                # de-schedule NOW: deferring finish_request one step let the
                # scheduler hand this request another slot and the finish
                # accounting run twice (metrics double-counted)
                self.scheduler.running.remove(req)
            pairs.append((it, token_id))
        self._deferred = (sched.items, pairs)
        if pt is not None:
            t3 = time.perf_counter()
            pt["sync+append"] += t3 - t2
            pt["steps"] += 1
            pt["tokens"] += len(sampler_out.token_ids)
            if self._timing_autoprint and pt["steps"] >= 128:
                import sys as _sys

                n = pt.pop("steps")
                toks = pt.pop("tokens", 0)
                pt.pop("prefill_steps", 0)
                parts = {k: round(v / n * 1e3, 2) for k, v in pt.items()}
                print(f"[step-timing/pipelined] {n} steps ({toks} tokens): "
                      f"per-step ms {parts}", file=_sys.stderr, flush=True)
                pt.clear()
        if getattr(self, "_bench_win", None) is not None:
            self._bench_tick(len(sampler_out.token_ids))
        self.metrics.num_running.set(len(self.scheduler.running))
        self.metrics.num_waiting.set(len(self.scheduler.waiting))
        self.metrics.kv_usage.set(
            1.0 - self.block_manager.num_free_blocks / max(1, self.block_manager.num_blocks)
        )
        return outputs

    def _drain_deferred(self) -> list[RequestOutput]:
        d = self._deferred
        self._deferred = None
        if d is None:
            return []
        items, pairs = d
        from .request import RequestStatus

        for it, token_id in pairs:
            req = it.request
            if req.status == RequestStatus.FINISHED_ABORTED:
                # aborted since; detok state no longer matters
                continue
            new_text = self.detokenizer.append_token(req, token_id)
            if not req.status.is_finished:
                self.stop_checker.check_text(req, new_text)
            # length/EOS finishes decided in the sync-append phase still
            # need this (their final) token's text in the output — skipping
            # them dropped the last text piece from every cheap finish

        outputs: list[RequestOutput] = []
        seen = set()
        for it in items:
            req = it.request
            if id(req) in seen:
                continue
            seen.add(id(req))
            out = req.make_output()
            if out is not None:
                outputs.append(out)
            if req.status.is_finished:
                self._account_finish(req)
        return outputs

    def _account_finish(self, req: Request) -> None:
        # Idempotent finish bookkeeping: a stop-string finish decided in a
        # drain can see the request again in the next step's deferred list.
        if getattr(req, "_finish_accounted", False):
            return
        req._finish_accounted = True
        self.scheduler.finish_request(req)
        m = req.metrics
        self.metrics.request_success.inc()
        self.metrics.prompt_tokens.inc(req.num_prompt_tokens)
        self.metrics.generation_tokens.inc(req.num_output_tokens)
        if m.first_token_time and m.first_scheduled_time:
            self.metrics.ttft.observe(m.first_token_time - m.arrival_time)
            if req.num_output_tokens > 1 and m.last_token_time:
                self.metrics.time_per_output_token.observe(
                    (m.last_token_time - m.first_token_time)
                    / (req.num_output_tokens - 1)
                )

    # ------------------------------------------------------------------
    def _step_sync(self) -> list[RequestOutput]:
        pt = self.phase_times
        if pt is not None:
            import torch as _torch

            _sync = _torch.cuda.synchronize if self.worker.device == "cuda" else (lambda: None)
            _sync()
            t0 = time.perf_counter()
        if self.spec_enabled:
            from . import spec as _spec

            if self.spec_draft_model:
                eligible = []
                for req in self.scheduler.running:
                    req.spec_draft = []
                    if _spec.eligible(req):
                        eligible.append(req)
                if eligible:
                    self.worker.draft.propose(eligible, self.spec_k)
            else:
                for req in self.scheduler.running:
                    req.spec_draft = (
                        _spec.propose(req, self.spec_k, self.model_config.max_model_len)
                        if _spec.eligible(req) else []
                    )
        sched = self.scheduler.schedule()
        if sched.is_empty:
            return []
        if pt is not None:
            t1 = time.perf_counter()
            pt["schedule"] += t1 - t0
        result = self.worker.execute(sched)
        for it in sched.items:
            it.request.num_computed_tokens += it.num_new_tokens
            self.block_manager.register_prefix(it.request)
        if pt is not None:
            _sync()
            t2 = time.perf_counter()
            pt["execute+sample"] += t2 - t1
            pt["build_batch"] += getattr(self.worker, "last_build_time", 0.0)

        now = time.time()
        spec_token_count = 0

        # speculative results: accept the longest matching draft prefix plus
        # the model's bonus token; roll computed back past rejected draft KV
        for it, preds in (result.spec_results or []):
            req = it.request
            draft = req.spec_draft
            req.spec_draft = []
            if req.status.is_finished:
                continue
            accepted = 0
            while accepted < len(draft) and preds[accepted] == draft[accepted]:
                accepted += 1
            req.num_computed_tokens -= len(draft) - accepted
            new_tokens = draft[:accepted] + [preds[accepted]]
            spec_token_count += len(new_tokens)
            if self.spec_draft_model:
                # draft KV beyond the accepted prefix diverged: roll back
                req.draft_computed = min(req.draft_computed,
                                         req._draft_base + accepted)
            if req.metrics.first_token_time is None:
                req.metrics.first_token_time = now
            req.metrics.last_token_time = now
            for tok in new_tokens:
                if req.status.is_finished:
                    break
                req.output_token_ids.append(tok)
                new_text = self.detokenizer.append_token(req, tok)
                self.stop_checker.check(req, tok, new_text)

        sampler_out = result.sampler_output
        for it, token_id, lp in zip(
            self.worker._sampling_items, sampler_out.token_ids, sampler_out.logprobs
        ):
            req = it.request
            if req.status.is_finished:  # aborted mid-step
                continue
            if req.metrics.first_token_time is None:
                req.metrics.first_token_time = now
            req.metrics.last_token_time = now
            req.output_token_ids.append(token_id)
            if req.logprobs is not None:
                req.logprobs.append(lp)
                if lp and token_id in lp:
                    req.cumulative_logprob += lp[token_id].logprob
            if req.guided_state is not None:
                req.guided_state.advance(token_id)
            new_text = self.detokenizer.append_token(req, token_id)
            self.stop_checker.check(req, token_id, new_text)

        outputs: list[RequestOutput] = []
        seen = set()
        for it in sched.items:
            req = it.request
            if id(req) in seen:
                continue
            seen.add(id(req))
            out = req.make_output()
            if out is not None:
                outputs.append(out)
            if req.status.is_finished:
                self._account_finish(req)
        if pt is not None:
            t3 = time.perf_counter()
            pt["postprocess"] += t3 - t2
            pt["steps"] += 1
            pt["tokens"] += len(sampler_out.token_ids)
            pt["prefill_steps"] += 1 if sched.items and any(
                i.num_new_tokens > 1 for i in sched.items) else 0
            if self._timing_autoprint and pt["steps"] >= 128:
                import sys as _sys

                n = pt.pop("steps")
                toks = pt.pop("tokens")
                pf = pt.pop("prefill_steps")
                parts = {k: round(v / n * 1e3, 2) for k, v in pt.items()}
                print(f"[step-timing] {n} steps ({pf} with prefill, "
                      f"{toks} tokens): per-step ms {parts}",
                      file=_sys.stderr, flush=True)
                pt.clear()
        if getattr(self, "_bench_win", None) is not None:
            self._bench_tick(len(sampler_out.token_ids) + spec_token_count)
        self.metrics.num_running.set(len(self.scheduler.running))
        self.metrics.num_waiting.set(len(self.scheduler.waiting))
        self.metrics.kv_usage.set(
            1.0 - self.block_manager.num_free_blocks / max(1, self.block_manager.num_blocks)
        )
        return outputs

    def shutdown(self) -> None:
        self.worker.stop_workers()
