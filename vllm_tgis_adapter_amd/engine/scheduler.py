"""Continuous-batching scheduler.

Iteration-level scheduling with chunked prefill: every step assembles one
token batch of at most ``max_num_batched_tokens`` tokens across at most
``max_num_seqs`` requests.  Running requests are served first (decode = 1
token each, or the next prompt chunk while prefilling); waiting requests are
admitted with whatever budget remains.  If KV blocks run out mid-decode the
most-recently-arrived running request is preempted (blocks freed, recomputed
on re-admission).  Abort frees KV blocks mid-flight (reference call site
grpc_server.py:292,388).

Token-accounting invariant: ``num_computed_tokens`` counts tokens whose KV is
in cache.  A request *samples* in the step where its computed count reaches
``num_tokens`` (= prompt + generated so far): the model's logits at the last
computed position produce the next token.  Steady-state decode is the 1-token
special case; preemption recovery (recompute prompt+generated) is the same
code path.
"""

from __future__ import annotations

import time
from collections import deque
from dataclasses import dataclass, field
from typing import Optional

from .block_manager import BlockManager
from .config import SchedulerConfig
from .request import Request, RequestStatus


@dataclass
class ScheduledItem:
    request: Request
    num_new_tokens: int  # tokens whose KV is computed this step
    samples: bool        # True if this step produces a sampled token


@dataclass
class SchedulerOutput:
    items: list[ScheduledItem] = field(default_factory=list)
    preempted: list[Request] = field(default_factory=list)

    @property
    def total_tokens(self) -> int:
        return sum(i.num_new_tokens for i in self.items)

    @property
    def is_empty(self) -> bool:
        return not self.items


class Scheduler:
    def __init__(self, config: SchedulerConfig, block_manager: BlockManager):
        self.config = config
        self.block_manager = block_manager
        self.waiting: deque[Request] = deque()
        self.running: list[Request] = []
        self._requests: dict[str, Request] = {}

    # ------------------------------------------------------------------
    def add_request(self, request: Request) -> None:
        self._requests[request.request_id] = request
        self.waiting.append(request)

    def abort_request(self, request_id: str) -> Optional[Request]:
        req = self._requests.get(request_id)
        if req is None or req.status.is_finished:
            return None
        req.finish(RequestStatus.FINISHED_ABORTED)
        if req in self.running:
            self.running.remove(req)
        else:
            try:
                self.waiting.remove(req)
            except ValueError:
                pass
        self.block_manager.free(req)
        self._requests.pop(request_id, None)
        return req

    def get_request(self, request_id: str) -> Optional[Request]:
        return self._requests.get(request_id)

    def has_unfinished(self) -> bool:
        return bool(self.waiting or self.running)

    def finish_request(self, req: Request) -> None:
        """Called by the engine after stop-condition checks."""
        if req in self.running:
            self.running.remove(req)
        self.block_manager.free(req)
        self._requests.pop(req.request_id, None)

    # ------------------------------------------------------------------
    def _schedule_chunk(self, req: Request, budget: int) -> tuple[int, bool]:
        """(chunk, samples) under the token budget; 0 chunk => skip."""
        remaining = req.num_tokens - req.num_computed_tokens + len(req.spec_draft)
        chunk = min(remaining, budget)
        samples = chunk == remaining
        if req.spec_draft and chunk < remaining:
            # draft didn't fit the budget; drop it for this step
            req.spec_draft = []
            remaining = req.num_tokens - req.num_computed_tokens
            chunk = min(remaining, budget)
            samples = chunk == remaining
        return chunk, samples

    def schedule(self) -> SchedulerOutput:
        out = SchedulerOutput()
        budget = self.config.max_num_batched_tokens

        # 1. Serve running requests, oldest first; preempt newest on KV OOM.
        for req in list(self.running):
            if budget <= 0:
                break
            if req not in self.running:  # got preempted by an earlier victim pick
                continue
            chunk, samples = self._schedule_chunk(req, budget)
            if chunk <= 0:
                continue
            target_tokens = req.num_computed_tokens + chunk
            ok = True
            while not self.block_manager.can_grow_to(req, target_tokens):
                victim = self._pick_victim(exclude=req)
                if victim is None:
                    ok = False
                    break
                self._preempt(victim, out)
                out.items = [i for i in out.items if i.request is not victim]
            if not ok:
                continue
            self.block_manager.grow_to(req, target_tokens)
            out.items.append(ScheduledItem(req, chunk, samples))
            budget -= chunk

        # 2. Admit waiting requests with the remaining budget.
        #
        # Prefill-admission batching: with a busy decode batch, a steady
        # trickle of arrivals would otherwise make nearly EVERY step a mixed
        # prefill+decode step — which cannot replay the hipGraph decode
        # buckets and runs eager (~2-3x slower per step under sustained gRPC
        # load, tools/serve_bench.py r1).  Hold arrivals briefly so prefills
        # amortize into fewer mixed steps; an idle engine admits immediately.
        if (
            self.waiting
            and out.items
            and len(self.waiting) < self.config.prefill_admit_batch
        ):
            oldest = self.waiting[0]
            age = time.time() - (oldest.metrics.arrival_time or 0.0)
            if age < self.config.prefill_admit_delay_s:
                return out
        while self.waiting and budget > 0 and len(self.running) < self.config.max_num_seqs:
            req = self.waiting[0]
            if req.num_computed_tokens == 0 and not req.block_ids:
                self.block_manager.match_prefix(req)
            chunk, samples = self._schedule_chunk(req, budget)
            if chunk <= 0:
                break
            target_tokens = req.num_computed_tokens + chunk
            if not self.block_manager.can_grow_to(req, target_tokens):
                # Don't preempt running work to admit new work.
                break
            self.waiting.popleft()
            self.block_manager.grow_to(req, target_tokens)
            now = time.time()
            if req.metrics.first_scheduled_time is None:
                req.metrics.first_scheduled_time = now
                req.metrics.time_in_queue = now - req.metrics.arrival_time
            req.status = RequestStatus.RUNNING
            self.running.append(req)
            out.items.append(ScheduledItem(req, chunk, samples))
            budget -= chunk

        return out

    # ------------------------------------------------------------------
    def _pick_victim(self, exclude: Request) -> Optional[Request]:
        for req in reversed(self.running):
            if req is not exclude:
                return req
        return None

    def _preempt(self, req: Request, out: SchedulerOutput) -> None:
        self.running.remove(req)
        self.block_manager.free(req)
        # Recompute-from-scratch preemption: KV is rebuilt on re-admission
        # (already-generated tokens are recomputed like prompt tokens).
        req.num_computed_tokens = 0
        req.spec_draft = []
        req.draft_computed = 0  # draft-model KV freed with the blocks
        req.status = RequestStatus.PREEMPTED
        self.waiting.appendleft(req)
        out.preempted.append(req)
