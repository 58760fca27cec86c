"""Incremental detokenization + stop-sequence scanning.

Streamed text must be decoded token-by-token without re-decoding the whole
output, while never emitting the replacement char of a half-finished UTF-8 /
byte-level merge, and while holding back text that could be the prefix of a
stop sequence (TGIS contract: ≤6 stop seqs of ≤240 bytes; validation.py:10-11).
"""

from __future__ import annotations

from typing import Optional

from .request import Request, RequestStatus


class Detokenizer:
    def __init__(self, tokenizer):
        self.tokenizer = tokenizer
        self._special = set(tokenizer.all_special_tokens)
        # id -> (token, token-or-"" when skipping specials); lazily filled
        self._tok_cache: dict[int, tuple[str, str]] = {}
        # id -> decoded piece (skip-specials variant); lazily filled
        self._piece_cache: dict[int, tuple[str, str]] = {}
        self._ctx_free = self._probe_context_free()

    def _probe_context_free(self) -> bool:
        """True when per-token decode concatenates to the joint decode, so
        the steady-state hot path can append cached pieces (one dict lookup)
        instead of two windowed convert_tokens_to_string calls per token."""
        try:
            n = min(len(self.tokenizer), 4096)
            ids = [i for i in range(2, n, max(1, n // 64))][:48]
            toks = self.tokenizer.convert_ids_to_tokens(ids)
            pieces = [self.tokenizer.convert_tokens_to_string([t]) for t in toks]
            for i in range(0, len(ids) - 1, 2):
                joint = self.tokenizer.convert_tokens_to_string(
                    [toks[i], toks[i + 1]])
                if joint != pieces[i] + pieces[i + 1]:
                    return False
            return True
        except Exception:
            return False

    def _piece(self, token_id: int) -> tuple[str, str]:
        hit = self._piece_cache.get(token_id)
        if hit is None:
            tok = self.tokenizer.convert_ids_to_tokens([token_id])[0]
            s = self.tokenizer.convert_tokens_to_string([tok])
            hit = (s, "" if tok in self._special else s)
            self._piece_cache[token_id] = hit
        return hit

    def _convert(self, ids: list[int], skip_special: bool) -> list[str]:
        toks = self.tokenizer.convert_ids_to_tokens(ids)
        if skip_special:
            return [t if t not in self._special else "" for t in toks]
        return toks

    def _one_token(self, token_id: int, skip_special: bool) -> str:
        hit = self._tok_cache.get(token_id)
        if hit is None:
            tok = self.tokenizer.convert_ids_to_tokens([token_id])[0]
            hit = (tok, "" if tok in self._special else tok)
            self._tok_cache[token_id] = hit
        return hit[1] if skip_special else hit[0]

    def append_token(self, req: Request, token_id: int) -> str:
        """Incrementally decode one new token; returns the new text fragment."""
        skip = req.sampling_params.skip_special_tokens

        # fast path: context-free decoder, no pending partial merge -> the
        # cached piece IS the new text (the windowed double-decode below
        # cost ~2 Rust calls per token per request, a visible slice of the
        # per-step postprocess at batch 512)
        if self._ctx_free and req.read_offset == len(req.prev_token_texts):
            piece = self._piece(token_id)[1 if skip else 0]
            if "�" not in piece:
                req.prev_token_texts.append(self._one_token(token_id, skip))
                req.prefix_offset = req.read_offset
                req.read_offset = len(req.prev_token_texts)
                if piece:
                    req.output_text += piece
                return piece

        new_tok = self._one_token(token_id, skip)
        req.prev_token_texts.append(new_tok)

        toks = req.prev_token_texts
        if req.read_offset == 0 and req.prefix_offset == 0 and len(toks) > 1:
            pass  # state already consistent

        prefix_text = self.tokenizer.convert_tokens_to_string(
            [t for t in toks[req.prefix_offset:req.read_offset] if t]
        )
        full_text = self.tokenizer.convert_tokens_to_string(
            [t for t in toks[req.prefix_offset:] if t]
        )
        if len(full_text) > len(prefix_text) and not full_text.endswith("�"):
            new_text = full_text[len(prefix_text):]
            req.prefix_offset = req.read_offset
            req.read_offset = len(toks)
            req.output_text += new_text
            return new_text
        return ""


class StopChecker:
    """Applies max/min token limits, EOS and stop-string semantics."""

    def __init__(self, max_model_len: int):
        self.max_model_len = max_model_len

    def check(self, req: Request, token_id: int, new_text: str) -> None:
        if self.check_cheap(req, token_id):
            return
        self.check_text(req, new_text)

    def check_cheap(self, req: Request, token_id: int) -> bool:
        """Detokenization-free finish conditions (EOS; length limits for
        requests without stop strings — those keep reference precedence by
        deciding after the string scan).  Safe to run before detok, so the
        pipelined step applies it at token-append time while stop-string
        scanning rides the deferred phase."""
        p = req.sampling_params

        # EOS (suppressed below min_tokens by the sampler; double-check here)
        if (
            req.eos_token_id is not None
            and token_id == req.eos_token_id
            and req.num_output_tokens >= p.min_tokens
        ):
            # eos token excluded from output text by skip_special_tokens
            req.finish(RequestStatus.FINISHED_STOPPED, stop_reason=None)
            return True
        if not p.stop:
            if p.max_tokens is not None and req.num_output_tokens >= p.max_tokens:
                req.finish(RequestStatus.FINISHED_LENGTH)
                return True
            if req.num_tokens >= self.max_model_len:
                req.finish(RequestStatus.FINISHED_LENGTH)
                return True
        return False

    def check_text(self, req: Request, new_text: str) -> None:
        """Stop-string scan (+ length limits for stop-string requests)."""
        p = req.sampling_params

        # stop strings
        if p.stop and req.num_output_tokens >= p.min_tokens:
            window_start = max(0, len(req.output_text) - len(new_text) - 240)
            window = req.output_text[window_start:]
            best: Optional[tuple[int, str]] = None
            for s in p.stop:
                idx = window.find(s)
                if idx != -1 and (best is None or idx < best[0]):
                    best = (idx, s)
            if best is not None:
                idx, s = best
                end = window_start + idx + (len(s) if p.include_stop_str_in_output else 0)
                req.output_text = req.output_text[:end]
                req.holdback_len = 0
                req.finish(RequestStatus.FINISHED_STOPPED, stop_reason=s)
                return
            # hold back a possible stop-string prefix from streaming
            req.holdback_len = _longest_stop_prefix(req.output_text, p.stop)

        # length limits
        if p.max_tokens is not None and req.num_output_tokens >= p.max_tokens:
            req.finish(RequestStatus.FINISHED_LENGTH)
            return
        if req.num_tokens >= self.max_model_len:
            req.finish(RequestStatus.FINISHED_LENGTH)
            return


def _longest_stop_prefix(text: str, stops: list[str]) -> int:
    """Length of the longest proper prefix of any stop string that is a
    suffix of ``text`` (the bytes we must not stream yet)."""
    best = 0
    for s in stops:
        for plen in range(min(len(s) - 1, len(text)), 0, -1):
            if text.endswith(s[:plen]):
                best = max(best, plen)
                break
    return best
