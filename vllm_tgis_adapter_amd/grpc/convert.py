"""Engine-result -> TGIS proto conversion.

Wire BEHAVIOR here is pinned to the reference service (stop-reason mapping
grpc_server.py:663-699, token detail / logprob / rank / top-N layout
:701-756, tokenize-and-cap semantics :758-800) and locked in by the parity
tests; the implementation is this framework's own — table-driven reason
resolution and batch-first detail assembly rather than the reference's
branch chains.
"""

from __future__ import annotations

from typing import TYPE_CHECKING, Optional

from ..logging import init_logger
from . import proto

if TYPE_CHECKING:
    from ..engine.types import CompletionOutput, PosLogprobs

logger = init_logger(__name__)

_SR = proto.StopReasonValue


class StopInfo:
    """Resolved wire stop reason + the sequence text to echo, if any."""

    __slots__ = ("reason", "sequence")

    def __init__(self, reason: int, sequence: str = ""):
        self.reason = reason
        self.sequence = sequence


def resolve_stop(
    output: "CompletionOutput", *, capped: bool, deadline_hit: bool, tokenizer
) -> StopInfo:
    """Map the engine finish state onto the TGIS StopReason enum.

    ``capped`` marks requests whose max_tokens came from the model-length
    cap rather than the client (length-finish then reports TOKEN_LIMIT).
    """
    finish = output.finish_reason
    if finish is None:
        return StopInfo(_SR.TIME_LIMIT if deadline_hit else _SR.NOT_FINISHED)
    if finish == "length":
        return StopInfo(_SR.TOKEN_LIMIT if capped else _SR.MAX_TOKENS)
    if finish == "abort":
        return StopInfo(_SR.TIME_LIMIT if deadline_hit else _SR.CANCELLED)
    if finish != "stop":
        logger.warning("Unrecognized finish_reason: %s", finish)
        return StopInfo(_SR.CANCELLED)

    # "stop": a stop string, a stop/eos token id, or the default eos
    detail = output.stop_reason
    if isinstance(detail, str):
        return StopInfo(_SR.STOP_SEQUENCE, detail)
    if isinstance(detail, int):
        return StopInfo(_SR.EOS_TOKEN, tokenizer.convert_ids_to_tokens(detail))
    if detail is None:
        return StopInfo(_SR.EOS_TOKEN, getattr(tokenizer, "eos_token", None) or "")
    logger.warning("Unexpected stop_reason type: %s", type(detail))
    return StopInfo(_SR.STOP_SEQUENCE)


def append_token_details(
    dest,
    ids: list[int],
    per_pos_logprobs,
    *,
    want_logprob: bool,
    want_rank: bool,
    top_n: int,
    tokenizer,
    skip_first: int = 0,
) -> None:
    """Fill a repeated TokenInfo field for the given token positions.

    Layout per position: token text; optionally the token's logprob and/or
    rank (a spec-decode dummy rank of -1 is published as 0); optionally the
    ``top_n`` highest-probability alternatives, each with text (+ logprob
    when logprobs were requested).
    """
    if skip_first:
        ids = ids[skip_first:]
        if per_pos_logprobs is not None:
            per_pos_logprobs = per_pos_logprobs[skip_first:]

    texts = tokenizer.convert_ids_to_tokens(ids)
    for pos, text in enumerate(texts):
        info = proto.TokenInfo(text=text)
        lp_map: "PosLogprobs | None" = (
            per_pos_logprobs[pos] if per_pos_logprobs else None
        )
        if lp_map is not None:
            _fill_position(info, ids[pos], lp_map, want_logprob, want_rank,
                           top_n, tokenizer)
        dest.append(info)


def requested_logprob_count(resp_options, greedy: bool) -> Optional[int]:
    """How many logprobs/position the engine must track for this request.

    The count covers the requested top-N alternatives plus room for the
    sampled token itself; under greedy decoding with logprobs the sampled
    token IS the top-1, so one slot overlaps.  None = no logprobs needed.
    """
    base = 1 if (resp_options.token_logprobs or resp_options.token_ranks) else 0
    n = resp_options.top_n_tokens
    if n:
        base += n
        if greedy and resp_options.token_logprobs:
            base -= 1
    return base or None


def tokenize_with_caps(
    prompt: str,
    sampling_params,
    *,
    tokenizer,
    add_special_tokens: bool,
    truncate_to: Optional[int],
    max_model_len: int,
    default_max_new: int,
    validate_input_fn,
) -> tuple[list[int], bool]:
    """Encode the prompt and settle the request's max_tokens budget.

    Returns (input token ids, capped) where ``capped`` records that the
    effective max_tokens came from the model context limit rather than the
    client — downstream a length-finish then reports TOKEN_LIMIT instead
    of MAX_TOKENS (reference behavior grpc_server.py:787-798).
    Raises ValueError for over-long input (TGIS error strings).
    """
    enc_kwargs: dict = {"add_special_tokens": add_special_tokens}
    if truncate_to is not None:
        enc_kwargs["truncation"] = True
        enc_kwargs["max_length"] = truncate_to
    ids = tokenizer(prompt, **enc_kwargs).input_ids
    n_in = len(ids)

    validate_input_fn(sampling_params, n_in, max_model_len)

    room = max_model_len - n_in
    requested = sampling_params.max_tokens
    if requested is None:
        sampling_params.max_tokens = min(default_max_new, room)
        return ids, True
    if requested > room:
        sampling_params.max_tokens = room
        return ids, True
    return ids, False


def _fill_position(info, token_id, lp_map, want_logprob, want_rank, top_n,
                   tokenizer) -> None:
    entry = lp_map.get(token_id)
    if entry is not None and (want_logprob or want_rank):
        if want_logprob:
            info.logprob = entry.logprob
        if want_rank:
            info.rank = max(0, entry.rank if entry.rank is not None else 0)
    if not top_n:
        return
    best = sorted(lp_map.items(), key=lambda kv: kv[1].logprob, reverse=True)
    best = best[:top_n]
    alt_texts = tokenizer.convert_ids_to_tokens([tid for tid, _ in best])
    for alt_text, (_, alt) in zip(alt_texts, best):
        top = info.top_tokens.add()
        top.text = alt_text
        if want_logprob:
            top.logprob = alt.logprob
