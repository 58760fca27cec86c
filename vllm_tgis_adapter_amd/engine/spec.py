"""Speculative decoding, n-gram / prompt-lookup flavor.

No draft model: the proposer finds the most recent earlier occurrence of the
sequence's last bigram and drafts the tokens that followed it — effective on
text with repetition (code, JSON, retrieval contexts).  Draft tokens ride
the engine's existing multi-token chunk path (they are scheduled exactly
like a prefill chunk, so the causal prefill attention kernel verifies them
in ONE forward); the worker argmaxes the chunk's logits rows and the engine
accepts the longest matching prefix plus the model's bonus token, rolling
``num_computed_tokens`` back past any rejected draft KV (those cache slots
are overwritten when the real tokens reach them).

Enabled per-engine with ``--speculator-name ngram`` (reference arg surface
tgis_utils/args.py:165-168 → speculative_model); greedy, unconstrained
requests only — everything else falls back to normal decode transparently.
"""

from __future__ import annotations

from .request import Request

NGRAM = 2  # lookup key length


def is_ngram_spec(model_name: str | None) -> bool:
    return model_name is not None and model_name.strip("[]").lower() == "ngram"


def eligible(req: Request) -> bool:
    p = req.sampling_params
    return (
        p.temperature == 0.0
        and not (p.min_tokens and req.num_output_tokens < p.min_tokens)
        and p.logprobs is None
        and p.prompt_logprobs is None
        and not p.logits_processors
        and req.guided_state is None
        and req.num_tokens - req.num_computed_tokens == 1  # decode-ready
    )


def propose(req: Request, max_draft: int, max_model_len: int) -> list[int]:
    """Draft up to ``max_draft`` tokens by prompt lookup of the last bigram."""
    toks = req.all_token_ids
    n = len(toks)
    if n < NGRAM + 1:
        return []
    budget = min(max_draft, max_model_len - n - 1)
    if budget <= 0:
        return []
    key = tuple(toks[n - NGRAM:])
    # most recent earlier occurrence of the bigram
    for i in range(n - NGRAM - 1, -1, -1):
        if tuple(toks[i:i + NGRAM]) == key:
            draft = toks[i + NGRAM:i + NGRAM + budget]
            return list(draft)
    return []
