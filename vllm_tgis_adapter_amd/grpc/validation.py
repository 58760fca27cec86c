"""TGIS-compatible parameter/input validation.

The error STRINGS are wire contract — TGIS clients match on them — and must
stay byte-identical to the reference enumeration (reference
grpc/validation.py:18-61, itself mirroring TGIS router/src/validation.rs).
The checks reimplement the documented TGIS limits (reference
validation.py:64-144).
"""

from __future__ import annotations

import typing
from enum import Enum

from . import proto

MAX_TOP_N_TOKENS = 10
MAX_STOP_SEQS = 6
MAX_STOP_SEQ_LENGTH = 240

# Reject (True) vs silently ignore (False) sampling parameters in greedy mode.
STRICT_PARAMETER_VALIDATION = False


class TGISValidationError(str, Enum):
    """TGIS parameter-validation failure strings (wire contract)."""

    TopP = "top_p must be > 0.0 and <= 1.0"
    TopK = "top_k must be strictly positive"
    TypicalP = "typical_p must be <= 1.0"
    RepetitionPenalty = "repetition_penalty must be > 0.0 and <= 2.0"
    LengthPenalty = "length_penalty.decay_factor must be >= 1.0 and <= 10.0"
    MaxNewTokens = "max_new_tokens must be <= {0}"
    MinNewTokens = "min_new_tokens must be <= max_new_tokens"
    InputLength = (
        "input tokens ({0}) plus prefix length ({1}) plus "
        "min_new_tokens ({2}) must be <= {3}"
    )
    InputLength2 = "input tokens ({0}) plus prefix length ({1}) must be < {2}"
    Tokenizer = "tokenizer error {0}"
    StopSequences = (
        "can specify at most {0} non-empty stop sequences, each "
        "not more than {1} UTF8 bytes"
    )
    TokenDetail = (
        "must request input and/or generated tokens to request extra token detail"
    )
    PromptPrefix = "can't retrieve prompt prefix with id '{0}': {1}"
    SampleParametersGreedy = (
        "sampling parameters aren't applicable in greedy decoding mode"
    )
    TopN = "top_n_tokens ({0}) must be <= {1}"
    AdapterNotFound = "can't retrieve adapter with id '{0}': {1}"
    AdaptersDisabled = "adapter_id supplied but no adapter store was configured"
    AdapterUnsupported = "adapter type {0} is not currently supported"
    InvalidAdapterID = (
        "Invalid adapter id '{0}', must contain only alphanumeric, _ and - and /"
    )

    def error(self, *args, **kwargs) -> typing.NoReturn:
        raise ValueError(self.value.format(*args, **kwargs))


def validate_input(sampling_params, token_num: int, max_model_len: int) -> None:
    """Reject prompts that cannot fit the model context."""
    n_min = sampling_params.min_tokens
    if token_num >= max_model_len:
        TGISValidationError.InputLength2.error(token_num, 0, max_model_len)
    if token_num + n_min > max_model_len:
        TGISValidationError.InputLength.error(token_num, 0, n_min, max_model_len)


def validate_params(params, server_cap: int) -> None:
    """Raise ValueError carrying the exact TGIS string when any field of a
    proto Parameters violates the documented TGIS limits.  Check ORDER is
    part of observable behavior (the first violated limit names the error)
    and follows the reference."""
    err = TGISValidationError

    if params.decoding.HasField("length_penalty"):
        decay = params.decoding.length_penalty.decay_factor
        if decay < 1.0 or decay > 10.0:
            err.LengthPenalty.error()

    rep = params.decoding.repetition_penalty  # 0 == unset == no penalty
    if rep < 0 or rep > 2:
        err.RepetitionPenalty.error()

    stopping = params.stopping
    if stopping.max_new_tokens > server_cap:
        err.MaxNewTokens.error(server_cap)
    if stopping.min_new_tokens > (stopping.max_new_tokens or server_cap):
        err.MinNewTokens.error()

    seqs = list(stopping.stop_sequences)
    ok_seqs = len(seqs) <= MAX_STOP_SEQS and all(
        0 < len(s) <= MAX_STOP_SEQ_LENGTH for s in seqs
    )
    if not ok_seqs:
        err.StopSequences.error(MAX_STOP_SEQS, MAX_STOP_SEQ_LENGTH)

    resp = params.response
    if resp.top_n_tokens > MAX_TOP_N_TOKENS:
        err.TopN.error(resp.top_n_tokens, MAX_TOP_N_TOKENS)
    detail = resp.token_logprobs or resp.token_ranks or resp.top_n_tokens
    if detail and not (resp.input_tokens or resp.generated_tokens):
        err.TokenDetail.error()

    s = params.sampling
    if STRICT_PARAMETER_VALIDATION and params.method == proto.GREEDY and (
        s.temperature or s.top_k or s.top_p or s.typical_p
    ):
        err.SampleParametersGreedy.error()
    if s.top_k < 0:
        err.TopK.error()
    if s.top_p < 0 or s.top_p > 1:
        err.TopP.error()
    if s.typical_p > 1:
        err.TypicalP.error()
