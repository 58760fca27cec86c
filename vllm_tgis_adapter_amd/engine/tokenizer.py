"""Tokenizer facade.

HF-compatible tokenizer behind the engine (reference consumes vLLM's
AnyTokenizer; call sites grpc_server.py:770-779,839-874,685-688 need
encode/encode_plus with add_special_tokens + offsets, convert_ids_to_tokens
and eos handling).  For synthetic presets (no network) a byte-level tokenizer
with the preset's vocab size is built programmatically.
"""

from __future__ import annotations

import os
from functools import lru_cache

from .config import ModelConfig


def build_synthetic_tokenizer(vocab_size: int):
    """Byte-level tokenizer with exactly ``vocab_size`` entries.

    BPE with an empty merge list over the ByteLevel alphabet: every
    pre-token decomposes into single-byte tokens, so any input encodes; the
    remaining id space is filled so convert_ids_to_tokens works for any
    sampled id.
    """
    from tokenizers import Tokenizer, decoders, pre_tokenizers, processors
    from tokenizers.models import BPE
    from transformers import PreTrainedTokenizerFast

    specials = ["<unk>", "<s>", "</s>", "<pad>"]
    alphabet = sorted(pre_tokenizers.ByteLevel.alphabet())
    vocab: dict[str, int] = {}
    for tok in specials:
        vocab[tok] = len(vocab)
    for ch in alphabet:
        vocab[ch] = len(vocab)
    i = 0
    while len(vocab) < vocab_size:
        filler = f"<extra_{i}>"
        if filler not in vocab:
            vocab[filler] = len(vocab)
        i += 1

    tok = Tokenizer(BPE(vocab=vocab, merges=[], unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    tok.post_processor = processors.TemplateProcessing(
        single="<s> $A",
        pair="<s> $A $B",
        special_tokens=[("<s>", vocab["<s>"])],
    )
    return PreTrainedTokenizerFast(
        tokenizer_object=tok,
        bos_token="<s>",
        eos_token="</s>",
        unk_token="<unk>",
        pad_token="<pad>",
        clean_up_tokenization_spaces=False,
    )


@lru_cache(maxsize=8)
def _cached_synthetic(vocab_size: int):
    return build_synthetic_tokenizer(vocab_size)


def get_tokenizer(model_config: ModelConfig):
    path = model_config.weights_path or model_config.model
    if os.path.isdir(path) and (
        os.path.exists(os.path.join(path, "tokenizer.json"))
        or os.path.exists(os.path.join(path, "tokenizer_config.json"))
    ):
        from transformers import AutoTokenizer

        return AutoTokenizer.from_pretrained(path)
    return _cached_synthetic(model_config.vocab_size)
