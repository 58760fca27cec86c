"""Per-request logits processors.

Both are implemented natively (the reference wraps HF's TypicalLogitsWarper;
reference tgis_utils/logits_processors.py:7-47 defines the semantics):

* typical_p — locally-typical sampling (Meister et al.): keep the smallest
  set of tokens whose |-log p - H| deviation is lowest and whose mass
  reaches ``mass``; others are masked to -inf.
* exponential-decay length penalty — after ``start_index`` generated tokens,
  boost the EOS logit by |eos| * (decay^tokens_past - 1).
"""

from __future__ import annotations

import torch


class TypicalLogitsWarperWrapper:
    def __init__(self, mass: float):
        self.mass = mass

    def __call__(self, token_ids: list[int], logits: torch.Tensor) -> torch.Tensor:
        # operates on a single [vocab] row
        log_probs = torch.log_softmax(logits.float(), dim=-1)
        probs = log_probs.exp()
        entropy = -(probs * log_probs).nansum()
        deviation = (entropy + log_probs).abs()  # |-log p - H|
        order = torch.argsort(deviation)
        cum_mass = probs[order].cumsum(dim=-1)
        # keep tokens until the cumulative mass exceeds `mass` (inclusive)
        keep_count = int((cum_mass < self.mass).sum().item()) + 1
        keep = order[:keep_count]
        out = torch.full_like(logits, float("-inf"))
        out[keep] = logits[keep]
        return out


class ExpDecayLengthPenaltyWarper:
    def __init__(self, length_penalty: tuple[int, float], eos_token_id: int):
        self.start, self.penalty = length_penalty
        self.eos_token_id = eos_token_id

    def __call__(self, token_ids: list[int], logits: torch.Tensor) -> torch.Tensor:
        tokens_past = max(0, len(token_ids) - self.start)
        factor = pow(self.penalty, tokens_past)
        if factor != 1:
            eos_logit = logits[self.eos_token_id]
            # penalty applied on |logit| so negative logits move up too
            logits[self.eos_token_id] = eos_logit + torch.abs(eos_logit) * (factor - 1)
        return logits
