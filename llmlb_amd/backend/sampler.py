"""Token sampling: greedy, temperature, top-p/top-k/min-p, OpenAI
frequency/presence penalties and HF-style repetition penalty.  The
all-greedy fast path (the serving common case) is one argmax with no
host round trip; per-sequence filters run only for sequences that ask
for them."""
from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class SamplingParams:
    temperature: float = 0.0     # 0 = greedy
    top_p: float = 1.0
    top_k: int = 0               # 0 = off
    min_p: float = 0.0           # 0 = off
    frequency_penalty: float = 0.0   # OpenAI: count-scaled
    presence_penalty: float = 0.0    # OpenAI: flat per seen token
    repetition_penalty: float = 1.0  # HF: divide positive / mult negative
    max_new_tokens: int = 128
    stop_token_ids: tuple = ()
    seed: int | None = None
    logprobs: int = 0            # 0 = off, else top-k alternatives
    logit_bias: tuple = ()       # ((token_id, bias), ...) -100..100

    def needs_history(self) -> bool:
        return (self.frequency_penalty != 0.0
                or self.presence_penalty != 0.0
                or self.repetition_penalty != 1.0)

    def needs_slow_path(self) -> bool:
        return self.needs_history() or bool(self.logit_bias)


def _apply_penalties(lg: torch.Tensor, p: SamplingParams,
                     out_ids: list[int], prompt_ids: list[int]):
    """In place on fp32 logits for one sequence."""
    if p.frequency_penalty or p.presence_penalty:
        if out_ids:
            seen = torch.tensor(out_ids, device=lg.device)
            uniq, counts = seen.unique(return_counts=True)
            lg[uniq] -= (p.frequency_penalty * counts.float()
                         + p.presence_penalty)
    if p.repetition_penalty != 1.0:
        ids = list(prompt_ids) + list(out_ids)
        if ids:
            uniq = torch.tensor(ids, device=lg.device).unique()
            vals = lg[uniq]
            lg[uniq] = torch.where(vals > 0, vals / p.repetition_penalty,
                                   vals * p.repetition_penalty)


def sample(logits: torch.Tensor, params_list: list[SamplingParams],
           generators: list[torch.Generator | None],
           histories: list[tuple[list[int], list[int]]] | None = None
           ) -> torch.Tensor:
    """logits: [S, vocab] -> [S] int64 next tokens.

    ``histories[i]`` = (prompt_ids, out_ids) — only consulted for
    sequences whose params request a penalty."""
    S = logits.shape[0]
    if all(p.temperature < 1e-5 and not p.needs_slow_path()
           for p in params_list):
        return logits.argmax(dim=-1)  # all-greedy fast path (no H2D)
    out = torch.empty(S, dtype=torch.long, device=logits.device)
    for i in range(S):
        p = params_list[i]
        if p.temperature < 1e-5 and not p.needs_slow_path():
            out[i] = logits[i].argmax(dim=-1)
            continue
        # clone: .float() aliases when logits are already fp32 and the
        # penalty filters mutate in place
        lg = logits[i].float().clone()
        if p.needs_history() and histories is not None:
            prompt_ids, out_ids = histories[i]
            _apply_penalties(lg, p, out_ids, prompt_ids)
        for tok_id, bias in p.logit_bias:
            if 0 <= tok_id < lg.shape[-1]:
                lg[tok_id] += bias
        if p.temperature < 1e-5:
            out[i] = lg.argmax(dim=-1)
            continue
        lg = lg / max(p.temperature, 1e-5)
        if p.top_k > 0 and p.top_k < lg.shape[-1]:
            kth = lg.topk(p.top_k).values[-1]
            lg[lg < kth] = float("-inf")
        probs = torch.softmax(lg, dim=-1)
        if p.min_p > 0.0:
            probs[probs < p.min_p * probs.max()] = 0.0
            probs /= probs.sum()
        if p.top_p < 1.0:
            sorted_p, idx = probs.sort(descending=True)
            csum = sorted_p.cumsum(-1)
            mask = csum - sorted_p > p.top_p
            sorted_p[mask] = 0.0
            sorted_p /= sorted_p.sum()
            pick = torch.multinomial(sorted_p, 1,
                                     generator=generators[i])
            out[i] = idx[pick]
        else:
            out[i] = torch.multinomial(probs, 1,
                                       generator=generators[i])
    return out
