"""Token sampling: greedy, temperature, top-p (batched, on device)."""
from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class SamplingParams:
    temperature: float = 0.0     # 0 = greedy
    top_p: float = 1.0
    max_new_tokens: int = 128
    stop_token_ids: tuple = ()
    seed: int | None = None


def sample(logits: torch.Tensor, params_list: list[SamplingParams],
           generators: list[torch.Generator | None]) -> torch.Tensor:
    """logits: [S, vocab] -> [S] int64 next tokens."""
    S = logits.shape[0]
    if all(p.temperature < 1e-5 for p in params_list):
        return logits.argmax(dim=-1)  # all-greedy fast path (no H2D)
    temps = torch.tensor(
        [max(p.temperature, 0.0) for p in params_list],
        device=logits.device, dtype=torch.float32)
    greedy = temps < 1e-5
    out = torch.empty(S, dtype=torch.long, device=logits.device)
    if greedy.any():
        out[greedy] = logits[greedy].argmax(dim=-1)
    if (~greedy).any():
        for i in range(S):
            if greedy[i]:
                continue
            p = params_list[i]
            lg = (logits[i].float() / max(p.temperature, 1e-5))
            probs = torch.softmax(lg, dim=-1)
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
