"""Sampling parameters (the local equivalent of the OpenAI sampling knobs the
reference forwards: temperature, top_p, max_tokens, stop, seed, n —
k_llms/resources/completions/completions.py:42-64)."""

from __future__ import annotations

from typing import Dict, List, Optional, Union

from pydantic import BaseModel


class SamplingParams(BaseModel):
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = 0  # 0 = disabled
    max_tokens: Optional[int] = None
    stop: Optional[Union[str, List[str]]] = None
    seed: Optional[int] = None
    frequency_penalty: float = 0.0
    presence_penalty: float = 0.0
    # OpenAI `logit_bias`: {token_id: bias in [-100, 100]} added to the
    # logits before sampling (the reference forwards it to the remote API;
    # served natively here)
    logit_bias: Optional[Dict[int, float]] = None
    logprobs: bool = False
    # number of top alternative tokens to report per position (OpenAI
    # `top_logprobs`, 0 = off; requires logprobs=True at the API layer)
    top_logprobs: int = 0

    @property
    def stop_list(self) -> List[str]:
        if self.stop is None:
            return []
        return [self.stop] if isinstance(self.stop, str) else list(self.stop)
