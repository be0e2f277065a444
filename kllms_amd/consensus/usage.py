"""Usage summation across n completion streams.

Equivalent of the reference's consolidate_consensus_usage
(ref consensus_utils.py:1458-1516) as a plain utility with no external
dependency: sums CompletionUsage objects including token-detail substructures.
The local engine also uses it to aggregate per-stream usage (prompt tokens
counted once — shared prefill — completion tokens summed).
"""

from __future__ import annotations

from typing import Iterable, Optional

from ..types.openai_compat import CompletionTokensDetails, CompletionUsage, PromptTokensDetails


def consolidate_consensus_usage(usages: Iterable[Optional[CompletionUsage]]) -> Optional[CompletionUsage]:
    usages = list(usages)
    if not usages:
        return None
    out = CompletionUsage(prompt_tokens=0, completion_tokens=0, total_tokens=0)
    for u in usages:
        if u is None:
            continue
        out.prompt_tokens += u.prompt_tokens or 0
        out.completion_tokens += u.completion_tokens or 0
        out.total_tokens += u.total_tokens or 0

        ptd = u.prompt_tokens_details
        if ptd is not None:
            if out.prompt_tokens_details is None:
                out.prompt_tokens_details = PromptTokensDetails()
            for field in ("audio_tokens", "cached_tokens"):
                v = getattr(ptd, field)
                if v is not None:
                    cur = getattr(out.prompt_tokens_details, field) or 0
                    setattr(out.prompt_tokens_details, field, cur + v)

        ctd = u.completion_tokens_details
        if ctd is not None:
            if out.completion_tokens_details is None:
                out.completion_tokens_details = CompletionTokensDetails()
            for field in (
                "audio_tokens",
                "accepted_prediction_tokens",
                "rejected_prediction_tokens",
                "reasoning_tokens",
            ):
                v = getattr(ctd, field)
                if v is not None:
                    cur = getattr(out.completion_tokens_details, field) or 0
                    setattr(out.completion_tokens_details, field, cur + v)
    return out
