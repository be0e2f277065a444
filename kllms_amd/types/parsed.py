"""KLLMsParsedChatCompletion return type (reference: k_llms/types/parsed.py:7-15)."""

from typing import Any, Dict, Optional

from pydantic import Field

from .openai_compat import ParsedChatCompletion


class KLLMsParsedChatCompletion(ParsedChatCompletion):
    """ParsedChatCompletion extended with per-field consensus likelihoods."""

    likelihoods: Optional[Dict[str, Any]] = Field(
        default=None,
        description=(
            "Object defining the uncertainties of the fields extracted when using "
            "consensus. Follows the same structure as the extraction object."
        ),
    )
