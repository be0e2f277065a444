"""KLLMsChatCompletion return type (reference: k_llms/types/completions.py:7-15)."""

from typing import Any, Dict, Optional

from pydantic import Field

from .openai_compat import ChatCompletion


class KLLMsChatCompletion(ChatCompletion):
    """ChatCompletion extended with per-field consensus likelihoods.

    ``likelihoods`` mirrors the structure of the extraction object: nested
    dicts for objects, lists for arrays, floats at leaves.
    """

    likelihoods: Optional[Dict[str, Any]] = Field(
        default=None,
        description=(
            "Object defining the uncertainties of the fields extracted when using "
            "consensus. Follows the same structure as the extraction object."
        ),
    )
