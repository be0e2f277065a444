"""Public response types (reference: k_llms/types/__init__.py:1-4)."""

from .completions import KLLMsChatCompletion
from .parsed import KLLMsParsedChatCompletion

__all__ = ["KLLMsParsedChatCompletion", "KLLMsChatCompletion"]
