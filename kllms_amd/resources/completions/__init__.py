"""Completions resource (reference: k_llms/resources/completions/__init__.py:1-2)."""

from .completions import AsyncCompletions, Completions

__all__ = ["Completions", "AsyncCompletions"]
