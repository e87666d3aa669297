"""Instant-playlist AI layer.

Reference: /root/reference/tasks/ai/ — planner + tool implementations +
per-vendor providers (providers/openai.py, providers/gemini.py,
providers/mistral.py). The planner/tools/rerank live in
web/api_chat.py; this package holds the vendor adapters
(ai/providers.py), each translating the same tool-calling plan request
into the vendor's wire format and parsing tool calls back.
"""

from audiomuse_amd.ai.providers import plan_with_llm  # noqa: F401
