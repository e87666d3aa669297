"""LLM provider adapters for the instant-playlist planner.

Reference: /root/reference/tasks/ai/providers/ — openai.py (tool-calling
chat completions), gemini.py (generateContent + functionDeclarations),
mistral.py (OpenAI-compatible chat with Mistral auth/model defaults).
Each adapter turns one planning prompt + tool schema into a vendor
request and parses the returned tool calls into the planner's
[{"tool": name, "args": {...}}] form. All network failures return None
(the planner falls back to the deterministic heuristic plan).

Outbound URLs pass the SSRF guard (utils/logging_utils.py:
validate_outbound_url) before any request is made — same policy as the
reference's ssrf_guard.py.
"""

from __future__ import annotations

import json
import logging
from typing import Dict, List, Optional

from audiomuse_amd import config as C
from audiomuse_amd.utils.logging_utils import validate_outbound_url

logger = logging.getLogger(__name__)

_SYSTEM = ("Plan music-library tool calls for the user's playlist "
           "request. Use only the provided tools.")

_DEFAULT_BASE = {
    "openai": "https://api.openai.com/v1",
    "mistral": "https://api.mistral.ai/v1",
    "gemini": "https://generativelanguage.googleapis.com/v1beta",
}
_DEFAULT_MODEL = {
    "openai": "gpt-4o-mini",
    "mistral": "mistral-small-latest",
    "gemini": "gemini-2.0-flash",
}


def _base_and_key(provider: str) -> tuple:
    """Per-vendor URL/key config (reference providers/* +
    PARAMETERS.md: OPENAI_SERVER_URL/OLLAMA_SERVER_URL/*_API_KEY);
    AI_BASE_URL / AI_API_KEY are the generic overrides."""
    vendor_base = {
        "openai": C.OPENAI_SERVER_URL,
        "ollama": C.OLLAMA_SERVER_URL,
        "mistral": _DEFAULT_BASE["mistral"],
        "gemini": _DEFAULT_BASE["gemini"],
    }.get(provider) or _DEFAULT_BASE.get(provider, "")
    vendor_key = {
        "openai": C.OPENAI_API_KEY,
        "gemini": C.GEMINI_API_KEY,
        "mistral": C.MISTRAL_API_KEY,
    }.get(provider, "")
    base = (C.AI_BASE_URL or vendor_base).rstrip("/")
    key = C.AI_API_KEY or vendor_key
    validate_outbound_url(base)
    return base, key


def _model_for(provider: str) -> str:
    vendor_model = {
        "openai": C.OPENAI_MODEL_NAME,
        "ollama": C.OLLAMA_MODEL_NAME,
        "gemini": C.GEMINI_MODEL_NAME,
        "mistral": C.MISTRAL_MODEL_NAME,
    }.get(provider, "")
    return C.AI_MODEL_NAME or vendor_model or _DEFAULT_MODEL.get(
        provider, "")


def build_openai_request(provider: str, prompt: str,
                         tools: Dict[str, Dict]) -> Dict:
    """OpenAI-compatible chat/completions body (openai.py + mistral.py:
    Mistral's chat API is OpenAI-wire-compatible)."""
    return {
        "model": _model_for(provider),
        "messages": [{"role": "system", "content": _SYSTEM},
                     {"role": "user", "content": prompt}],
        "temperature": C.AI_TOOLCALL_TEMPERATURE,
        "top_p": C.AI_TOOLCALL_TOP_P,
        "max_tokens": C.AI_TOOLCALL_NUM_PREDICT,
        # ollama's OpenAI-compatible endpoint honors these; cloud
        # vendors ignore unknown sampler fields
        **({"top_k": C.AI_TOOLCALL_TOP_K, "min_p": C.AI_TOOLCALL_MIN_P}
           if provider == "ollama" else {}),
        "tools": [{"type": "function",
                   "function": {"name": name,
                                "parameters": schema or {"type": "object"}}}
                  for name, schema in tools.items()],
        "tool_choice": "auto",
    }


def parse_openai_response(body: Dict) -> Optional[List[Dict]]:
    msg = body["choices"][0]["message"]
    plan = []
    for tc in msg.get("tool_calls") or []:
        fn = tc.get("function", {})
        try:
            args = json.loads(fn.get("arguments") or "{}")
        except Exception:
            args = {}
        plan.append({"tool": fn.get("name"), "args": args})
    return plan or None


def build_gemini_request(prompt: str, tools: Dict[str, Dict]) -> Dict:
    """generateContent body with functionDeclarations (gemini.py)."""
    return {
        "system_instruction": {"parts": [{"text": _SYSTEM}]},
        "contents": [{"role": "user", "parts": [{"text": prompt}]}],
        "tools": [{"functionDeclarations": [
            {"name": name,
             "parameters": schema or {"type": "object", "properties": {}}}
            for name, schema in tools.items()]}],
        "tool_config": {"function_calling_config": {"mode": "AUTO"}},
    }


def parse_gemini_response(body: Dict) -> Optional[List[Dict]]:
    plan = []
    for cand in body.get("candidates") or []:
        for part in cand.get("content", {}).get("parts", []):
            fc = part.get("functionCall")
            if fc:
                plan.append({"tool": fc.get("name"),
                             "args": dict(fc.get("args") or {})})
    return plan or None


def plan_with_llm(prompt: str, tools: Dict[str, Dict],
                  provider: Optional[str] = None,
                  post=None) -> Optional[List[Dict]]:
    """One tool-calling request to the configured provider; None on any
    failure or when AI_PROVIDER is none. `post` injects the HTTP call in
    tests (defaults to requests.post)."""
    provider = (provider or C.AI_PROVIDER or "none").lower()
    if provider in ("", "none"):
        return None
    if provider not in _DEFAULT_BASE and provider != "ollama":
        logger.warning("unknown AI provider %r", provider)
        return None
    try:
        if post is None:
            import requests

            post = requests.post
        base, key = _base_and_key(provider)
        if provider == "gemini":
            model = _model_for(provider)
            r = post(f"{base}/models/{model}:generateContent",
                     headers={"x-goog-api-key": key},
                     json=build_gemini_request(prompt, tools),
                     timeout=C.AI_REQUEST_TIMEOUT_SECONDS)
            r.raise_for_status()
            return parse_gemini_response(r.json())
        r = post(f"{base}/chat/completions",
                 headers={"Authorization": f"Bearer {key}"},
                 json=build_openai_request(provider, prompt, tools),
                 timeout=C.AI_REQUEST_TIMEOUT_SECONDS)
        r.raise_for_status()
        return parse_openai_response(r.json())
    except Exception:
        logger.debug("LLM plan failed; falling back to heuristic",
                     exc_info=True)
        return None
