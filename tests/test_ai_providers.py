"""Vendor adapter tests for the instant-playlist planner
(ai/providers.py; reference: tasks/ai/providers/). All offline: the
HTTP call is injected."""

import json

import pytest

from audiomuse_amd import config as C
from audiomuse_amd.ai import providers as P

TOOLS = {"seed_search": {"type": "object",
                         "properties": {"seeds": {"type": "array"}}},
         "text_match": None}


class FakeResp:
    def __init__(self, body):
        self.body = body

    def raise_for_status(self):
        pass

    def json(self):
        return self.body


def test_none_provider_short_circuits(monkeypatch):
    monkeypatch.setattr(C, "AI_PROVIDER", "none")
    assert P.plan_with_llm("p", TOOLS) is None
    assert P.plan_with_llm("p", TOOLS, provider="unknown-vendor") is None


def test_openai_request_and_parse(monkeypatch):
    monkeypatch.setattr(C, "AI_MODEL_NAME", "")
    calls = {}

    def post(url, headers=None, json=None, timeout=None):
        calls["url"] = url
        calls["body"] = json
        return FakeResp({"choices": [{"message": {"tool_calls": [
            {"function": {"name": "seed_search",
                          "arguments": "{\"seeds\": [\"a\"], \"n\": 5}"}},
            {"function": {"name": "text_match",
                          "arguments": "not json"}},
        ]}}]})

    plan = P.plan_with_llm("road trip", TOOLS, provider="openai", post=post)
    assert calls["url"] == "https://api.openai.com/v1/chat/completions"
    body = calls["body"]
    assert body["model"] == "gpt-4o-mini"
    assert body["messages"][1]["content"] == "road trip"
    names = [t["function"]["name"] for t in body["tools"]]
    assert names == ["seed_search", "text_match"]
    # declared schema rides through; missing schema gets the default
    assert body["tools"][0]["function"]["parameters"]["properties"]
    assert plan == [{"tool": "seed_search", "args": {"seeds": ["a"], "n": 5}},
                    {"tool": "text_match", "args": {}}]


def test_mistral_defaults(monkeypatch):
    monkeypatch.setattr(C, "AI_MODEL_NAME", "")
    calls = {}

    def post(url, **kw):
        calls["url"] = url
        calls["body"] = kw["json"]
        return FakeResp({"choices": [{"message": {}}]})

    plan = P.plan_with_llm("x", TOOLS, provider="mistral", post=post)
    assert calls["url"] == "https://api.mistral.ai/v1/chat/completions"
    assert calls["body"]["model"] == "mistral-small-latest"
    assert plan is None  # no tool calls -> heuristic fallback upstream


def test_gemini_request_and_parse(monkeypatch):
    monkeypatch.setattr(C, "AI_MODEL_NAME", "")
    calls = {}

    def post(url, headers=None, json=None, timeout=None):
        calls["url"] = url
        calls["headers"] = headers
        calls["body"] = json
        return FakeResp({"candidates": [{"content": {"parts": [
            {"text": "thinking..."},
            {"functionCall": {"name": "seed_search",
                              "args": {"seeds": ["b"], "n": 3}}},
        ]}}]})

    plan = P.plan_with_llm("chill", TOOLS, provider="gemini", post=post)
    assert calls["url"].endswith("/models/gemini-2.0-flash:generateContent")
    assert "x-goog-api-key" in calls["headers"]
    decls = calls["body"]["tools"][0]["functionDeclarations"]
    assert [d["name"] for d in decls] == ["seed_search", "text_match"]
    assert calls["body"]["contents"][0]["parts"][0]["text"] == "chill"
    assert plan == [{"tool": "seed_search", "args": {"seeds": ["b"], "n": 3}}]


def test_network_failure_returns_none():
    def post(url, **kw):
        raise OSError("no route")

    assert P.plan_with_llm("x", TOOLS, provider="openai", post=post) is None


def test_ssrf_guard_blocks_internal_base(monkeypatch):
    monkeypatch.setattr(C, "AI_BASE_URL", "http://169.254.169.254/latest")

    def post(url, **kw):  # must never be reached
        raise AssertionError("request was sent to a blocked URL")

    assert P.plan_with_llm("x", TOOLS, provider="openai", post=post) is None
