"""Answer verification with strict JSON protocol
(reference src/core/llm/answer_verifier.py:27-87: temp 0.0, max 512 tokens,
first-{...}-blob JSON parse, verdict pass|warn|fail, never raises —
defaults to warn; on verdict=fail the pipeline swaps in revised_answer,
reference graph/nodes.py:471-472)."""

from __future__ import annotations

import json
import logging
import re
from typing import Any, TypedDict

from sentio_amd.pipeline.prompt_builder import PromptBuilder

logger = logging.getLogger(__name__)


class VerifyResult(TypedDict, total=False):
    verdict: str
    citations_ok: bool
    notes: list[str]
    revised_answer: str


def extract_json_dict(text: str) -> dict[str, Any] | None:
    """Robust JSON extraction (reference src/core/llm/reply_extractor.py):
    markdown fences, then largest balanced {...} blob, then light repair
    (trailing commas, Python constants)."""
    fence = re.search(r"```(?:json)?\s*(\{.*?\})\s*```", text, re.DOTALL)
    candidates = []
    if fence:
        candidates.append(fence.group(1))
    start = text.find("{")
    end = text.rfind("}")
    if start >= 0 and end > start:
        candidates.append(text[start : end + 1])
    for cand in candidates:
        for attempt in (cand, _repair(cand)):
            try:
                obj = json.loads(attempt)
                if isinstance(obj, dict):
                    return obj
            except Exception:
                continue
    return None


def _repair(s: str) -> str:
    s = re.sub(r",\s*([}\]])", r"\1", s)          # trailing commas
    s = re.sub(r"\bTrue\b", "true", s)
    s = re.sub(r"\bFalse\b", "false", s)
    s = re.sub(r"\bNone\b", "null", s)
    return s


class AnswerVerifier:
    def __init__(self, generator, max_tokens: int = 512):
        self.generator = generator
        self.builder = PromptBuilder()
        self.max_tokens = max_tokens

    def verify(self, query: str, context: str, answer: str) -> VerifyResult:
        prompt = self.builder.build_verify_prompt(query=query, context=context,
                                                  answer=answer)
        try:
            text = self.generator.generate(
                [prompt], max_new_tokens=self.max_tokens, temperature=0.0
            )[0]
            return self._normalize(extract_json_dict(text))
        except Exception as exc:
            logger.warning("verify() failed: %s", exc)
            return VerifyResult(verdict="warn", citations_ok=False,
                                notes=["verifier_error"])

    @staticmethod
    def _normalize(data: dict[str, Any] | None) -> VerifyResult:
        if data is None:
            return VerifyResult(verdict="warn", citations_ok=False,
                                notes=["invalid_json"])
        out = VerifyResult()
        out["verdict"] = str(data.get("verdict", "warn"))
        if out["verdict"] not in ("pass", "warn", "fail"):
            out["verdict"] = "warn"
        out["citations_ok"] = bool(data.get("citations_ok", False))
        out["notes"] = [str(n) for n in (data.get("notes") or [])][:8]
        if isinstance(data.get("revised_answer"), str):
            out["revised_answer"] = data["revised_answer"]
        return out
