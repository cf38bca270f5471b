"""Prompt templating (reference src/core/llm/prompt_builder.py:22-162:
prompts/*.md loaded once with class-level cache, {instruction}/{context}/
{query} substitution, per-mode instruction strings, hardcoded fallbacks)."""

from __future__ import annotations

from pathlib import Path

PROMPT_DIR = Path(__file__).parent / "prompts"

MODE_INSTRUCTIONS = {
    "fast": "Answer briefly and directly from the context.",
    "balanced": "Answer accurately from the context with concise explanations.",
    "quality": "Answer thoroughly and precisely from the context, weighing all sources.",
    "creative": "Answer from the context with engaging, well-structured prose.",
}

_FALLBACK_TEMPLATES = {
    "profile.md": "You are a retrieval-grounded assistant. Cite sources with [n].",
    "retrieve.md": "{instruction}\n\nContext:\n{context}\n\nQuestion: {query}\n\nAnswer:",
    "verify.md": (
        'Audit the answer against the context. Reply ONLY JSON: {"verdict": '
        '"pass"|"warn"|"fail", "citations_ok": bool, "notes": [], '
        '"revised_answer"?: str}\nQuestion:\n{query}\nContext:\n{context}\n'
        "Answer:\n{answer}\nJSON:"
    ),
}


class PromptBuilder:
    _cache: dict[str, str] = {}

    def __init__(self, mode: str = "balanced"):
        self.mode = mode

    @classmethod
    def _load(cls, name: str) -> str:
        if name in cls._cache:
            return cls._cache[name]
        path = PROMPT_DIR / name
        try:
            text = path.read_text()
        except OSError:
            text = _FALLBACK_TEMPLATES.get(name, "")
        cls._cache[name] = text
        return text

    def system_prompt(self) -> str:
        return self._load("profile.md").strip()

    def build_qa_prompt(self, query: str, context: str, mode: str | None = None) -> str:
        template = self._load("retrieve.md")
        instruction = MODE_INSTRUCTIONS.get(mode or self.mode, MODE_INSTRUCTIONS["balanced"])
        return (
            template.replace("{instruction}", instruction)
            .replace("{context}", context)
            .replace("{query}", query)
        )

    def build_verify_prompt(self, query: str, context: str, answer: str) -> str:
        template = self._load("verify.md")
        return (
            template.replace("{query}", query)
            .replace("{context}", context)
            .replace("{answer}", answer)
        )

    def fallback_text(self, kind: str = "default") -> str:
        return self._load(f"fallback_{kind}.md").strip()
