"""Context preparation (reference src/core/llm/generator.py:193-254):
numbered entries `[n] Source: … | page: … | score: …` + citation footer."""

from __future__ import annotations

from sentio_amd.models.document import Document

FOOTER = (
    "\n\nUse the numbered sources above. Cite with [n] after facts. "
    "If information is insufficient, say what is missing."
)


def prepare_context(documents: list[Document]) -> str:
    if not documents:
        return ""
    parts: list[str] = []
    for idx, doc in enumerate(documents, start=1):
        content = doc.text or doc.metadata.get("content", "")
        if not content:
            continue
        source = str(doc.metadata.get("source", f"Document {idx}"))
        page = doc.metadata.get("page") or doc.metadata.get("page_number")
        score = (
            doc.metadata.get("score")
            or doc.metadata.get("hybrid_score")
            or doc.metadata.get("rerank_score")
            or doc.metadata.get("dense_score")
        )
        header_bits = [f"[{idx}] Source: {source}"]
        if page is not None:
            header_bits.append(f"page: {page}")
        if isinstance(score, (int, float)):
            header_bits.append(f"score: {float(score):.3f}")
        parts.append(" | ".join(header_bits) + "\n" + content)
    if not parts:
        return "No content available in retrieved documents."
    return "\n\n".join(parts) + FOOTER


def numbered_context(documents: list[Document]) -> str:
    """Simple numbered context used by the verifier
    (reference graph/nodes.py:451-460)."""
    lines = []
    for idx, doc in enumerate(documents, start=1):
        content = doc.text or doc.metadata.get("content", "")
        lines.append(f"[{idx}] {content}")
    return "\n\n".join(lines)
