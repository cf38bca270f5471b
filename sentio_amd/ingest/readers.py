"""File readers for bulk ingest
(reference src/core/ingest/ingest.py:172-289: txt/md/html/json/yaml readers,
PDF/DOCX behind optional deps, directory walker)."""

from __future__ import annotations

import json
import logging
import re
from pathlib import Path

from sentio_amd.models.document import Document

logger = logging.getLogger(__name__)

TEXT_SUFFIXES = {".txt", ".md", ".rst", ".py", ".log"}


def _strip_html(text: str) -> str:
    text = re.sub(r"<(script|style)[^>]*>.*?</\1>", " ", text, flags=re.S | re.I)
    text = re.sub(r"<[^>]+>", " ", text)
    return re.sub(r"\s+", " ", text).strip()


def read_file(path: Path) -> Document | None:
    suffix = path.suffix.lower()
    try:
        if suffix in TEXT_SUFFIXES:
            text = path.read_text(errors="replace")
        elif suffix in (".html", ".htm"):
            text = _strip_html(path.read_text(errors="replace"))
        elif suffix == ".json":
            data = json.loads(path.read_text(errors="replace"))
            text = json.dumps(data, indent=1)
        elif suffix in (".yaml", ".yml"):
            import yaml

            data = yaml.safe_load(path.read_text(errors="replace"))
            text = json.dumps(data, indent=1, default=str)
        elif suffix == ".pdf":
            text = _read_pdf(path)
            if text is None:
                return None
        elif suffix == ".docx":
            text = _read_docx(path)
            if text is None:
                return None
        else:
            return None
    except Exception as exc:
        logger.warning("failed to read %s: %s", path, exc)
        return None
    if not text.strip():
        return None
    return Document(text=text, metadata={"source": str(path), "filename": path.name})


def read_directory(directory: str, recursive: bool = True) -> list[Document]:
    root = Path(directory)
    if not root.is_dir():
        raise FileNotFoundError(directory)
    pattern = "**/*" if recursive else "*"
    docs = []
    for path in sorted(root.glob(pattern)):
        if path.is_file():
            doc = read_file(path)
            if doc is not None:
                docs.append(doc)
    return docs


def _read_pdf(path: Path) -> str | None:
    """PDF text extraction (reference ingest.py:172-223 used PyPDF2).
    PyPDF2 is absent from this deployment image; gated import keeps the
    capability wired without a hard dependency."""
    try:
        import PyPDF2  # type: ignore
    except ImportError:
        logger.warning("PyPDF2 not installed; skipping PDF %s", path)
        return None
    try:
        with open(path, "rb") as f:
            reader = PyPDF2.PdfReader(f)
            return "\n".join(page.extract_text() or "" for page in reader.pages)
    except Exception as exc:
        logger.warning("failed to read PDF %s: %s", path, exc)
        return None


def _read_docx(path: Path) -> str | None:
    """DOCX extraction (reference ingest.py used python-docx); gated."""
    try:
        import docx  # type: ignore
    except ImportError:
        logger.warning("python-docx not installed; skipping DOCX %s", path)
        return None
    try:
        d = docx.Document(str(path))
        return "\n".join(p.text for p in d.paragraphs)
    except Exception as exc:
        logger.warning("failed to read DOCX %s: %s", path, exc)
        return None
