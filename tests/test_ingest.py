"""Ingest pipeline tests (reference src/tests/test_document_ingestor_*.py
capability: load→chunk→embed→store, file readers, directory walker)."""

from __future__ import annotations

import json

from sentio_amd.config import Settings
from sentio_amd.models.document import Document
from sentio_amd.serving.container import ServiceContainer


def _container():
    s = Settings()
    s.mock_compute = True
    s.device = "cpu"
    s.chunk_size = 64
    s.chunk_overlap = 8
    return ServiceContainer(s)


def test_ingest_document_chunks_and_indexes():
    c = _container()
    text = "sentence one about GPUs. " * 20
    result = c.ingestor().ingest_document(Document(text=text, id="doc1"))
    assert result["chunks"] >= 2
    assert len(c.dense_index()) == result["chunks"]
    assert c.bm25_index().n_docs == result["chunks"]
    # chunks carry parent_id
    some = c.dense_index().get_document(c.dense_index().doc_ids[0])
    assert some.metadata["parent_id"] == "doc1"


def test_ingest_documents_stats_accumulate():
    c = _container()
    docs = [Document(text=f"document number {i} " * 30, id=f"d{i}")
            for i in range(3)]
    result = c.ingestor().ingest_documents(docs)
    assert result["documents"] == 3
    assert result["chunks"] == len(c.dense_index())


def test_ingest_directory_readers(tmp_path):
    (tmp_path / "a.txt").write_text("plain text file " * 30)
    (tmp_path / "b.md").write_text("# heading\nmarkdown body " * 20)
    (tmp_path / "c.html").write_text(
        "<html><body><p>html body text</p><script>junk()</script></body></html>")
    (tmp_path / "d.json").write_text(json.dumps({"k": "json value"}))
    (tmp_path / "skip.bin").write_bytes(b"\x00\x01")
    from sentio_amd.ingest.ingestor import ingest_directory

    c = _container()
    result = ingest_directory(str(tmp_path), c.ingestor())
    assert result["documents"] == 4          # .bin skipped
    assert len(c.dense_index()) >= 4
    # html reader stripped the script
    texts = [c.dense_index().get_document(i).text
             for i in c.dense_index().doc_ids]
    assert not any("junk()" in t for t in texts)


def test_ingested_docs_are_retrievable():
    c = _container()
    c.ingestor().ingest_documents([
        Document(text="the capybara is the largest living rodent", id="capy"),
        Document(text="MI355X has 288 GB of HBM3E memory", id="gpu"),
    ])
    docs = c.retriever().retrieve("HBM3E memory GPU", top_k=2)
    assert docs
    assert any("HBM3E" in d.text for d in docs)


def test_clear_endpoint_resets_indexes():
    c = _container()
    c.ingestor().ingest_documents([Document(text="x " * 100, id="a")])
    assert len(c.dense_index()) > 0
    c.clear_indexes()
    assert len(c.dense_index()) == 0
    assert c.bm25_index().n_docs == 0
    # pipeline still functional after clear
    c.ingestor().ingest_documents([Document(text="y " * 100, id="b")])
    assert len(c.dense_index()) > 0
