from sentio_amd.ingest.chunker import TextChunker  # noqa: F401
from sentio_amd.ingest.ingestor import DocumentIngestor, ingest_directory  # noqa: F401
