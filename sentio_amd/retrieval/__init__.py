from sentio_amd.retrieval.base import BaseRetriever, ScorerPlugin  # noqa: F401
from sentio_amd.retrieval.dense import DenseRetriever  # noqa: F401
from sentio_amd.retrieval.sparse import BM25Retriever  # noqa: F401
from sentio_amd.retrieval.hybrid import HybridRetriever  # noqa: F401
from sentio_amd.retrieval.factory import create_retriever  # noqa: F401
