"""sentio_amd — an MI355X-native RAG serving engine.

A ground-up rebuild of the capability surface of chernistry/sentio
(reference: /root/reference, a remote-API RAG system — see SURVEY.md) as an
on-device engine for AMD Instinct MI355X (gfx950, CDNA4):

* every remote compute call of the reference (Jina embeddings, Jina rerank,
  OpenAI-compatible generation, Qdrant vector search — reference
  src/core/embeddings/providers/jina.py:165, src/core/rerankers/jina_reranker.py:172,
  src/core/llm/providers/openai.py:117, src/core/retrievers/dense.py:64)
  becomes a hand-written CDNA4 HIP kernel or an on-device engine built on them;
* cross-shard aggregation uses RCCL collectives over xGMI
  (torch.distributed backend "nccl" on ROCm);
* the HTTP surface (/chat, /embed, /health*, /metrics, /info, /clear) and the
  retrieve→rerank→select→generate→verify pipeline semantics match the
  reference (src/core/graph/factory.py:94-188).
"""

__version__ = "0.1.0"

from sentio_amd.config import Settings, settings  # noqa: F401
from sentio_amd.models.document import Document  # noqa: F401
