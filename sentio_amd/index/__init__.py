from sentio_amd.index.bm25 import BM25Index  # noqa: F401
from sentio_amd.index.dense import DenseIndex  # noqa: F401
from sentio_amd.index.fusion import fuse  # noqa: F401
