from sentio_amd.pipeline.graph import (  # noqa: F401
    GraphConfig,
    RagPipeline,
    build_basic_graph,
    build_streaming_graph,
)
from sentio_amd.pipeline.state import RAGState, create_initial_state  # noqa: F401
