"""Demo runner: build the pipeline and answer one query from the CLI
(reference scripts/run_graph.py:47-118 capability — without its bit-rot:
the reference used attribute access on a TypedDict and imported a
nonexistent symbol; this one runs).

Usage:
    python scripts/run_pipeline.py "what is the MI355X?" [--mock]
    python scripts/run_pipeline.py "..." --ingest-dir ./docs
"""

from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("query")
    p.add_argument("--mock", action="store_true",
                   help="deterministic hash engines (no GPU needed)")
    p.add_argument("--ingest-dir", default=None,
                   help="directory of documents to ingest first")
    p.add_argument("--top-k", type=int, default=3)
    args = p.parse_args()

    from sentio_amd.config import Settings
    from sentio_amd.models.document import Document
    from sentio_amd.serving.container import ServiceContainer
    from sentio_amd.serving.handlers import ChatHandler

    s = Settings()
    if args.mock:
        s.mock_compute = True
        s.device = "cpu"
    c = ServiceContainer(s)
    c.initialize_all()

    if args.ingest_dir:
        from sentio_amd.ingest.ingestor import ingest_directory

        stats = ingest_directory(args.ingest_dir, c.ingestor())
        print(f"[ingest] {json.dumps(stats)}", file=sys.stderr)
    else:
        demo = [
            Document(text="MI355X is AMD's CDNA4 data-center GPU with 288 GB "
                          "of HBM3E and 256 compute units.", id="d1",
                     metadata={"source": "demo"}),
            Document(text="RCCL provides all-reduce and all-gather "
                          "collectives over xGMI links between GPUs.", id="d2",
                     metadata={"source": "demo"}),
            Document(text="Flash attention tiles the softmax(QK^T)V product "
                          "so the score matrix never hits HBM.", id="d3",
                     metadata={"source": "demo"}),
        ]
        c.ingestor().ingest_documents(demo)

    out = ChatHandler(c).process(args.query, top_k=args.top_k)
    print(json.dumps(out, indent=2, default=str))


if __name__ == "__main__":
    main()
