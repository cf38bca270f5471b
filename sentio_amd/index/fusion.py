"""Result fusion: rrf / weighted_rrf / comb_sum.

Behavior parity with the reference's hybrid fusion
(reference src/core/retrievers/hybrid.py:204-259):

* rrf:          score[id] += 1/(rrf_k + rank) per source list
* weighted_rrf: same, dense/sparse lists weighted by dense_weight/sparse_weight
* comb_sum:     min-max normalize each source's raw scores (all-equal → 1.0),
                then weighted sum; plugin lists get weight 0.2
* scorer-plugin scores are added directly to the fused score
* final ranking: fused score desc, truncated to top_k

On-device, the per-shard candidate lists are tiny (top_k ids+scores), so
fusion runs on host over the merged candidate union — the kernels' job is
producing the per-source top-k lists, the merge is latency-trivial.
"""

from __future__ import annotations

from collections import defaultdict
from typing import Sequence

Hit = tuple[str, float]  # (doc_id, raw score), already rank-ordered


def _minmax(values: dict[str, float]) -> dict[str, float]:
    if not values:
        return {}
    vmin = min(values.values())
    vmax = max(values.values())
    if vmax <= vmin:
        return {k: 1.0 for k in values}
    scale = vmax - vmin
    return {k: (v - vmin) / scale for k, v in values.items()}


def fuse(
    dense_hits: Sequence[Hit],
    sparse_hits: Sequence[Hit],
    method: str = "rrf",
    top_k: int = 10,
    rrf_k: int = 60,
    dense_weight: float = 0.7,
    sparse_weight: float = 0.3,
    plugin_hits: Sequence[Hit] = (),
    plugin_weight: float = 0.2,
) -> list[Hit]:
    if method not in ("rrf", "weighted_rrf", "comb_sum"):
        raise ValueError(f"Unknown fusion_method: {method}")

    fused: dict[str, float] = defaultdict(float)

    if method in ("rrf", "weighted_rrf"):
        dw = 1.0 if method == "rrf" else float(dense_weight)
        sw = 1.0 if method == "rrf" else float(sparse_weight)
        for rank, (doc_id, _s) in enumerate(dense_hits):
            fused[doc_id] += dw * (1.0 / (rrf_k + rank))
        for rank, (doc_id, _s) in enumerate(sparse_hits):
            fused[doc_id] += sw * (1.0 / (rrf_k + rank))
        for rank, (doc_id, _s) in enumerate(plugin_hits):
            fused[doc_id] += 1.0 / (rrf_k + rank)
    else:  # comb_sum
        for doc_id, ns in _minmax({d: s for d, s in dense_hits}).items():
            fused[doc_id] += float(dense_weight) * ns
        for doc_id, ns in _minmax({d: s for d, s in sparse_hits}).items():
            fused[doc_id] += float(sparse_weight) * ns
        for doc_id, ns in _minmax({d: s for d, s in plugin_hits}).items():
            fused[doc_id] += plugin_weight * ns

    ranked = sorted(fused.items(), key=lambda kv: kv[1], reverse=True)
    return ranked[:top_k]


def add_plugin_scores(
    fused: list[Hit], doc_scores: dict[str, float]
) -> list[Hit]:
    """Add scorer-plugin scores directly (reference hybrid.py:275-285) and
    re-rank."""
    merged = {d: s for d, s in fused}
    for doc_id, s in doc_scores.items():
        merged[doc_id] = merged.get(doc_id, 0.0) + float(s)
    return sorted(merged.items(), key=lambda kv: kv[1], reverse=True)
