"""Load balancing: which span of blocks should a joining server host?

Same semantics as the reference (`server/block_selection.py:12-95`): compute
per-block aggregate throughput over ONLINE servers, place our span at the
minimum-throughput window, and rebalance only if doing so would raise the
swarm's bottleneck throughput by more than `balance_quality` vs a greedy
re-placement of every server (eps guards against oscillation).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from petals_amd.data_structures import RemoteModuleInfo, RemoteSpanInfo, ServerState


def compute_throughputs(spans: Dict[str, RemoteSpanInfo], *, total_blocks: int) -> np.ndarray:
    # deterministic accumulation order: FP addition is not associative, so an
    # undefined order yields slightly different sums for the same server set and
    # can cause excess block replacements (swarm churn)
    throughputs = np.zeros(total_blocks)
    for peer_id in sorted(spans):
        span = spans[peer_id]
        if span.state != ServerState.OFFLINE:
            throughputs[span.start : span.end] += span.throughput
    return throughputs


def _choose_best_start(throughputs: np.ndarray, num_blocks: int) -> int:
    options = (
        (sorted(throughputs[i : i + num_blocks]), i)
        for i in range(0, len(throughputs) - num_blocks + 1)
    )
    return min(options)[-1]


def choose_best_blocks(num_blocks: int, module_infos: List[Optional[RemoteModuleInfo]]) -> List[int]:
    from petals_amd.data_structures import compute_spans

    spans = compute_spans(module_infos)
    throughputs = compute_throughputs(spans, total_blocks=len(module_infos))
    start = _choose_best_start(throughputs, num_blocks)
    return list(range(start, start + num_blocks))


def should_choose_other_blocks(
    local_peer_id: str,
    module_infos: List[Optional[RemoteModuleInfo]],
    balance_quality: float,
) -> bool:
    if balance_quality > 1.0:
        return True  # forces rebalancing on each check (for tests)

    from petals_amd.data_structures import compute_spans

    spans = compute_spans(module_infos)
    initial_throughput = compute_throughputs(spans, total_blocks=len(module_infos)).min()
    eps = 1e-3

    if local_peer_id not in spans:
        return True
    local_span = spans[local_peer_id]
    throughputs = compute_throughputs(spans, total_blocks=len(module_infos))
    throughputs[local_span.start : local_span.end] -= local_span.throughput * (1 + eps)

    new_start = _choose_best_start(throughputs, local_span.length)
    if local_span.start == new_start:
        return False  # already in the best place

    local_span.start, local_span.end = new_start, new_start + local_span.length
    throughputs[local_span.start : local_span.end] += local_span.throughput * (1 + eps)

    moved = True
    while moved:
        servers = list(spans.keys())
        np.random.shuffle(servers)
        moved = False
        for peer_id in servers:
            span = spans[peer_id]
            throughputs[span.start : span.end] -= span.throughput * (1 + eps)
            new_start = _choose_best_start(throughputs, span.length)
            if span.start != new_start:
                span.start, span.end = new_start, new_start + span.length
                moved = True
            throughputs[span.start : span.end] += span.throughput * (1 + eps)

    new_throughput = throughputs.min()
    if new_throughput < initial_throughput or new_throughput < eps:
        return False
    actual_quality = initial_throughput / new_throughput
    return actual_quality < balance_quality - eps
