"""Materialized view of DHT block state (parity: client/routing/sequence_info.py)."""

from __future__ import annotations

import dataclasses
import time
from typing import Dict, List, Optional, Sequence, Tuple

from petals_amd.data_structures import ModuleUID, RemoteModuleInfo, RemoteSpanInfo, ServerState


@dataclasses.dataclass
class RemoteSequenceInfo:
    block_uids: Tuple[ModuleUID, ...]
    block_infos: List[Optional[RemoteModuleInfo]]
    spans_by_priority: List[RemoteSpanInfo]  # sorted by length, longest first
    spans_containing_block: Tuple[List[RemoteSpanInfo], ...]
    last_updated_time: float

    @classmethod
    def make_empty(cls, block_uids: Sequence[ModuleUID]) -> "RemoteSequenceInfo":
        block_uids = tuple(block_uids)
        empty = tuple([] for _ in block_uids)
        return cls(block_uids, [None] * len(block_uids), [], empty, last_updated_time=-float("inf"))

    def __len__(self):
        return len(self.block_uids)

    def update_(self, new_block_infos: List[Optional[RemoteModuleInfo]]):
        assert len(new_block_infos) == len(self.block_uids)
        self.block_infos = list(new_block_infos)
        self.spans_by_priority, self.spans_containing_block = self.compute_spans(self.block_infos)
        self.last_updated_time = time.monotonic()

    @staticmethod
    def compute_spans(block_infos: List[Optional[RemoteModuleInfo]]):
        num_blocks = len(block_infos)
        spans: Dict[str, RemoteSpanInfo] = {}
        active: Dict[str, RemoteSpanInfo] = {}
        all_spans: List[RemoteSpanInfo] = []
        for block_idx, info in enumerate(block_infos):
            servers = info.servers if info is not None else {}
            for peer_id, server_info in servers.items():
                if server_info.state != ServerState.ONLINE:
                    continue
                span = active.get(peer_id)
                if span is not None and span.end == block_idx:
                    span.end = block_idx + 1
                else:
                    span = RemoteSpanInfo(peer_id=peer_id, start=block_idx, end=block_idx + 1, server_info=server_info)
                    active[peer_id] = span
                    all_spans.append(span)
            # drop spans that did not extend into this block
            for peer_id in list(active.keys()):
                if active[peer_id].end <= block_idx:
                    del active[peer_id]

        spans_by_priority = sorted(all_spans, key=lambda s: s.length, reverse=True)
        spans_containing_block = tuple([] for _ in range(num_blocks))
        for span in all_spans:
            for i in range(span.start, span.end):
                spans_containing_block[i].append(span)
        return spans_by_priority, spans_containing_block
