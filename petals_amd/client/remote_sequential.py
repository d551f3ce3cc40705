"""nn.Module facade over the span chain (parity: client/remote_sequential.py)."""

from __future__ import annotations

import contextvars
from contextlib import contextmanager
from typing import Optional

import torch
from torch import nn

from petals_amd.client.inference_session import InferenceSession
from petals_amd.client.routing.sequence_manager import RemoteSequenceManager
from petals_amd.client.sequential_autograd import _RemoteSequentialAutogradFunction
from petals_amd.data_structures import UID_DELIMITER
from petals_amd.utils.misc import DUMMY

_active_session = contextvars.ContextVar("active_session", default=None)


class RemoteSequential(nn.Module):
    """A chain of remote transformer blocks acting like nn.Sequential."""

    def __init__(
        self,
        config,  # a ModelConfig with ClientConfig attached (see models/*/model.py)
        *,
        sequence_manager: Optional[RemoteSequenceManager] = None,
        start_block: Optional[int] = None,
        end_block: Optional[int] = None,
        dht=None,
    ):
        super().__init__()
        self.config = config
        if sequence_manager is None:
            if start_block is None:
                start_block = 0
            if end_block is None:
                end_block = config.num_blocks
            block_uids = tuple(f"{config.dht_prefix}{UID_DELIMITER}{i}" for i in range(start_block, end_block))
            sequence_manager = RemoteSequenceManager(config.client, block_uids, dht=dht)
        self.sequence_manager = sequence_manager

    def forward(self, inputs: torch.Tensor, prompts: torch.Tensor = DUMMY, **kwargs) -> torch.Tensor:
        assert inputs.ndim == 3, "inputs must be [batch, seq, hidden]"
        if self.active_session is None:
            return _RemoteSequentialAutogradFunction.apply(inputs, prompts, self.sequence_manager)
        return self.active_session.step(inputs, prompts, **kwargs)

    @property
    def active_session(self) -> Optional[InferenceSession]:
        return _active_session.get()

    @property
    def position(self) -> int:
        session = self.active_session
        return session.position if session else 0

    @contextmanager
    def use_session(self, session: Optional[InferenceSession]):
        token = _active_session.set(session)
        try:
            yield session
        finally:
            _active_session.reset(token)

    @contextmanager
    def inference_session(self, **kwargs) -> InferenceSession:
        with InferenceSession(self.sequence_manager, **kwargs) as session, self.use_session(session):
            yield session

    def __getitem__(self, ix) -> "RemoteSequential":
        if isinstance(ix, int):
            ix = slice(ix, ix + 1)
        assert isinstance(ix, slice) and (ix.step is None or ix.step == 1)
        start, stop = ix.indices(len(self))[:2]
        sub_uids = self.sequence_manager.block_uids[start:stop]
        sub_manager = RemoteSequenceManager(self.sequence_manager.config, sub_uids, dht=self.sequence_manager.dht)
        return RemoteSequential(self.config, sequence_manager=sub_manager)

    def __len__(self):
        return len(self.sequence_manager.block_uids)

    def extra_repr(self) -> str:
        return f"modules={self.sequence_manager.block_uids[0]}..{self.sequence_manager.block_uids[-1]}"
