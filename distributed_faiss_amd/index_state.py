# MI355X-native sharded ANN engine — index state machine states.
#
# Surface and aggregation rules kept identical to the reference's
# distributed_faiss/index_state.py:11-36: a cluster of shards is TRAINING
# if any shard is TRAINING; else NOT_TRAINED if any shard is; else ADD if
# any shard is still draining its add buffer; else TRAINED.

from enum import Enum
from typing import List


class IndexState(Enum):
    NOT_TRAINED = 1
    TRAINING = 2
    ADD = 3
    TRAINED = 4

    @staticmethod
    def get_aggregated_states(states: List["IndexState"]) -> "IndexState":
        states = set(states)
        assert len(states) > 0
        if len(states) == 1:
            return states.pop()
        if IndexState.TRAINING in states:
            return IndexState.TRAINING
        if IndexState.NOT_TRAINED in states:
            return IndexState.NOT_TRAINED
        if IndexState.ADD in states:
            return IndexState.ADD
        return IndexState.TRAINED
