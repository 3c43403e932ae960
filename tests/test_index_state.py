# State aggregation rules — pins reference tests/test_index_state.py:14-22.
from distributed_faiss_amd.index_state import IndexState

T, N, A, TR = (
    IndexState.TRAINED,
    IndexState.NOT_TRAINED,
    IndexState.ADD,
    IndexState.TRAINING,
)


def test_aggregation_rules():
    agg = IndexState.get_aggregated_states
    assert agg([T, T]) == T
    assert agg([N]) == N
    assert agg([T, TR]) == TR
    assert agg([T, N]) == N
    assert agg([T, A]) == A
    assert agg([A, N, TR, T]) == TR
    assert agg([A, N, T]) == N
    assert agg([A, T]) == A
