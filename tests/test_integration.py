# Integration: Index / IndexServer / IndexClient flows, oracle backend.
#
# Ports the reference's integration suite (tests/test_integration.py) onto
# the in-process server shell: the reference spawned 4 real TCP servers on
# localhost ports + 1 single server (reference tests/test_integration.py:
# 54-73); here the same topology is 4+1 in-process IndexServer objects
# registered under port numbers, exercised through the same server-list
# file construction the reference client uses.
#
# BASELINE.json configs[0] is exactly this path (flat dim=64..512, multi
# local CPU servers, no GPU).
import os
import random
import string
import tempfile

import numpy as np
import pytest

from distributed_faiss_amd import (
    IndexCfg,
    IndexClient,
    IndexServer,
    IndexState,
)
from oracle import OracleProvider

EMBED_DIM = 64


def rand_meta(n, nchars=5, rng=None):
    rng = rng or random
    return [
        "".join(rng.choices(string.ascii_uppercase + string.digits, k=nchars))
        for _ in range(n)
    ]


@pytest.fixture
def topology(tmp_path):
    """4 multi-servers + 1 single server, oracle backend, server-list files."""
    prov = OracleProvider()
    multi = [IndexServer(i, str(tmp_path / "multi"), provider=prov) for i in range(4)]
    ports = [2237, 2238, 2239, 2240]
    for s, p in zip(multi, ports):
        s.start_blocking(p)
    single = IndexServer(0, str(tmp_path / "single"), provider=prov)
    single.start_blocking(2241)

    def server_list_file(port_list):
        f = tempfile.NamedTemporaryFile(
            "w", suffix=".txt", delete=False, dir=str(tmp_path)
        )
        f.write(f"{len(port_list)}\n")
        for p in port_list:
            f.write(f"localhost,{p}\n")
        f.close()
        return f.name

    yield {
        "multi": multi,
        "ports": ports,
        "single": single,
        "single_port": 2241,
        "list_file": server_list_file,
    }
    from distributed_faiss_amd.server import unregister_inproc_server

    for p in ports + [2241]:
        unregister_inproc_server(p)


def make_client(topology, ports):
    return IndexClient(topology["list_file"](ports))


def test_train_num_honored(topology):
    # reference tests/test_integration.py:117-146
    train_num = 10
    cfg = IndexCfg(index_builder_type="flat", dim=EMBED_DIM, train_num=train_num)
    client = make_client(topology, [topology["single_port"]])
    index_id = "train_num"
    client.create_index(index_id, cfg)
    rng = np.random.default_rng(0)

    def add_data(ndoc):
        emb = rng.random((ndoc, EMBED_DIM), dtype=np.float32)
        client.add_index_data(index_id, emb, rand_meta(ndoc), train_async_if_triggered=False)
        return client.get_state(index_id)

    state = add_data(train_num - 1)
    assert state == IndexState.NOT_TRAINED
    state = add_data(1)
    assert state != IndexState.NOT_TRAINED
    _wait_trained(client, index_id)

    q = rng.random((4, EMBED_DIM), dtype=np.float32)
    results = client.search(q, 4, index_id)
    assert results[0].shape == (4, 4)
    client.save_index(index_id)
    results = client.search(q, 4, index_id)
    assert results[0].shape == (4, 4)
    client.close()

    client2 = make_client(topology, [topology["single_port"]])
    assert client2.load_index(index_id, cfg)
    assert client2.get_state(index_id) == IndexState.TRAINED
    results = client2.search(q, 4, index_id)
    assert results[0].shape == (4, 4)
    client2.close()


def _wait_trained(client, index_id, timeout=30):
    import time

    t0 = time.time()
    while time.time() - t0 < timeout:
        if client.get_state(index_id) == IndexState.TRAINED:
            return
        time.sleep(0.05)
    raise TimeoutError("index never reached TRAINED")


def test_search_quality_same_for_multiple_clients(topology):
    # THE results-parity invariant (reference tests/test_integration.py:
    # 205-265): sharded search over 4 servers returns exactly equal scores
    # and metadata to a single flat index holding the same data.
    index_id = "quality"
    cfg = IndexCfg(index_builder_type="flat", dim=EMBED_DIM)
    single_client = make_client(topology, [topology["single_port"]])
    single_client.create_index(index_id, cfg)
    clients = [make_client(topology, topology["ports"]) for _ in range(4)]

    assert single_client.get_state(index_id) == IndexState.NOT_TRAINED
    rng = np.random.default_rng(1)
    pyrng = random.Random(1)
    for client in clients:
        client.create_index(index_id, cfg)
        for _ in range(pyrng.randint(1, 4)):
            n = pyrng.randint(1, 1280)
            emb = rng.random((n, EMBED_DIM), dtype=np.float32)
            meta = rand_meta(n, rng=pyrng)
            client.add_index_data(index_id, emb, meta, train_async_if_triggered=False)
            single_client.add_index_data(index_id, emb, meta, train_async_if_triggered=False)
        assert client.get_state(index_id) == IndexState.NOT_TRAINED

    clients[0].sync_train(index_id)
    single_client.sync_train(index_id)
    _wait_trained(clients[0], index_id)
    _wait_trained(single_client, index_id)

    assert clients[0].get_ntotal(index_id) == single_client.get_ntotal(index_id)
    q = rng.random((16, EMBED_DIM), dtype=np.float32)
    scores_aggr, meta_aggr = clients[0].search(q, 5, index_id)
    scores_single, meta_single = single_client.search(q, 5, index_id)
    assert (scores_aggr == scores_single).all()
    assert meta_aggr == meta_single


def test_index_client_multiple_server_balance(topology):
    # reference tests/test_integration.py:267-330: exact round-robin balance
    index_id = "balance"
    cfg = IndexCfg(index_builder_type="flat", dim=EMBED_DIM)
    clients = [make_client(topology, topology["ports"]) for _ in range(4)]
    num_docs_per_batch = 1280
    num_batches = 4
    rng = np.random.default_rng(2)
    for client in clients:
        client.create_index(index_id, cfg)
        for _ in range(num_batches):
            emb = rng.random((num_docs_per_batch, EMBED_DIM), dtype=np.float32)
            client.add_index_data(index_id, emb, rand_meta(num_docs_per_batch),
                                  train_async_if_triggered=False)
    clients[0].sync_train(index_id)
    _wait_trained(clients[0], index_id)
    total = 4 * num_batches * num_docs_per_batch
    assert clients[0].get_ntotal(index_id) == total
    # per-server balance is exact: every client starts at a random server but
    # round-robins, and batches-per-client is a multiple of num_servers
    for srv in topology["multi"]:
        assert srv.get_ntotal(index_id) == total // 4

    q = rng.random((16, EMBED_DIM), dtype=np.float32)
    scores, meta = clients[0].search(q, 5, index_id)
    assert scores.shape == (16, 5)
    assert len(meta) == 16 and len(meta[0]) == 5
    clients[0].drop_index(index_id)
    assert clients[0].get_ntotal(index_id) == 0


def test_config_to_file_and_reload(topology, tmp_path):
    # reference tests/test_integration.py:332-385: cfg.json persisted with
    # the index and honored on load (with override)
    index_id = "cfg_persist"
    cfg = IndexCfg(index_builder_type="flat", dim=EMBED_DIM, train_num=8, nprobe=3)
    client = make_client(topology, [topology["single_port"]])
    client.create_index(index_id, cfg)
    rng = np.random.default_rng(3)
    emb = rng.random((20, EMBED_DIM), dtype=np.float32)
    client.add_index_data(index_id, emb, rand_meta(20), train_async_if_triggered=False)
    _wait_trained(client, index_id)
    client.save_index(index_id)
    cfg_path = topology["single"].get_config_path(index_id)
    assert os.path.isfile(cfg_path)
    loaded_cfg = IndexCfg.from_json(cfg_path)
    assert loaded_cfg.dim == EMBED_DIM

    client2 = make_client(topology, [topology["single_port"]])
    assert client2.load_index(index_id, cfg=None)
    assert client2.cfg.dim == EMBED_DIM
    D, meta = client2.search(rng.random((2, EMBED_DIM), dtype=np.float32), 3, index_id)
    assert D.shape == (2, 3)


def test_get_centroids_ivf(topology):
    # reference tests/test_integration.py:387-416 (ivf_simple centroids)
    index_id = "centroids"
    ncent = 8
    cfg = IndexCfg(index_builder_type="ivf_simple", dim=EMBED_DIM, metric="l2",
                   centroids=ncent, train_num=256, nprobe=4)
    client = make_client(topology, [topology["single_port"]])
    client.create_index(index_id, cfg)
    rng = np.random.default_rng(4)
    emb = rng.random((400, EMBED_DIM), dtype=np.float32)
    client.add_index_data(index_id, emb, None, train_async_if_triggered=False)
    _wait_trained(client, index_id)
    cents = client.get_centroids(index_id)
    assert len(cents) == 1
    assert cents[0].shape == (ncent, EMBED_DIM)


def test_ivfpq_end_to_end_and_set_nprobe(topology):
    index_id = "ivfpq_e2e"
    cfg = IndexCfg(index_builder_type="knnlm", dim=EMBED_DIM, metric="l2",
                   centroids=8, code_size=8)
    client = make_client(topology, topology["ports"])
    client.create_index(index_id, cfg)
    rng = np.random.default_rng(5)
    for _ in range(16):
        emb = rng.random((128, EMBED_DIM), dtype=np.float32)
        client.add_index_data(index_id, emb, rand_meta(128), train_async_if_triggered=False)
    client.sync_train(index_id)
    _wait_trained(client, index_id)
    client.set_nprobe(index_id, 8)
    q = rng.random((4, EMBED_DIM), dtype=np.float32)
    D, meta = client.search(q, 5, index_id)
    assert D.shape == (4, 5)
    assert all(len(row) == 5 for row in meta)
    assert all(m is not None for m in meta[0])  # full probe: results exist


def test_dot_negation_convention_end_to_end(topology):
    # quirk 2: for metric "dot" client.search returns NEGATED scores; the
    # best hit (max dot) comes first with the most negative value
    index_id = "dot_conv"
    cfg = IndexCfg(index_builder_type="flat", dim=EMBED_DIM, metric="dot", train_num=4)
    client = make_client(topology, [topology["single_port"]])
    client.create_index(index_id, cfg)
    rng = np.random.default_rng(6)
    emb = rng.random((32, EMBED_DIM), dtype=np.float32)
    client.add_index_data(index_id, emb, list(range(32)), train_async_if_triggered=False)
    _wait_trained(client, index_id)
    q = emb[7:8] * 10.0
    D, meta = client.search(q, 3, index_id)
    # shard-level (engine) scores are positive dots; merged output negated
    assert D[0, 0] <= D[0, 1] <= D[0, 2]
    assert D[0, 0] < 0
    assert meta[0][0] == 7  # self-match wins


def test_search_with_filter(topology):
    index_id = "filter"
    cfg = IndexCfg(index_builder_type="flat", dim=EMBED_DIM, train_num=4)
    client = make_client(topology, [topology["single_port"]])
    client.create_index(index_id, cfg)
    rng = np.random.default_rng(7)
    emb = rng.random((64, EMBED_DIM), dtype=np.float32)
    meta = [("keep" if i % 2 else "drop", i) for i in range(64)]
    client.add_index_data(index_id, emb, meta, train_async_if_triggered=False)
    _wait_trained(client, index_id)
    q = rng.random((3, EMBED_DIM), dtype=np.float32)
    scores, meta_out = client.search_with_filter(q, 5, index_id, filter_pos=0,
                                                 filter_value="drop")
    for row in meta_out:
        assert all(m[0] == "keep" for m in row)


def test_server_list_parsing(tmp_path):
    # reference tests/test_client.py:19-39 behavior: count line + host,port
    from distributed_faiss_amd.client import IndexClient

    p = tmp_path / "servers.txt"
    p.write_text("2\nhost1,1237\nhost2,1238\n")
    res = IndexClient.read_server_list(str(p), initial_timeout=0.01,
                                       total_max_timeout=0.05)
    assert res == [("host1", 1237), ("host2", 1238)]

    p2 = tmp_path / "bad.txt"
    p2.write_text("3\nhost1,1237\nhost2,1238\n")
    with pytest.raises(AssertionError):
        IndexClient.read_server_list(str(p2), initial_timeout=0.01,
                                     total_max_timeout=0.05)
