# TCP transport (multi-node mode): the reference's
# rpc semantics — transparent method proxy, traceback-string error
# propagation re-raised as ServerException (reference rpc.py:126-138),
# many concurrent client processes against one server (reference
# tests/test_rpc.py:57-63 — threads here).
import tempfile
import threading
import time

import numpy as np
import pytest

from distributed_faiss_amd import IndexCfg, IndexClient, IndexServer, IndexState
from distributed_faiss_amd.rpc import ServerException, TcpClient, TcpServer
from oracle import OracleProvider


@pytest.fixture
def tcp_server(tmp_path):
    srv = IndexServer(0, str(tmp_path), provider=OracleProvider())
    tcp = TcpServer(srv, port=0)  # ephemeral port
    tcp.start_background()
    yield srv, tcp
    tcp.stop()


def test_proxy_roundtrip(tcp_server):
    srv, tcp = tcp_server
    cli = TcpClient("127.0.0.1", tcp.port)
    assert cli.get_rank() == 0
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4)
    assert cli.create_index("t", cfg) is True
    emb = np.random.default_rng(0).random((10, 16), dtype=np.float32)
    cli.add_index_data("t", emb, ["m%d" % i for i in range(10)], False)
    for _ in range(100):
        if cli.get_state("t") == IndexState.TRAINED:
            break
        time.sleep(0.05)
    assert cli.get_state("t") == IndexState.TRAINED
    D, meta, _ = cli.search("t", emb[:2], 3, False)
    assert D.shape == (2, 3)
    assert cli.get_ntotal("t") == 10
    cli.close()


def test_server_exception_propagates(tcp_server):
    srv, tcp = tcp_server
    cli = TcpClient("127.0.0.1", tcp.port)
    with pytest.raises(ServerException, match="no index"):
        cli.search("missing", np.zeros((1, 4), np.float32), 1, False)
    with pytest.raises(ServerException, match="unknown method"):
        cli.definitely_not_a_method()
    cli.close()


def test_index_client_over_tcp(tmp_path):
    # full IndexClient flow against two TCP servers resolved from a
    # server-list file (ports not in the in-process registry -> TCP)
    prov = OracleProvider()
    tcps = []
    for r in range(2):
        srv = IndexServer(r, str(tmp_path / str(r)), provider=prov)
        tcp = TcpServer(srv, port=0)
        tcp.start_background()
        tcps.append(tcp)
    with tempfile.NamedTemporaryFile("w", suffix=".txt", delete=False,
                                     dir=str(tmp_path)) as f:
        f.write("2\n")
        for tcp in tcps:
            f.write(f"127.0.0.1,{tcp.port}\n")
        path = f.name

    cli = IndexClient(path)
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4)
    cli.create_index("x", cfg)
    rng = np.random.default_rng(1)
    emb = rng.random((40, 16), dtype=np.float32)
    meta = list(range(40))
    for i in range(0, 40, 10):
        cli.add_index_data("x", emb[i:i + 10], meta[i:i + 10],
                           train_async_if_triggered=False)
    for _ in range(100):
        if cli.get_state("x") == IndexState.TRAINED:
            break
        time.sleep(0.05)
    assert cli.get_ntotal("x") == 40
    q = emb[5:6] * 2.0
    D, m = cli.search(q, 3, "x")
    assert m[0][0] == 5  # self-match under dot
    cli.close()
    for tcp in tcps:
        tcp.stop()


def test_concurrent_tcp_clients(tcp_server):
    srv, tcp = tcp_server
    cfg = IndexCfg(index_builder_type="flat", dim=8, train_num=4)
    boot = TcpClient("127.0.0.1", tcp.port)
    boot.create_index("c", cfg)
    emb = np.random.default_rng(2).random((50, 8), dtype=np.float32)
    boot.add_index_data("c", emb, None, False)
    for _ in range(100):
        if boot.get_state("c") == IndexState.TRAINED:
            break
        time.sleep(0.05)
    errors = []

    def worker():
        try:
            c = TcpClient("127.0.0.1", tcp.port)
            for _ in range(10):
                D, meta, _ = c.search("c", emb[:3], 2, False)
                assert D.shape == (3, 2)
            c.close()
        except Exception as e:  # pragma: no cover
            errors.append(e)

    ts = [threading.Thread(target=worker) for _ in range(10)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(60)
    assert not errors
    boot.close()
