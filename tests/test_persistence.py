# Persistence / restart flows (reference call stack (4)-(5)):
# save -> load_index round trip, buffer replay, cfg.json override,
# autosave watcher, get_ids, concurrent clients (reference
# tests/test_rpc.py:57-63 concurrency smoke, in-process).
import os
import pickle
import time

import numpy as np
import pytest

from distributed_faiss_amd import IndexCfg, IndexClient, IndexServer, IndexState
from distributed_faiss_amd.index import Index, get_index_files
from oracle import OracleProvider


def _wait(pred, timeout=20):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if pred():
            return True
        time.sleep(0.05)
    return False


@pytest.fixture
def prov():
    return OracleProvider()


def test_index_files_layout():
    f = get_index_files("/x/y")
    assert f == ("/x/y/index.dfann", "/x/y/meta.pkl", "/x/y/buffer.pkl",
                 "/x/y/cfg.json")


def test_save_then_from_storage_dir(tmp_path, prov):
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4,
                   index_storage_dir=str(tmp_path))
    idx = Index(cfg, provider=prov)
    rng = np.random.default_rng(0)
    emb = rng.random((20, 16), dtype=np.float32)
    idx.add_batch(emb, [f"m{i}" for i in range(20)],
                  train_async_if_triggered=False)
    assert _wait(lambda: idx.get_state() == IndexState.TRAINED)
    assert idx.save()
    idx2 = Index.from_storage_dir(str(tmp_path), cfg, provider=prov)
    assert idx2 is not None
    assert idx2.engine.ntotal == 20
    D, meta, _ = idx2.search(emb[:2], 3)
    assert meta[0][0] == "m0" or meta[0][0] is not None


def test_from_storage_dir_requires_meta(tmp_path, prov):
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4,
                   index_storage_dir=str(tmp_path))
    idx = Index(cfg, provider=prov)
    idx.add_batch(np.ones((5, 16), np.float32), None,
                  train_async_if_triggered=False)
    assert _wait(lambda: idx.get_state() == IndexState.TRAINED)
    idx.save()
    os.remove(os.path.join(str(tmp_path), "meta.pkl"))
    # reference index.py:300-309: meta file is a hard requirement
    with pytest.raises(RuntimeError, match="meta"):
        Index.from_storage_dir(str(tmp_path), cfg, provider=prov)


def test_buffer_replay_on_load(tmp_path, prov):
    # reference index.py:333-337: meta covering index+buffer triggers
    # re-add of the buffered batches on load
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4,
                   index_storage_dir=str(tmp_path))
    idx = Index(cfg, provider=prov)
    emb = np.random.default_rng(1).random((6, 16), dtype=np.float32)
    idx.add_batch(emb, list("abcdef"), train_async_if_triggered=False)
    assert _wait(lambda: idx.get_state() == IndexState.TRAINED)
    assert _wait(lambda: idx.engine.ntotal == 6)
    idx.save()
    # append a fake buffered batch + extend metadata to cover it
    extra = np.random.default_rng(2).random((3, 16), dtype=np.float32)
    _, meta_file, buffer_file, _ = get_index_files(str(tmp_path))
    with open(buffer_file, "wb") as f:
        pickle.dump([extra], f)
    with open(meta_file, "rb") as f:
        meta = pickle.load(f)
    with open(meta_file, "wb") as f:
        pickle.dump(meta + ["x", "y", "z"], f)
    idx2 = Index.from_storage_dir(str(tmp_path), cfg, ignore_buffer=False,
                                  provider=prov)
    assert _wait(lambda: idx2.engine.ntotal == 9)


def test_metadata_mismatch_truncates(tmp_path, prov):
    # reference index.py:338-344: meta not covering buffer -> buffer ignored
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4,
                   index_storage_dir=str(tmp_path))
    idx = Index(cfg, provider=prov)
    emb = np.random.default_rng(1).random((6, 16), dtype=np.float32)
    idx.add_batch(emb, list("abcdef"), train_async_if_triggered=False)
    assert _wait(lambda: idx.get_state() == IndexState.TRAINED)
    assert _wait(lambda: idx.engine.ntotal == 6)
    idx.save()
    extra = np.random.default_rng(2).random((3, 16), dtype=np.float32)
    _, _, buffer_file, _ = get_index_files(str(tmp_path))
    with open(buffer_file, "wb") as f:
        pickle.dump([extra], f)  # meta NOT extended
    idx2 = Index.from_storage_dir(str(tmp_path), cfg, ignore_buffer=False,
                                  provider=prov)
    assert idx2.engine.ntotal == 6
    assert len(idx2.id_to_metadata) == 6


def test_autosave_watcher(tmp_path, prov):
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4,
                   save_interval_sec=1, index_storage_dir=str(tmp_path))
    idx = Index(cfg, provider=prov)
    emb = np.random.default_rng(3).random((8, 16), dtype=np.float32)
    idx.add_batch(emb, None, train_async_if_triggered=False)
    assert _wait(lambda: idx.get_state() == IndexState.TRAINED)
    index_file = get_index_files(str(tmp_path))[0]
    assert _wait(lambda: os.path.exists(index_file), timeout=10)


def test_get_ids_custom_idx(tmp_path, prov):
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4,
                   custom_meta_id_idx=1)
    srv = IndexServer(0, str(tmp_path), provider=prov)
    srv.serve(3301)
    cli = IndexClient(servers=[srv])
    cli.create_index("ids", cfg)
    emb = np.random.default_rng(4).random((5, 16), dtype=np.float32)
    meta = [("a", 10), ("b", 11), ("c", 12), ("d", 13), ("e", 14)]
    cli.add_index_data("ids", emb, meta, train_async_if_triggered=False)
    assert _wait(lambda: cli.get_state("ids") == IndexState.TRAINED)
    assert cli.get_ids("ids") == {10, 11, 12, 13, 14}


def test_concurrent_clients_search(tmp_path, prov):
    # reference tests/test_rpc.py:57-63: many concurrent clients against
    # one server; here threads against the in-process server
    import threading

    srv = IndexServer(0, str(tmp_path), provider=prov)
    srv.serve(3302)
    cli = IndexClient(servers=[srv])
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4)
    cli.create_index("conc", cfg)
    emb = np.random.default_rng(5).random((100, 16), dtype=np.float32)
    cli.add_index_data("conc", emb, None, train_async_if_triggered=False)
    assert _wait(lambda: cli.get_state("conc") == IndexState.TRAINED)
    errors = []

    def worker():
        try:
            c = IndexClient(servers=[srv])
            c.create_index("conc", cfg)  # no-op server-side; sets client cfg
            for _ in range(5):
                D, meta = c.search(emb[:4], 3, "conc")
                assert D.shape == (4, 3)
        except Exception as e:  # pragma: no cover
            errors.append(e)

    ts = [threading.Thread(target=worker) for _ in range(10)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(60)
    assert not errors


def test_drop_and_recreate(tmp_path, prov):
    srv = IndexServer(0, str(tmp_path), provider=prov)
    srv.serve(3303)
    cli = IndexClient(servers=[srv])
    cfg = IndexCfg(index_builder_type="flat", dim=16, train_num=4)
    cli.create_index("dr", cfg)
    emb = np.random.default_rng(6).random((10, 16), dtype=np.float32)
    cli.add_index_data("dr", emb, None, train_async_if_triggered=False)
    assert _wait(lambda: cli.get_state("dr") == IndexState.TRAINED)
    assert cli.get_ntotal("dr") == 10
    cli.drop_index("dr")
    assert cli.get_ntotal("dr") == 0
    cli.create_index("dr", IndexCfg(index_builder_type="flat", dim=16,
                                    train_num=4))
    cli.add_index_data("dr", emb[:4], None, train_async_if_triggered=False)
    assert _wait(lambda: cli.get_state("dr") == IndexState.TRAINED)
    assert cli.get_ntotal("dr") == 4
