# IndexCfg surface: JSON round-trip, unknown-kwarg tolerance (reference
# quirk 7: shipped configs carry dead keys that must land in .extra),
# metric mapping. Pins reference test TestIndexCfg
# (tests/test_integration.py:419-421) with the reference's own fixture
# key set (tests/test_index_config.json).
import json

import pytest

from distributed_faiss_amd.index_cfg import (
    IndexCfg,
    METRIC_INNER_PRODUCT,
    METRIC_L2,
)
from distributed_faiss_amd.engine_spec import resolve_engine_spec, infer_n_centroids


def test_from_json_with_dead_keys(tmp_path):
    # same key set as the reference fixture tests/test_index_config.json:
    # factory_type is a DEAD key and must be silently absorbed into extra
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps({
        "index_storage_dir": "/tmp/save",
        "dim": "1024",
        "factory_type": "IVF{centroids},SQ8",
        "centroids": "1000",
    }))
    cfg = IndexCfg.from_json(str(p))
    assert cfg.dim == 1024
    assert cfg.index_storage_dir == "/tmp/save"
    assert cfg.extra["factory_type"] == "IVF{centroids},SQ8"
    assert cfg.centroids == "1000"  # not coerced until factory resolution


def test_json_round_trip(tmp_path):
    cfg = IndexCfg(index_builder_type="ivf_simple", dim=64, centroids=16,
                   metric="l2", nprobe=4, train_num=100)
    p = tmp_path / "cfg.json"
    p.write_text(cfg.to_json_string())
    # to_json_string dumps .extra as a nested dict; from_json absorbs it
    loaded = json.loads(p.read_text())
    loaded.update(loaded.pop("extra", {}))
    cfg2 = IndexCfg(**loaded)
    assert cfg2.dim == 64 and cfg2.centroids == 16 and cfg2.metric == "l2"
    assert cfg2.nprobe == 4 and cfg2.train_num == 100


def test_from_json_flattens_extra(tmp_path):
    # to_json_string serializes cfg.extra as a top-level "extra" key;
    # from_json must flatten it back so engine knobs (code_size,
    # bits_per_vector, seed — read by engine_spec at train time) survive
    # a cfg.json round trip instead of nesting as extra["extra"].
    cfg = IndexCfg(index_builder_type="knnlm", dim=128, centroids=64,
                   metric="l2", code_size=16, bits_per_vector=8, seed=7)
    p = tmp_path / "cfg.json"
    p.write_text(cfg.to_json_string())
    cfg2 = IndexCfg.from_json(str(p))
    assert cfg2.extra["code_size"] == 16
    assert cfg2.extra["bits_per_vector"] == 8
    assert cfg2.extra["seed"] == 7
    assert "extra" not in cfg2.extra
    spec = resolve_engine_spec(cfg2, 1000)
    assert spec["m"] == 16 and spec["seed"] == 7


def test_metric_mapping():
    assert IndexCfg(metric="dot").get_metric() == METRIC_INNER_PRODUCT
    assert IndexCfg(metric="l2").get_metric() == METRIC_L2
    with pytest.raises(RuntimeError):
        IndexCfg(metric="cosine").get_metric()


def test_builder_flat_ignores_metric():
    # reference quirk 3 (index.py:94): builder "flat" is ALWAYS inner product
    cfg = IndexCfg(index_builder_type="flat", dim=32, metric="l2")
    spec = resolve_engine_spec(cfg, 1000)
    assert spec["type"] == "flat"
    assert spec["metric"] == METRIC_INNER_PRODUCT


def test_builder_knnlm_defaults_and_nprobe_overwrite():
    # reference index.py:43-48: code_size default 64, nprobe reset to 1
    cfg = IndexCfg(index_builder_type="knnlm", dim=128, centroids=64,
                   metric="l2", nprobe=32)
    spec = resolve_engine_spec(cfg, 1000)
    assert spec["type"] == "ivfpq" and spec["m"] == 64 and spec["nbits"] == 8
    assert spec["nprobe"] == 1 and cfg.nprobe == 1


def test_builder_ivfsq_is_fp16():
    # reference quirk 5 (index.py:65): builder "ivfsq" uses QT_fp16
    cfg = IndexCfg(index_builder_type="ivfsq", dim=128, centroids=64, metric="l2")
    spec = resolve_engine_spec(cfg, 1000)
    assert spec["type"] == "ivfsq" and spec["sq_type"] == "fp16"


def test_factory_string_paths():
    cfg = IndexCfg(faiss_factory="IVF{centroids},SQ8", dim=64, centroids="1000",
                   metric="l2")
    spec = resolve_engine_spec(cfg, 5000)
    assert spec["type"] == "ivfsq" and spec["sq_type"] == "8bit" and spec["nlist"] == 1000

    cfg = IndexCfg(faiss_factory="IVF{centroids},PQ8", dim=64, centroids=0, metric="dot")
    spec = resolve_engine_spec(cfg, 10000)
    # centroids inferred: 2*sqrt(10000) = 200 (reference index.py:497-508)
    assert spec["nlist"] == 200 and spec["type"] == "ivfpq" and spec["m"] == 8

    cfg = IndexCfg(faiss_factory="IVF32,Flat", dim=16, metric="l2")
    spec = resolve_engine_spec(cfg, 10000)
    assert spec["type"] == "ivf_flat" and spec["nlist"] == 32


def test_infer_n_centroids_thresholds():
    # reference index.py:497-508 (note 10e5 == 1e6 as written)
    assert infer_n_centroids(10000) == 200
    assert infer_n_centroids(int(10e5)) == 65536
    assert infer_n_centroids(int(10e6)) == 262144
    assert infer_n_centroids(int(10e7)) == 1048576


def test_out_of_scope_builders_raise():
    # ivf_gpu stays out of scope (the faiss-GPU wrapper is what this
    # engine replaces); hnswsq resolves since round 2
    cfg = IndexCfg(index_builder_type="ivf_gpu", dim=16, metric="l2", centroids=4)
    with pytest.raises(NotImplementedError):
        resolve_engine_spec(cfg, 100)


def test_builder_hnswsq():
    # reference index.py:51-60: store_n/ef_construction from extra,
    # L2 asserted
    cfg = IndexCfg(index_builder_type="hnswsq", dim=32, metric="l2",
                   nprobe=48, store_n=24, ef_construction=80)
    spec = resolve_engine_spec(cfg, 100)
    assert spec["type"] == "hnswsq" and spec["m"] == 24
    assert spec["ef_construction"] == 80 and spec["nprobe"] == 48
    cfg2 = IndexCfg(index_builder_type="hnswsq", dim=32, metric="l2")
    spec2 = resolve_engine_spec(cfg2, 100)
    assert spec2["m"] == 128 and spec2["ef_construction"] == 100  # defaults
    with pytest.raises(AssertionError):  # reference asserts L2
        resolve_engine_spec(
            IndexCfg(index_builder_type="hnswsq", dim=32, metric="dot"), 100)
