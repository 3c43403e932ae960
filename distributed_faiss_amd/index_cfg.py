# MI355X-native sharded ANN engine — index configuration.
#
# Public surface kept identical to the reference's
# distributed_faiss/index_cfg.py (IndexCfg, reference index_cfg.py:11-63):
# same constructor keyword set and defaults, unknown kwargs silently
# collected into `extra` (reference index_cfg.py:27,42 — shipped configs
# carry dead keys like `factory_type` that must be accepted), same JSON
# round-trip (from_json / to_json_string, reference index_cfg.py:54-61).
#
# The only difference: `get_metric()` returns our own metric enum values
# instead of the faiss module's (reference index_cfg.py:44-52 maps
# "dot" -> faiss.METRIC_INNER_PRODUCT, "l2" -> faiss.METRIC_L2). The
# integer values are the same as faiss's enum values (0, 1) so serialized
# configs mean the same thing.

import json

# faiss MetricType enum values, restated (faiss Index.h: METRIC_INNER_PRODUCT=0,
# METRIC_L2=1), consumed the way reference index_cfg.py:44-52 does.
METRIC_INNER_PRODUCT = 0
METRIC_L2 = 1


class IndexCfg:
    def __init__(
        self,
        index_builder_type: str = None,
        faiss_factory: str = None,
        dim: int = 768,
        train_num: int = 0,
        train_ratio: int = 1.0,
        centroids: int = 0,
        metric: str = "dot",
        nprobe: int = 1,
        infer_centroids=False,
        buffer_bsz: int = 50000,
        save_interval_sec: int = -1,
        index_storage_dir: str = None,
        custom_meta_id_idx: int = 0,
        **kwargs,
    ):
        self.index_builder_type = index_builder_type
        self.faiss_factory = faiss_factory
        self.dim = int(dim)
        self.train_num = train_num
        self.train_ratio = train_ratio
        self.centroids = centroids
        self.metric = metric
        self.nprobe = nprobe
        self.infer_centroids = infer_centroids
        self.buffer_bsz = buffer_bsz
        self.save_interval_sec = save_interval_sec
        self.index_storage_dir = index_storage_dir
        self.custom_meta_id_idx = custom_meta_id_idx
        self.extra = kwargs

    def get_metric(self):
        metric = self.metric
        if metric == "dot":
            return METRIC_INNER_PRODUCT
        elif metric == "l2":
            return METRIC_L2
        raise RuntimeError("Only dot and l2 metrics are supported.")

    @classmethod
    def from_json(cls, json_path):
        with open(json_path, "r") as f:
            kwargs = json.load(f)
        # to_json_string serializes `extra` as a top-level key (it dumps
        # __dict__); feeding it back through **kwargs would NEST it
        # (cfg.extra == {"extra": {...}}), silently dropping engine knobs
        # like code_size/bits_per_vector/seed on a train-from-round-trip.
        # Flatten it back into kwargs — the on-disk format and the
        # reference surface are unchanged.
        extra = kwargs.pop("extra", None)
        if isinstance(extra, dict):
            for k, v in extra.items():
                kwargs.setdefault(k, v)
        return cls(**kwargs)

    def to_json_string(self):
        return json.dumps(self, default=lambda o: o.__dict__, sort_keys=True, indent=4)

    def __repr__(self) -> str:
        return f"<IndexCFG: {self.__dict__}>"
