# Shard-local index state machine.
#
# Re-implements the reference's distributed_faiss/index.py Index class
# (buffer -> train -> add -> search -> save lifecycle) on top of a pluggable
# compute engine. Behavior mirrored line-for-line where observable:
#   * add_batch buffering, fp32 cast, metadata length check, train trigger
#     at cfg.train_num              (reference index.py:138-176, quirk 10)
#   * training-data selection: first train_num rows in arrival order,
#     shuffled in place after slicing (reference index.py:194-211, quirk 6
#     — the reference shuffle uses the UNSEEDED global numpy RNG; kept)
#   * buffer drained through engine.add in cfg.buffer_bsz chunks with
#     _maybe_save after each chunk   (reference index.py:403-433)
#   * search under index_lock + metadata gather with -1 -> None
#     (reference index.py:241-270, quirk 9: ids are arrival positions)
#   * persistence layout {storage}/{index_id}/{rank}/: engine payload
#     ("index.dfann" — our own format, NOT a faiss file), meta.pkl,
#     buffer.pkl, cfg.json          (reference index.py:103-108,435-474)
#   * from_storage_dir: meta file REQUIRED and must cover ntotal; buffer
#     replay optional               (reference index.py:284-344)
#
# The engine is injected via `provider` (create(spec) / load(path)); the
# product default is the HIP/gfx950 engine (hip_engine.HipProvider), which
# raises immediately if the native library is unavailable — there is no
# silent CPU fallback. Tests inject the oracle provider.

import _thread
import logging
import os
import pickle
import threading
import time
from typing import List, Optional, Tuple, Union

import numpy as np

from .engine_spec import resolve_engine_spec
from .index_cfg import IndexCfg
from .index_state import IndexState

logger = logging.getLogger()


def get_index_files(index_storage_dir: str) -> Tuple[str, str, str, str]:
    # reference index.py:103-108 layout; engine payload renamed (our format)
    index_file = os.path.join(index_storage_dir, "index.dfann")
    meta_file = os.path.join(index_storage_dir, "meta.pkl")
    buffer_file = os.path.join(index_storage_dir, "buffer.pkl")
    cfg_file = os.path.join(index_storage_dir, "cfg.json")
    return index_file, meta_file, buffer_file, cfg_file


def _default_provider():
    from .hip_engine import HipProvider  # fails loudly if libdfann.so is absent

    return HipProvider()


class Index:
    def __init__(self, cfg: IndexCfg, provider=None):
        self.cfg = cfg
        self.provider = provider if provider is not None else _default_provider()
        self.embeddings_buffer = []
        self.total_data = 0
        self.id_to_metadata = []
        self.buffer_lock = threading.Lock()
        self.index_lock = threading.Lock()
        self.state = IndexState.NOT_TRAINED
        self.engine = None

        self.index_save_time = time.time()
        self.index_saved_size = 0

        if cfg.save_interval_sec > 0:
            self._run_save_watcher()

    # -- lifecycle ---------------------------------------------------------

    def drop_index(self):
        with self.buffer_lock:
            self.embeddings_buffer = []
            self.total_data = 0
            self.id_to_metadata = []
        with self.index_lock:
            self.engine = None
            self.state = IndexState.NOT_TRAINED

    def add_batch(
        self,
        embeddings: np.ndarray,
        metadata: Optional[List[object]],
        train_async_if_triggered: bool = True,
    ):
        embeddings_num = embeddings.shape[0]
        if not metadata:
            metadata = [None] * embeddings_num
        if embeddings_num != len(metadata):
            raise RuntimeError("metadata length should match the batch size of the embeddings")

        embeddings = embeddings.astype(np.float32)  # reference index.py:151

        with self.buffer_lock:
            self.embeddings_buffer.append(embeddings)
            self.id_to_metadata.extend(metadata)
            self.total_data += embeddings_num
            total_data = self.total_data

        state = self.get_state()
        if state == IndexState.TRAINED and total_data >= 0:
            self.add_buffer_to_index()
        elif state == IndexState.NOT_TRAINED and 0 < self.cfg.train_num <= total_data:
            if train_async_if_triggered:
                _thread.start_new_thread(self.train, ())
            else:
                self.train()

    def get_idx_data_num(self) -> Tuple[int, int]:
        with self.buffer_lock:
            buf_total = self.total_data
        index_total = 0
        with self.index_lock:
            if self.engine is not None:
                index_total = self.engine.ntotal
        return buf_total, index_total

    def train(self) -> None:
        with self.index_lock:
            if self.state in [IndexState.TRAINING, IndexState.TRAINED, IndexState.ADD]:
                return
            self.state = IndexState.TRAINING
        cfg = self.cfg

        with self.buffer_lock:
            embeddings = self.embeddings_buffer
            dim = cfg.dim
            if dim == 0:  # guess from data (reference index.py:196-198)
                dim = embeddings[0].shape[1]
                cfg.dim = dim
            if cfg.train_num > 0:
                train_num = cfg.train_num
            elif cfg.train_ratio >= 1.0:
                train_num = self.total_data
            else:
                train_num = int(cfg.train_ratio * self.total_data)
            all_data_as_np_array = np.concatenate(embeddings, axis=0)

        train_data = all_data_as_np_array[:train_num]
        np.random.shuffle(train_data)  # unseeded, as the reference (index.py:211)
        total_data_size = all_data_as_np_array.shape[0]

        spec = resolve_engine_spec(cfg, total_data_size)
        engine = self.provider.create(spec)
        logger.info(f"Training index with array shaped {train_data.shape}")
        engine.train(train_data)

        with self.index_lock:
            self.engine = engine
            self.state = IndexState.TRAINED
        self.add_buffer_to_index()

    def add_buffer_to_index(self) -> None:
        add_to_index = False
        with self.index_lock:
            if self.state == IndexState.TRAINED:
                add_to_index = True
                self.state = IndexState.ADD
        if add_to_index:
            _thread.start_new_thread(self._add_buffer_to_idx, ())

    def search(
        self, query_batch: np.ndarray, top_k: int = 100, return_embeddings: bool = False
    ) -> Tuple[np.ndarray, List[List[object]], Optional[np.ndarray]]:
        scores, _indexes, results_meta, embs = self.search_full(
            query_batch, top_k, return_embeddings)
        return scores, results_meta, embs

    def search_full(
        self, query_batch: np.ndarray, top_k: int = 100, return_embeddings: bool = False
    ) -> Tuple[np.ndarray, np.ndarray, List[List[object]], Optional[np.ndarray]]:
        """search() plus the shard-local ids — the distributed client's
        fan-out needs ids alongside metadata so the RCCL-gathered merge
        can map winners back (ref client.py:290,297-298 does this with
        synthetic ids; here the (shard, slot) pair does it)."""
        query_batch = np.ascontiguousarray(query_batch, dtype=np.float32)
        with self.index_lock:
            if self.state != IndexState.TRAINED:
                raise RuntimeError(f"Server index is not trained. state: {self.state}")
            if return_embeddings:
                scores, indexes, embs = self.engine.search_and_reconstruct(query_batch, top_k)
            else:
                scores, indexes = self.engine.search(query_batch, top_k)
                embs = None

        return scores, indexes, self.metadata_for(indexes), embs

    def metadata_for(self, indexes: np.ndarray) -> List[List[object]]:
        """Map an (nq, k) id matrix to metadata (-1 -> None), the gather
        the reference does inline at index.py:182-191."""
        nq, n = indexes.shape
        with self.buffer_lock:
            return [
                [
                    self.id_to_metadata[indexes[i, j]] if indexes[i, j] != -1 else None
                    for j in range(n)
                ]
                for i in range(nq)
            ]

    def search_ids_dev(self, qt, top_k: int):
        """Device-resident search: (D, I) stay in HBM (torch cuda
        tensors). Serving-step hot path for the distributed client — no
        metadata epilogue, no D2H. HIP engine only."""
        with self.index_lock:
            if self.state != IndexState.TRAINED:
                raise RuntimeError(f"Server index is not trained. state: {self.state}")
            return self.engine.search_dev(qt, top_k)

    # -- persistence -------------------------------------------------------

    def save(self) -> bool:
        state = self.get_state()
        if state == IndexState.TRAINED:
            return self._maybe_save(ignore_time=True)
        elif state == IndexState.ADD:
            self.index_save_time = 0  # trigger save when the add drain finishes
        else:
            return False

    @classmethod
    def from_storage_dir(
        cls,
        index_storage_dir: str,
        cfg: IndexCfg = None,
        ignore_buffer: bool = True,
        provider=None,
    ) -> Union[None, "Index"]:
        index_file, meta_file, buffer_file, cfg_file = get_index_files(index_storage_dir)
        if not os.path.exists(index_file):
            return None

        if provider is None:
            provider = _default_provider()
        engine = provider.load(index_file)

        if os.path.exists(meta_file):
            with open(meta_file, "rb") as reader:
                meta = pickle.load(reader)
            assert (
                len(meta) >= engine.ntotal
            ), "Deserialized meta list should be at least of index size"
        else:
            raise RuntimeError("no meta file found. Can't use index.")

        buffer = []
        if (not ignore_buffer) and buffer_file and os.path.exists(buffer_file):
            with open(buffer_file, "rb") as reader:
                buffer = pickle.load(reader)

        if cfg is None:
            if os.path.isfile(cfg_file):
                cfg = IndexCfg.from_json(cfg_file)
            else:
                cfg = IndexCfg()

        buffer_size = sum(v.shape[0] for v in buffer)
        result = cls(cfg, provider=provider)
        result.engine = engine
        result.state = IndexState.TRAINED
        result.upd_cfg(cfg)

        if len(meta) == engine.ntotal + buffer_size:
            result.id_to_metadata = meta
            result.embeddings_buffer = buffer
            result.total_data = buffer_size
            if buffer_size > 0:
                result.add_buffer_to_index()
        else:
            logger.warning(
                "Metadata size doesn't match combined index+buffer size: "
                "ignoring buffer, reducing metadata to index size"
            )
            result.id_to_metadata = meta[: engine.ntotal]
        return result

    # -- knobs / introspection --------------------------------------------

    def get_centroids(self):
        with self.index_lock:
            if self.state != IndexState.TRAINED:
                raise RuntimeError("Server index is not trained")
            return self.engine.get_centroids()

    def set_nprobe(self, nprobe: int):
        self.cfg.nprobe = nprobe
        with self.index_lock:
            if self.engine is not None:
                self.engine.nprobe = nprobe

    def get_state(self):
        with self.index_lock:
            return self.state

    def get_ids(self):
        id_idx = self.cfg.custom_meta_id_idx
        r = {meta[id_idx] for meta in self.id_to_metadata if meta}
        return r

    def upd_cfg(self, cfg: IndexCfg):
        self.cfg = cfg
        self._override_nprobe(cfg)

    # -- internals ---------------------------------------------------------

    def _add_buffer_to_idx(self):
        while True:
            bsz = self.cfg.buffer_bsz
            embeddings_to_add = []
            embeddings_to_add_total = 0
            with self.buffer_lock:
                for e in self.embeddings_buffer:
                    embeddings_to_add.append(e)
                    embeddings_to_add_total += e.shape[0]
                    if embeddings_to_add_total >= bsz:
                        break
                if embeddings_to_add_total > 0:
                    # (the reference mutates the buffer OUTSIDE the lock,
                    #  index.py:417-419; we keep it inside — same observable
                    #  behavior, fewer races)
                    self.embeddings_buffer = self.embeddings_buffer[len(embeddings_to_add):]
                    self.total_data -= embeddings_to_add_total

            if embeddings_to_add_total == 0:
                break
            add_data_as_np = np.concatenate(embeddings_to_add, axis=0)
            self.engine.add(add_data_as_np)
            self._maybe_save(ignore_time=False)

        with self.index_lock:
            self.state = IndexState.TRAINED

    def _maybe_save(self, ignore_time: bool = False) -> bool:
        if not ignore_time:
            if self.cfg.save_interval_sec <= 0:
                return False
            if time.time() - self.index_save_time < self.cfg.save_interval_sec:
                return False

        with self.buffer_lock, self.index_lock:
            if self.engine.ntotal == self.index_saved_size:
                return False
            index_storage_dir = self.cfg.index_storage_dir
            index_file, meta_file, buffer_file, cfg_file = get_index_files(index_storage_dir)
            os.makedirs(index_storage_dir, exist_ok=True)
            self.engine.save(index_file)
            with open(meta_file, mode="wb") as f:
                pickle.dump(self.id_to_metadata, f)
            with open(buffer_file, mode="wb") as f:
                pickle.dump(self.embeddings_buffer, f)
            with open(cfg_file, mode="w") as f:
                f.write(self.cfg.to_json_string() + "\n")
            self.index_saved_size = self.engine.ntotal
            self.index_save_time = time.time()
            return True

    def _run_save_watcher(self):
        def _save(idx: "Index"):
            while True:
                time.sleep(idx.cfg.save_interval_sec)
                idx._maybe_save(ignore_time=False)

        _thread.start_new_thread(_save, (self,))

    def _override_nprobe(self, cfg: IndexCfg):
        if self.engine is not None:
            self.engine.nprobe = cfg.nprobe

    @staticmethod
    def infer_n_centroids(total_data_size):
        from .engine_spec import infer_n_centroids

        return infer_n_centroids(total_data_size)
