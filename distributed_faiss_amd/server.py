# In-process IndexServer shell.
#
# The reference runs one IndexServer PROCESS per shard behind a
# pickle-over-TCP socket loop (distributed_faiss/server.py:95-135,215-238).
# For the MI355X intra-node build the shards are GPUs in one process group
# and the TCP machinery is retired (BASELINE.json north_star; SURVEY.md §2
# row 'Server process'); this class keeps the server's METHOD surface —
# create/load/save/drop index, add_index_data, search, sync_train,
# async_train, get_state, get_ntotal, get_centroids, set_nprobe, get_ids,
# get_rank, index_loaded, stop — with identical signatures and the same
# {storage}/{index_id}/{rank}/ directory convention
# (reference server.py:382-388), so client code written against the
# reference drops in unchanged.
#
# A process-local port registry replaces socket discovery: tests (and the
# reference's own integration-test flow of writing a server-list file)
# register servers under a port number, and IndexClient resolves
# "host,port" lines against the registry.

import copy
import logging
import os
import threading
from typing import Optional, Tuple

import numpy as np

from .index import Index
from .index_cfg import IndexCfg
from .index_state import IndexState

logger = logging.getLogger()

# process-local replacement for TCP port binding
_PORT_REGISTRY = {}
_PORT_REGISTRY_LOCK = threading.Lock()


def register_inproc_server(port: int, server: "IndexServer"):
    with _PORT_REGISTRY_LOCK:
        _PORT_REGISTRY[int(port)] = server


def unregister_inproc_server(port: int):
    with _PORT_REGISTRY_LOCK:
        _PORT_REGISTRY.pop(int(port), None)


def resolve_inproc_server(port: int) -> "IndexServer":
    with _PORT_REGISTRY_LOCK:
        if int(port) not in _PORT_REGISTRY:
            raise ConnectionError(f"no in-process IndexServer registered on port {port}")
        return _PORT_REGISTRY[int(port)]


class IndexServer:
    def __init__(self, rank: int, index_storage_dir, provider=None):
        self.indexes = {}
        self.indexes_lock = threading.Lock()
        self.rank = rank
        self.index_storage_dir = index_storage_dir
        self.provider = provider

    # -- lifecycle ---------------------------------------------------------

    def start_blocking(self, port, v6=False, load_index=False):
        """Reference server.py:95-115 bound a socket; here: register and park."""
        if load_index:
            self.load_index()
        register_inproc_server(port, self)

    def serve(self, port):
        """Non-blocking registration (preferred for in-process use)."""
        register_inproc_server(port, self)

    def stop(self):
        for index_id in self.indexes:
            self.indexes[index_id].save()

    # -- index management --------------------------------------------------

    def create_index(self, index_id: str, cfg: IndexCfg):
        # The reference server receives cfg over pickle-RPC, so it mutates a
        # COPY and the client's object is untouched (server.py:240-254 via
        # rpc.py framing). Mirror that value semantics in-process.
        cfg = copy.deepcopy(cfg)
        index_storage_dir = self._get_storage_dir(index_id, cfg)
        cfg.index_storage_dir = index_storage_dir
        os.makedirs(index_storage_dir, exist_ok=True)
        with self.indexes_lock:
            if index_id not in self.indexes:
                self.indexes[index_id] = Index(cfg, provider=self.provider)
                return True
            return False

    def load_index(self, index_id: str = "default", cfg: IndexCfg = None) -> bool:
        cfg = copy.deepcopy(cfg)  # RPC value semantics, see create_index
        index_dir = self._get_storage_dir(index_id, cfg)
        if cfg:
            cfg.index_storage_dir = index_dir
        with self.indexes_lock:
            if index_id in self.indexes:
                if cfg:
                    self.indexes[index_id].upd_cfg(cfg)
                return True
            index = Index.from_storage_dir(index_dir, cfg, provider=self.provider)
            if index:
                self.indexes[index_id] = index
                return True
            return False

    def save_index(self, index_id: str):
        with self.indexes_lock:
            if index_id not in self.indexes:
                raise RuntimeError(f"Index with id={index_id} is not initialized")
            index = self.indexes[index_id]
        index.save()

    def drop_index(self, index_id: str):
        with self.indexes_lock:
            if index_id in self.indexes:
                del self.indexes[index_id]

    def index_loaded(self, index_id: str) -> bool:
        with self.indexes_lock:
            return (
                index_id in self.indexes
                and self.indexes[index_id].get_state() == IndexState.TRAINED
            )

    # -- data path ---------------------------------------------------------

    def add_index_data(
        self,
        index_id: str,
        embeddings: np.ndarray,
        metadata=None,
        train_async_if_triggered: bool = True,
    ):
        with self.indexes_lock:
            index = self.indexes[index_id]
        index.add_batch(embeddings, metadata, train_async_if_triggered)

    def search(
        self, index_id: str, query_batch: np.ndarray, top_k: int, return_embeddings: bool
    ) -> Tuple:
        index = self._get_index(index_id)
        return index.search(query_batch, top_k=top_k, return_embeddings=return_embeddings)

    def search_full(
        self, index_id: str, query_batch: np.ndarray, top_k: int,
        return_embeddings: bool = False
    ) -> Tuple:
        """search() plus shard-local ids (distributed-client fan-out)."""
        index = self._get_index(index_id)
        return index.search_full(query_batch, top_k=top_k,
                                 return_embeddings=return_embeddings)

    def search_ids_dev(self, index_id: str, qt, top_k: int):
        """Device-resident (D, I) search — the distributed client's timed
        serving step (no metadata epilogue, results stay in HBM)."""
        return self._get_index(index_id).search_ids_dev(qt, top_k)

    def adopt_index(self, index_id: str, index) -> None:
        """Install an externally built Index under an id (harness
        plumbing: bench.py builds its shard engine from device-resident
        synthetic data, then serves it through the reference surface)."""
        with self.indexes_lock:
            self.indexes[index_id] = index

    def sync_train(self, index_id: str):
        self._get_index(index_id).train()

    def async_train(self, index_id: str):
        # Reference quirk 4 (server.py:317): async_train calls Thread.run(),
        # not .start() — it is synchronous. Mirrored.
        self._get_index(index_id).train()

    def add_buffer_to_index(self, index_id: str):
        return self._get_index(index_id).add_buffer_to_index()

    # -- introspection / knobs --------------------------------------------

    def get_rank(self) -> int:
        return self.rank

    def get_state(self, index_id: str):
        return self._get_index(index_id).get_state()

    def get_ntotal(self, index_id: str) -> int:
        with self.indexes_lock:
            if index_id not in self.indexes:
                return 0
            index = self.indexes[index_id]
        return index.get_idx_data_num()[1]

    def get_aggregated_ntotal(self, index_id: str) -> int:
        with self.indexes_lock:
            index = self.indexes[index_id]
        return index.get_idx_data_num()[0]

    def get_centroids(self, index_id: str):
        return self._get_index(index_id).get_centroids()

    def set_nprobe(self, index_id: str, nprobe: int):
        return self._get_index(index_id).set_nprobe(nprobe)

    def get_ids(self, index_id: str = "default") -> set:
        with self.indexes_lock:
            index = self.indexes[index_id]
        return index.get_ids()

    def get_config_path(self, index_id: str):
        return os.path.join(self.index_storage_dir, index_id, str(self.rank), "cfg.json")

    def set_omp_num_threads(self, num_threads: int) -> None:
        pass  # no OpenMP in this engine; kept for surface parity

    # -- internals ---------------------------------------------------------

    def _get_index(self, index_id: str):
        with self.indexes_lock:
            if index_id not in self.indexes:
                raise RuntimeError("Server has no index with id={}".format(index_id))
            return self.indexes[index_id]

    def _get_storage_dir(self, index_id: str, cfg: Optional[IndexCfg]):
        index_storage_dir = cfg.index_storage_dir if cfg else None
        if not index_storage_dir:
            index_storage_dir = os.path.join(self.index_storage_dir, index_id, str(self.rank))
        else:
            index_storage_dir = os.path.join(index_storage_dir, str(self.rank))
        return index_storage_dir
