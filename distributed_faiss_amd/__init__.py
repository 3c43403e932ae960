# distributed_faiss_amd — MI355X-native sharded ANN search engine.
#
# Drop-in for the distributed-faiss hot path: the
# IVF/IVFPQ/IVFSQ build+search path of distributed_faiss/index.py plus the
# client fan-out/top-k merge of distributed_faiss/client.py, rebuilt
# MI355X-first: hand-written gfx950 HIP kernels behind a C-ABI library
# (include/dfann.h, csrc/), one GPU per shard, RCCL top-k merge over xGMI
# for the multi-GPU path (dist.py).
#
# Public surface mirrors the reference package:
#   IndexCfg, IndexState, Index, IndexServer, IndexClient

from .index_cfg import IndexCfg, METRIC_INNER_PRODUCT, METRIC_L2  # noqa: F401
from .index_state import IndexState  # noqa: F401
from .index import Index  # noqa: F401
from .server import IndexServer, register_inproc_server  # noqa: F401
from .client import IndexClient  # noqa: F401

__version__ = "0.1.0"
