# ORACLE — CPU restatement of the reference hot path's arithmetic.
#
# TEST INFRASTRUCTURE ONLY. Only tests/, __graft_entry__.smoke() and
# bench.py's cpu_baseline leg may import, call or execute anything in this
# package. The product path (distributed_faiss_amd/) never imports it and
# fails loudly when the HIP engine is missing on a GPU box.
#
# What this restates: the arithmetic the reference delegates to the
# un-vendored third-party `faiss-cpu>=1.7.2` wheel (reference setup.py:32),
# at the call sites of distributed_faiss/index.py:
#   - train      (index.py:217)  k-means coarse quantizer, PQ codebooks, SQ ranges
#   - add        (index.py:425)  coarse assign + PQ/SQ encode + list append
#   - search     (index.py:257)  coarse top-nprobe + list scan + top-k
#   - search_and_reconstruct (index.py:255)
#   - reconstruct_n (index.py:350, quantizer centroids)
# and the client merge (reference client.py:29-54, 265-310).
#
# PARITY PINNING STATUS: faiss cannot be imported or built
# in this container and its sources are not vendored under /root/reference,
# so for IVF/IVFPQ/IVFSQ numerical results this oracle is PARITY UNPINNED —
# it restates faiss v1.7's documented algorithms (residual IVFPQ with
# ksub=256 ADC LUT scan; SQ fp16 and 8-bit min/max affine codec; k-means
# with fixed iteration count) and WE generate the golden vectors
# (tests/golden/, script tests/golden/make_golden.py). What IS pinned by
# the reference's own tests and is ported verbatim into tests/:
#   - the merge known-answer test (reference tests/test_integration.py:181-203)
#   - the sharded==single-flat equality invariant
#     (reference tests/test_integration.py:205-265)
#
# Deliberate deviations from faiss (documented in DESIGN.md §oracle):
#   - k-means subsampling is strided, not random (the reference's own
#     training-data order is nondeterministic — index.py:211 shuffles with
#     the unseeded global numpy RNG — so random-subsample equivalence is
#     unobservable through the reference API anyway).
#   - k-means empty-cluster handling: split the largest cluster,
#     deterministic (faiss splits a probabilistically chosen one).
#   - ties everywhere break to the lower id (faiss heap behavior for
#     within-list scans; cross-list tie order in faiss is heap-order
#     dependent and unpinned).
#
# hnswsq tier (round 2): the SEARCH side is restated
# op-for-op (core.OracleHNSWSearch) and runs over the ENGINE'S OWN dumped
# graph — engine search results are bitwise equal to it
# (tests/test_hnsw.py). The BUILD side is NOT restated: the engine's
# batched wave insertion is a documented deviation from faiss's
# sequential insertion (parity unpinned there by construction); the
# build's pinned contracts are determinism (identical graph dumps) and
# recall property gates.

from .core import (  # noqa: F401
    METRIC_INNER_PRODUCT,
    METRIC_L2,
    splitmix64_seq,
    partial_shuffle_indices,
    kmeans,
    OracleFlat,
    OracleIVFFlat,
    OracleIVFPQ,
    OracleIVFSQ,
    make_oracle_engine,
    save_oracle_engine,
    load_oracle_engine,
    OracleProvider,
    aggregate_results,
)
