# Multi-process shard merge over torch.distributed (gloo, CPU, world 2) —
# covers the collective path of distributed_faiss_amd/dist.py that the
# 8-GPU bench uses with RCCL. The sharded result must equal a single
# oracle index over the union of the shards (the reference's
# sharded==single invariant, tests/test_integration.py:205-265, here at
# the engine level).
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker(rank, world, tmpdir):
    import torch
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    from distributed_faiss_amd.dist import allgather_shard_topk, merge_gathered
    from oracle import make_oracle_engine

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29581"
    dist.init_process_group("gloo", rank=rank, world_size=world)

    d, n_shard, nq, k = 16, 300, 7, 5
    rng = np.random.default_rng(100 + rank)
    xb = rng.standard_normal((n_shard, d), dtype=np.float32)
    q = np.random.default_rng(999).standard_normal((nq, d), dtype=np.float32)

    for metric, maximize in ((1, False), (0, True)):
        eng = make_oracle_engine({"type": "flat", "dim": d, "metric": metric})
        eng.add(xb)
        D, I = eng.search(q, k)
        Dall, Iall = allgather_shard_topk(torch.from_numpy(D), torch.from_numpy(I))
        Dm, s_idx, local = merge_gathered(Dall, Iall, k, maximize)

        # single-index truth on rank 0
        if rank == 0:
            xb_all = np.concatenate([
                np.random.default_rng(100 + r).standard_normal(
                    (n_shard, d), dtype=np.float32) for r in range(world)
            ])
            ref = make_oracle_engine({"type": "flat", "dim": d, "metric": metric})
            ref.add(xb_all)
            Dr, Ir = ref.search(q, k)
            # merged distances: negated for dot (reference quirk 2)
            np.testing.assert_allclose(Dm, -Dr if maximize else Dr,
                                       rtol=1e-5, atol=1e-5)
            # winner identity: shard*n_shard + local == global arrival id
            gids = s_idx * n_shard + local
            np.testing.assert_array_equal(gids, Ir)
    dist.destroy_process_group()


def test_gloo_shard_merge_world2(tmp_path):
    torch = pytest.importorskip("torch")
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0
