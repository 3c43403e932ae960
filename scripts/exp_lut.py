# A/B the GLUT scan knobs on ONE built index (build once, sweep
# DFANN_PQ_LUT_MB / DFANN_PQ_LUT_GLOBAL env overrides). Experiment
# harness only — bench.py stays the contract entry point.
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bench import WORKLOADS, gen_shard  # noqa: E402
from distributed_faiss_amd.hip_engine import HipEngine  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="ivfpq_100m8_d768_m64")
    ap.add_argument("--nprobe", type=int, default=8)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--settings", default="off;96;32;256;512",
                    help="';'-separated: 'off' or lut_mb values")
    args = ap.parse_args()
    cfg = WORKLOADS[args.config]
    k = cfg["k"]
    xb, q = gen_shard(cfg, 0, "cuda")
    spec = {"type": cfg["type"], "dim": cfg["d"], "metric": cfg["metric"],
            "nlist": cfg["nlist"], "m": cfg["m"], "nbits": cfg["nbits"],
            "sq_type": cfg.get("sq_type", "fp16"), "nprobe": args.nprobe,
            "seed": 1234, "coarse_bf16": cfg.get("coarse_bf16", 0),
            "max_ppc": cfg.get("max_ppc", 256),
            "ws_mb": cfg.get("ws_mb", 512)}
    eng = HipEngine(spec=spec)
    t0 = time.time()
    eng.train_dev(xb)
    eng.add_dev(xb)
    torch.cuda.synchronize()
    print(f"[exp] built in {time.time()-t0:.1f}s", file=sys.stderr)
    eng.nprobe = args.nprobe
    qb = q.contiguous()
    nq = qb.shape[0]
    if os.environ.get("EXP_SKEW"):
        import numpy as np
        off, _, _ = eng.get_lists()
        lens = np.diff(off)
        probes, _ = eng.coarse(qb.cpu().numpy(), args.nprobe)
        plens = lens[probes.reshape(-1)]
        tot = plens.sum()
        print(json.dumps({
            "list_len": {"max": int(lens.max()), "mean": float(lens.mean()),
                         "p99": int(np.percentile(lens, 99)),
                         "p999": int(np.percentile(lens, 99.9)),
                         "top10": np.sort(lens)[-10:].tolist()},
            "probed": {"rows_per_step": int(tot),
                       "max_block_rows": int(plens.max()),
                       "frac_rows_in_gt2048":
                           float(plens[plens > 2048].sum() / tot),
                       "frac_rows_in_gt8192":
                           float(plens[plens > 8192].sum() / tot),
                       "p99_block_rows": int(np.percentile(plens, 99))},
        }), file=sys.stderr, flush=True)
    for s in args.settings.split(";"):
        # tokens: "off" | "<mb>" (f32 GLUT) | "f16" | "f16:<mb>" | "bs:<n>"
        if s.startswith("bs:"):
            os.environ["DFANN_SCAN_BS"] = s.split(":", 1)[1]
            s = "off"
        if s == "off":
            os.environ["DFANN_PQ_LUT_GLOBAL"] = "0"
            os.environ["DFANN_PQ_LUT_F16"] = "0"
        elif s.startswith("f16"):
            os.environ["DFANN_PQ_LUT_GLOBAL"] = "-1"
            os.environ["DFANN_PQ_LUT_F16"] = "1"
            os.environ["DFANN_PQ_LUT_MB"] = (s.split(":", 1)[1]
                                             if ":" in s else "2048")
        else:
            os.environ["DFANN_PQ_LUT_GLOBAL"] = "1"
            os.environ["DFANN_PQ_LUT_F16"] = "0"
            os.environ["DFANN_PQ_LUT_MB"] = s
        for _ in range(args.warmup):
            eng.search_dev(qb, k)
        torch.cuda.synchronize()
        base = eng.get_timing()  # events accumulate; report deltas
        eng.set_timing(True)
        t0 = time.time()
        for _ in range(args.steps):
            eng.search_dev(qb, k)
        torch.cuda.synchronize()
        dt = time.time() - t0
        tm = eng.get_timing()
        eng.set_timing(False)
        print(json.dumps({
            "setting": s, "qps": nq * args.steps / dt,
            "ms_per_step": dt / args.steps * 1e3,
            "scan_ms_per_step": (tm["scan_ms"] - base["scan_ms"]) / args.steps,
            "gemm_ms_per_step": (tm["gemm_ms"] - base["gemm_ms"]) / args.steps,
            "merge_ms_per_step":
                (tm["merge_ms"] - base["merge_ms"]) / args.steps,
        }))


if __name__ == "__main__":
    main()
