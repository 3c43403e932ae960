#!/bin/bash
# round-2 sweep: persistent-scan A/B + nlist tuning at the headline shape.
# Writes gpurun_out/r2s_*.json (one bench JSON line each).
set -x
cd /root/repo
O=gpurun_out

# 0) bit-identity of the persistent scan vs the one-pair kernel (f16 GLUT)
python - << 'EOF' > $O/r2s_pers_check.log 2>&1
import os, sys
sys.path.insert(0, "/root/repo")
import numpy as np, torch
from distributed_faiss_amd.hip_engine import HipEngine
rng = np.random.default_rng(0)
d, n, nq, k, nlist, m = 128, 200_000, 2000, 10, 512, 64
# d % m == 0 -> dsub=2; also try m=16 (dsub=8)
for m in (64, 16):
    xb = rng.standard_normal((n, d), dtype=np.float32)
    q = rng.standard_normal((nq, d), dtype=np.float32)
    spec = {"type": "ivfpq", "dim": d, "metric": 1, "nlist": nlist, "m": m,
            "nbits": 8, "nprobe": 8, "seed": 1, "pq_lut_f16": 1}
    eng = HipEngine(spec=spec)
    eng.train(xb[:50_000]); eng.add(xb)
    os.environ["DFANN_SCAN_PERS"] = "0"
    D0, I0 = eng.search(q, k)
    os.environ["DFANN_SCAN_PERS"] = "1"
    D1, I1 = eng.search(q, k)
    assert np.array_equal(D0, D1) and np.array_equal(I0, I1), f"m={m} mismatch"
    print(f"m={m}: persistent scan bit-identical OK")
EOF
echo "PERS_CHECK_RC=$?"

B="python bench.py --steps 20 --warmup 5 --cpu-baseline 0"
# 1) persistent off (baseline with lut/scan split timing)
DFANN_SCAN_PERS=0 $B > $O/r2s_pers0.json 2> $O/r2s_pers0.log; echo RC=$?
# 2) persistent on (default blocks=2048)
$B > $O/r2s_pers1.json 2> $O/r2s_pers1.log; echo RC=$?
# 3) persistent blocks sweep
DFANN_SCAN_PERS_BLOCKS=1280 $B > $O/r2s_pb1280.json 2> $O/r2s_pb1280.log; echo RC=$?
DFANN_SCAN_PERS_BLOCKS=4096 $B > $O/r2s_pb4096.json 2> $O/r2s_pb4096.log; echo RC=$?
# 4) nlist tuning at the headline shape (engine tuning; same index family)
$B --nlist 32768 > $O/r2s_nl32k.json 2> $O/r2s_nl32k.log; echo RC=$?
$B --nlist 16384 > $O/r2s_nl16k.json 2> $O/r2s_nl16k.log; echo RC=$?
for f in $O/r2s_*.json; do echo "== $f"; python - "$f" << 'EOF'
import json, sys
try:
    d = json.load(open(sys.argv[1]))
    r = d["roofline"]
    print(round(d["value"]), "qps", round(d["ms_per_step"], 2), "ms/step",
          "nprobe", d["config"]["nprobe"], "recall", round(d["config"]["recall_at_10"], 4),
          "| scan", round(r["scan_ms"]/max(d["steps"],1)/3*d["steps"]/d["steps"], 3) if False else round(r["scan_ms"], 2),
          "lut", round(r["lut_ms"], 2), "gemm", round(r["gemm_ms"], 2),
          "merge", round(r["merge_ms"], 2), "| achieved", round(r["achieved"]), "GB/s")
except Exception as e:
    print("unparsed:", e)
EOF
done
