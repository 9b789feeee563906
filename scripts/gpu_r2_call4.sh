#!/bin/bash
# Round-2 GPU call 4: (a) hipGraph-capture-of-RCCL test (gates the
# ROC_GRAPH_MULTI default flip), (b) golden bf16 trajectory to pin GPU
# thresholds, (c) edge-weighted SpMM timing at Reddit scale.
cd "$(dirname "$0")/.." || exit 1
export TMPDIR=/tmp
mkdir -p gpurun_out
S=gpurun_out/r2c4_summary.txt
: > "$S"

echo "== RCCL graph-capture test ==" | tee -a "$S"
timeout 600 python -m pytest tests/test_rccl_gpu.py -q -m gpu --timeout=300 \
  > gpurun_out/r2c4_rccl.log 2>&1
echo "rc=$?" | tee -a "$S"
tail -4 gpurun_out/r2c4_rccl.log | tee -a "$S"

echo "== golden bf16 trajectory ==" | tee -a "$S"
timeout 300 python - 2>&1 <<'EOF' | tee -a "$S"
import sys, json, torch
sys.path.insert(0, "scripts")
from make_golden import train_trajectory
traj = train_trajectory(device="cuda:0", dtype=torch.bfloat16)
print("GOLDEN_BF16", json.dumps(traj))
EOF

echo "== spmm_edge timing (reddit shape, D=256) ==" | tee -a "$S"
timeout 420 python - 2>&1 <<'EOF' | tail -6 | tee -a "$S"
import time, torch
from roc_amd import synthetic_dataset, build_shard, edge_tensor
from roc_amd.ops import functional as F
g, feats, _, _, _ = synthetic_dataset("reddit", seed=1)
sh = build_shard(g, 0, 1).to("cuda:0")
x = feats.to("cuda:0").to(torch.bfloat16)
x = torch.nn.functional.pad(x, (0, 6))  # 602 -> 608
xs = x[:, :256].contiguous()  # D=256 like the hidden layer
w = edge_tensor(sh, init="gcn_norm")
for _ in range(3):
    out = F.scatter_gather_weighted(xs, w, sh)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    out = F.scatter_gather_weighted(xs, w, sh)
torch.cuda.synchronize()
tw = (time.perf_counter() - t0) / 10
for _ in range(3):
    ref = F.scatter_gather(xs, sh, normalize=True)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    ref = F.scatter_gather(xs, sh, normalize=True)
torch.cuda.synchronize()
tu = (time.perf_counter() - t0) / 10
err = (out.float() - ref.float()).abs().max().item()
print(f"spmm_edge D=256: {tw*1e3:.2f} ms  (fused-norm spmm: {tu*1e3:.2f} ms)"
      f"  max|diff|={err:.4f}")
EOF
echo DONE | tee -a "$S"
