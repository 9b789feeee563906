#!/bin/bash
# Round-2 GPU call 7: (a) instrumented RCCL-in-hipGraph capture test
# (faulthandler native stack on crash), (b) eager ws1 bench timing,
# (c) cProfile of eager epochs — quantifies the Python overhead the
# multi-GPU eager path would pay if capture can't include RCCL.
cd "$(dirname "$0")/.." || exit 1
export TMPDIR=/tmp
mkdir -p gpurun_out
S=gpurun_out/r2c7_summary.txt
: > "$S"

echo "== capture test (instrumented) ==" | tee -a "$S"
timeout 400 python -m pytest tests/test_rccl_gpu.py::test_rccl_graph_capture_ws1 \
  -q -m gpu --timeout=350 -s > gpurun_out/r2c7_capture.log 2>&1
echo "rc=$?" | tee -a "$S"
grep -E "stages|exitcode|Fatal|SIG|Thread|passed|failed|rccl|hip" \
  gpurun_out/r2c7_capture.log | head -25 | tee -a "$S"

echo "== eager ws1 bench ==" | tee -a "$S"
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 3 --exact-steps \
  --no-graph > gpurun_out/r2c7_eager.log 2>&1
echo "rc=$?" | tee -a "$S"
grep -h '"metric"' gpurun_out/r2c7_eager.log | tee -a "$S"

echo "== cProfile eager epochs ==" | tee -a "$S"
timeout 420 python - 2>&1 <<'EOF' | tail -30 | tee -a "$S"
import cProfile, io, pstats, time, torch, os
os.environ.setdefault("ROC_BENCH_CACHE", "/tmp")
from roc_amd import synthetic_dataset, build_model, AdamOptimizer, Trainer, build_shard
g, feats, labels, mask, c = synthetic_dataset("reddit", seed=1)
sh = build_shard(g, 0, 1)
import torch.nn.functional as tnf
feats = tnf.pad(feats, (0, 6))
dims = [608, 256, 64]
model = build_model("gcn", dims, dropout=0.5, seed=1)
opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
tr = Trainer(model, sh, feats, labels, mask, opt, device="cuda:0",
             compute_dtype=torch.bfloat16, grad_scale=1.0, num_classes=41)
for _ in range(3):
    tr.train_epoch()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    tr.train_epoch()
torch.cuda.synchronize()
print(f"eager epoch: {(time.perf_counter()-t0)/10*1e3:.2f} ms")
# CPU-side cost: time the launch phase WITHOUT sync (queue-ahead depth)
t0 = time.perf_counter()
for _ in range(10):
    tr.train_epoch()
cpu_side = (time.perf_counter() - t0) / 10 * 1e3
torch.cuda.synchronize()
print(f"python/launch per epoch (no sync): {cpu_side:.2f} ms")
pr = cProfile.Profile()
pr.enable()
for _ in range(10):
    tr.train_epoch()
pr.disable()
torch.cuda.synchronize()
s = io.StringIO()
pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(18)
print(s.getvalue())
EOF
echo DONE | tee -a "$S"
