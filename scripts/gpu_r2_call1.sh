#!/bin/bash
# Round-2 GPU call 1: RCCL-on-hardware validation.
# - full `pytest -m gpu` suite (incl. new tests/test_rccl_gpu.py)
# - ws=2-on-1-GPU bench over RCCL: eager vs hipGraph-multi vs overlap
cd "$(dirname "$0")/.." || exit 1
export TMPDIR=/tmp
mkdir -p gpurun_out
S=gpurun_out/r2c1_summary.txt
: > "$S"

echo "== pytest -m gpu ==" | tee -a "$S"
timeout 900 python -m pytest tests -q -m gpu --timeout=300 \
  > gpurun_out/r2c1_gputests.log 2>&1
echo "pytest rc=$?" | tee -a "$S"
tail -5 gpurun_out/r2c1_gputests.log | tee -a "$S"

# ws=2 both ranks on the one GPU, RCCL backend; small steps first
run_ws2 () {
  name="$1"; shift
  echo "== bench ws2 $name ==" | tee -a "$S"
  timeout 300 env ROC_DEVICE_OVERRIDE=0 "$@" \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29650 \
    bench.py --gpus 2 --steps 30 --warmup 4 --exact-steps \
    > "gpurun_out/r2c1_ws2_${name}.log" 2>&1
  echo "ws2 $name rc=$?" | tee -a "$S"
  grep -h '"metric"' "gpurun_out/r2c1_ws2_${name}.log" | tee -a "$S"
  tail -3 "gpurun_out/r2c1_ws2_${name}.log" >> "$S"
}

run_ws2 halo_eager            env ROC_COMM_MODE=halo
run_ws2 halo_overlap          env ROC_COMM_MODE=halo ROC_OVERLAP=1
run_ws2 halo_graph            env ROC_COMM_MODE=halo ROC_GRAPH_MULTI=1
run_ws2 ag_eager              env ROC_COMM_MODE=allgather ROC_AG_OVERLAP=0
run_ws2 ag_overlap            env ROC_COMM_MODE=allgather ROC_AG_OVERLAP=1
run_ws2 ag_graph              env ROC_COMM_MODE=allgather ROC_GRAPH_MULTI=1

echo "== bench ws1 (auto-extend check) ==" | tee -a "$S"
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 4 \
  > gpurun_out/r2c1_ws1.log 2>&1
echo "ws1 rc=$?" | tee -a "$S"
grep -h '"metric"' gpurun_out/r2c1_ws1.log | tee -a "$S"
tail -2 gpurun_out/r2c1_ws1.log >> "$S"
echo DONE | tee -a "$S"
