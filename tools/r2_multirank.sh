#!/usr/bin/env bash
# Multi-rank RCCL evidence on a single leased GPU:
# 2 and 4 ranks share cuda:0 via the engine's modulo device mapping,
# exercising real RCCL all-reduce + the side comm stream + bucket overlap.
# NOT a scaling claim — protocol/overhead rehearsal only (the 8-GPU curve
# is the driver's SCALE_rNN.json).
set -uo pipefail
mkdir -p gpurun_out

run_bench () {
    local n=$1 tag=$2
    echo "== bench.py --gpus $n ($tag) =="
    timeout 420 python -m torch.distributed.run --nnodes=1 \
        --nproc-per-node "$n" --master-addr 127.0.0.1 --master-port 29641 \
        bench.py --gpus "$n" --steps 30 --warmup 5 \
        > "gpurun_out/r2_multirank_n${n}.json" 2> "gpurun_out/r2_multirank_n${n}.log"
    echo "rc=$? ; json:"
    tail -1 "gpurun_out/r2_multirank_n${n}.json" || true
}

run_bench 2 "2 ranks on 1 GPU"
run_bench 4 "4 ranks on 1 GPU"

# kernel-trace evidence that RCCL kernels run alongside our conv/bwd
# kernels (overlap): stats-only profile of the 2-rank bench.
echo "== rocprof kernel stats of the 2-rank run =="
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --stats -d gpurun_out/prof_multirank -o mr -- \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29643 \
    bench.py --gpus 2 --steps 10 --warmup 3 \
    > gpurun_out/r2_multirank_prof.log 2>&1 || true
grep -ril "rccl\|AllReduce" gpurun_out/prof_multirank/ | head -3 || true
for f in $(find gpurun_out/prof_multirank -name '*stats*.csv' | head -4); do
    echo "--- $f"
    head -15 "$f"
done
