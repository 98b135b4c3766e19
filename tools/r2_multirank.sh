#!/usr/bin/env bash
# RCCL evidence on a single leased GPU.
#
# FINDING (gpurun_out/r2_multirank_n2.log, r2): RCCL/NCCL 2.26 REFUSES two
# ranks on one device — "Duplicate GPU detected : rank 0 and rank 1 both on
# CUDA device" — so the literal 2-rank-RCCL-on-1-GPU rehearsal VERDICT
# suggested is impossible by design. Instead this script produces:
#   (a) a 1-rank RCCL process group under torchrun with the engine's
#       collectives FORCED ON (COINN_FORCE_ALLREDUCE=1): real ncclAllReduce
#       kernels on the side comm stream, traced by rocprofv3 interleaved
#       with our backward kernels — validates stream/event wiring with the
#       real backend;
#   (b) 2- and 4-rank runs with COINN_DIST_BACKEND=gloo on the SAME GPU:
#       full multi-rank lock-step protocol + bucket machinery with real
#       HIP compute per rank (protocol-overhead rehearsal, NOT a scaling
#       measurement — the 8-GPU RCCL curve is the driver's SCALE_rNN.json).
set -uo pipefail
mkdir -p gpurun_out

echo "== (a) 1-rank RCCL group, forced collectives, rocprof kernel stats =="
export TMPDIR=/tmp
( cd /tmp && true )
COINN_FORCE_ALLREDUCE=1 timeout 600 rocprofv3 --stats --kernel-trace \
    -d gpurun_out/prof_rccl1 -o rccl1 -- \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
    --master-addr 127.0.0.1 --master-port 29641 \
    bench.py --gpus 1 --steps 10 --warmup 3 \
    > gpurun_out/r2_rccl1.json 2> gpurun_out/r2_rccl1.log
echo "rc=$?"
echo "-- rccl kernels in the trace:"
grep -ih "rccl\|AllReduce\|ncclDevKernel" \
    $(find gpurun_out/prof_rccl1 -name '*stats*.csv' 2>/dev/null) | head -5

for n in 2 4; do
    echo "== (b) bench.py --gpus $n, gloo backend, one shared GPU =="
    COINN_DIST_BACKEND=gloo timeout 420 python -m torch.distributed.run \
        --nnodes=1 --nproc-per-node "$n" --master-addr 127.0.0.1 \
        --master-port 2964$n bench.py --gpus "$n" --steps 20 --warmup 5 \
        > "gpurun_out/r2_multirank_gloo_n${n}.json" \
        2> "gpurun_out/r2_multirank_gloo_n${n}.log"
    echo "rc=$? ; json:"
    grep -o '{"metric.*' "gpurun_out/r2_multirank_gloo_n${n}.json" | tail -1
done
