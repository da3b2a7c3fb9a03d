#!/usr/bin/env bash
# Ring comm/compute overlap evidence on a MULTI-GPU MI355X node
# (north_star: >=90% of each ring step's K/V transfer hidden under the
# local tile kernel).  Two independent measurements:
#
# 1. bench.py's built-in probe (runs unattended in any N>1 launch,
#    incl. the round-end driver's 1/2/4/8 sweep): the printed JSON line
#    carries `ring: {compute_only_ms, comm_only_ms, exposed_comm_ms,
#    overlap_frac}` where exposed = full_step - compute_only and
#    overlap_frac = 1 - exposed/comm_only.
#
# 2. A kernel+RCCL trace of a few ring steps for the rocprof-level view
#    (kernel-trace only — rocprofv3 refuses --pmc combined with the
#    trace domains on this pool).  Overlap shows as RCCL kernels
#    (ncclDevKernel*) time-sharing the GPU with attn_fwd_kernel.
#
# Usage (8-GPU box):  bash tools/overlap_trace.sh 8
set -e
N=${1:-8}
cd /tmp && export TMPDIR=/tmp
OUT=${GRAFT_REPO_ROOT:-/root/repo}/gpurun_out/overlap_n$N
mkdir -p "$OUT"
rocprofv3 --kernel-trace --stats -d "$OUT" -o trace -- \
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29555 \
    ${GRAFT_REPO_ROOT:-/root/repo}/bench.py \
    --gpus "$N" --steps 2 --warmup 1 --seq 262144 --no-cpu-baseline \
  | tee "$OUT/bench.json"
echo "trace + bench line under $OUT; overlap_frac is in the JSON 'ring' key"
