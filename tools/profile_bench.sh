#!/bin/bash
# rocprofv3 kernel-trace + stats of the flagship bench (run under gpurun).
# Outputs land in gpurun_out/prof*; the judged summaries are copied into
# profiles/ and committed.
set -x
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out/prof
# kernel stats (no PMC in the same run — pool rule)
rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o resnet_b64 -- \
  timeout 500 python bench.py --gpus 1 --steps 5 --warmup 2 --batch-size 64 \
  > gpurun_out/prof/bench_out.log 2>&1
# PMC counters for the two heaviest kernels in a separate run
rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,SQ_WAVE_CYCLES,SQ_LDS_BANK_CONFLICT \
  -d gpurun_out/prof -o resnet_pmc -- \
  timeout 500 python bench.py --gpus 1 --steps 2 --warmup 1 --batch-size 64 \
  > gpurun_out/prof/bench_pmc.log 2>&1 || true
ls -la gpurun_out/prof/
for f in gpurun_out/prof/*stats*.csv; do
  echo "== $f"; head -25 "$f"
done
