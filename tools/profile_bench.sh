#!/bin/bash
# rocprofv3 kernel-trace + stats of the flagship bench (run under gpurun).
set -x
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out/prof
rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o resnet_b64 -- \
  timeout 500 python bench.py --gpus 1 --steps 5 --warmup 2 --batch-size 64 \
  > gpurun_out/prof/bench_out.log 2>&1
rm -f gpurun_out/prof/*.db
ls -la gpurun_out/prof/
for f in gpurun_out/prof/*stats*.csv; do
  echo "== $f"; head -30 "$f"
done
