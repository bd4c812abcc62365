#!/bin/bash
# rocprofv3 kernel stats of the final r2 configs (run under gpurun).
set -x
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out/prof2
rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof2 -o resnet_b1024 -- \
  timeout 400 python bench.py --gpus 1 --steps 4 --warmup 2 \
  > gpurun_out/prof2/resnet.log 2>&1
rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof2 -o bert_b32 -- \
  timeout 400 python tools/bench_nlp.py --model bert_base --steps 4 --warmup 2 \
  > gpurun_out/prof2/bert.log 2>&1
rm -f gpurun_out/prof2/*.db gpurun_out/prof2/*kernel_trace*
ls -la gpurun_out/prof2/
tail -1 gpurun_out/prof2/resnet.log
tail -1 gpurun_out/prof2/bert.log
