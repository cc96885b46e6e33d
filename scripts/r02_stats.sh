#!/bin/bash
# per-kernel step stats (csv) for profiles/ — 30B, llama, VL workloads
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
for m in qwen3-moe-30b llama3-8b qwen25-vl-7b; do
  timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d $R/gpurun_out/prof_$m -o step -- \
    python $R/bench.py --model $m --steps 2 --warmup 1 --no-cpu-baseline \
    > $R/gpurun_out/prof_$m.log 2>&1
  echo "$m rc=$?"
  find $R/gpurun_out/prof_$m -type f ! -name '*kernel_stats.csv' -delete
done
du -sh $R/gpurun_out
for m in qwen3-moe-30b llama3-8b qwen25-vl-7b; do
  echo "=== $m ==="; tail -2 $R/gpurun_out/prof_$m.log | head -1
  head -12 $R/gpurun_out/prof_$m/*kernel_stats.csv
done
