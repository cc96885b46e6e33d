#!/bin/bash
# VL re-measure (segment-batched ViT attention) + the step kernel-stats
# profile for profiles/. Keep gpurun_out well under the 64 MiB pull limit.
set -x
mkdir -p gpurun_out
timeout 600 python bench.py --model qwen25-vl-7b --steps 6 --warmup 2 > gpurun_out/bench_vl1.json 2> gpurun_out/bench_vl1.err
echo "vl1 rc=$?"
timeout 600 python bench.py --model qwen25-vl-7b --batch 4 --steps 6 --warmup 2 > gpurun_out/bench_vl4.json 2> gpurun_out/bench_vl4.err
echo "vl4 rc=$?"; tail -3 gpurun_out/bench_vl4.err
timeout 600 python bench.py --model qwen3-vl-moe-30b --steps 4 --warmup 2 --no-cpu-baseline > gpurun_out/bench_vlmoe.json 2> gpurun_out/bench_vlmoe.err
echo "vlmoe rc=$?"; tail -3 gpurun_out/bench_vlmoe.err
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_r02 -o r02step -- \
  python $GRAFT_REPO_ROOT/bench.py --steps 2 --warmup 1 --no-cpu-baseline \
  > $GRAFT_REPO_ROOT/gpurun_out/bench_30b_prof.log 2>&1
echo "rocprof rc=$?"
find $GRAFT_REPO_ROOT/gpurun_out/prof_r02 -type f ! -name '*stats*.csv' -delete
du -sh $GRAFT_REPO_ROOT/gpurun_out $GRAFT_REPO_ROOT/gpurun_out/*
echo "=== bench lines ==="
for f in bench_vl1 bench_vl4 bench_vlmoe; do echo "-- $f"; tail -c 1400 $GRAFT_REPO_ROOT/gpurun_out/$f.json; echo; done
echo "=== top kernels ==="
head -25 $GRAFT_REPO_ROOT/gpurun_out/prof_r02/*kernel_stats.csv
