#!/bin/bash
# round-2 bench re-run (first attempt's outputs were lost to the 64 MiB
# copy-back limit — rocprof kernel_trace.csv). Keep gpurun_out small.
set -x
mkdir -p gpurun_out
timeout 120 python -m pytest tests/test_gpu_kernels.py::test_ce_lowmem_matches_stash -x -q 2>&1 | tail -2
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_30b.json 2> gpurun_out/bench_30b.err
timeout 600 python bench.py --model llama3-8b --steps 10 --warmup 3 > gpurun_out/bench_llama.json 2> gpurun_out/bench_llama.err
timeout 600 python bench.py --model qwen25-vl-7b --batch 2 --steps 8 --warmup 2 > gpurun_out/bench_vl2.json 2> gpurun_out/bench_vl2.err
timeout 600 python bench.py --model qwen25-vl-7b --batch 4 --steps 8 --warmup 2 > gpurun_out/bench_vl4.json 2> gpurun_out/bench_vl4.err
echo "=== vl4 err tail ==="; tail -6 gpurun_out/bench_vl4.err
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_r02 -o r02step -- \
  python $GRAFT_REPO_ROOT/bench.py --steps 2 --warmup 1 --no-cpu-baseline \
  > $GRAFT_REPO_ROOT/gpurun_out/bench_30b_prof.log 2>&1
# the raw trace is huge; only the stats summaries travel back
find $GRAFT_REPO_ROOT/gpurun_out/prof_r02 -name '*kernel_trace*' -delete
du -sh $GRAFT_REPO_ROOT/gpurun_out
echo "=== bench lines ==="
for f in bench_30b bench_llama bench_vl2 bench_vl4; do echo "-- $f"; tail -c 1500 $GRAFT_REPO_ROOT/gpurun_out/$f.json; echo; done
