#!/bin/bash
# round-2 final GPU sweep: full gpu suite + smoke + headline benches + profile
set -x
mkdir -p gpurun_out
timeout 900 python -m pytest tests -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu rc=$?" | tee gpurun_out/sweep_rc.txt
timeout 420 python -c 'import __graft_entry__ as g; g.build(); g.smoke(); print("smoke OK")' >> gpurun_out/sweep_rc.txt 2>&1
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_30b.json 2> gpurun_out/bench_30b.err
echo "bench30b rc=$?" >> gpurun_out/sweep_rc.txt
timeout 600 python bench.py --model llama3-8b --steps 10 --warmup 3 > gpurun_out/bench_llama.json 2> gpurun_out/bench_llama.err
echo "llama rc=$?" >> gpurun_out/sweep_rc.txt
timeout 600 python bench.py --model qwen25-vl-7b --batch 2 --steps 8 --warmup 2 > gpurun_out/bench_vl2.json 2> gpurun_out/bench_vl2.err
echo "vl2 rc=$?" >> gpurun_out/sweep_rc.txt
timeout 600 python bench.py --model qwen25-vl-7b --batch 4 --steps 8 --warmup 2 > gpurun_out/bench_vl4.json 2> gpurun_out/bench_vl4.err
echo "vl4 rc=$?" >> gpurun_out/sweep_rc.txt
cd /tmp && export TMPDIR=/tmp
timeout 700 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/prof_r02 -o r02step -- \
  python $GRAFT_REPO_ROOT/bench.py --steps 3 --warmup 1 --no-cpu-baseline \
  > $GRAFT_REPO_ROOT/gpurun_out/bench_30b_prof.log 2>&1
echo "rocprof rc=$?" >> $GRAFT_REPO_ROOT/gpurun_out/sweep_rc.txt
tail -3 $GRAFT_REPO_ROOT/gpurun_out/pytest_gpu.log
cat $GRAFT_REPO_ROOT/gpurun_out/sweep_rc.txt
