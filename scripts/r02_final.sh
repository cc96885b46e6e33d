#!/bin/bash
# round-2 final validation sweep: full GPU suite, smoke, headline benches
# (with the cpu_baseline leg on the default 30B run)
set -x
mkdir -p gpurun_out
timeout 900 python -m pytest tests -x -q -m gpu > gpurun_out/final_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/final_pytest.log
timeout 420 python -c 'import __graft_entry__ as g; g.build(); g.smoke(); print("smoke OK")' 2>&1 | tail -2
timeout 900 python bench.py --steps 10 --warmup 3 > gpurun_out/final_30b.json 2> gpurun_out/final_30b.err
echo "30b rc=$?"
timeout 600 python bench.py --model llama3-8b --steps 10 --warmup 3 > gpurun_out/final_llama.json 2> gpurun_out/final_llama.err
echo "llama rc=$?"
timeout 600 python bench.py --model qwen25-vl-7b --steps 8 --warmup 2 --no-cpu-baseline > gpurun_out/final_vl.json 2> gpurun_out/final_vl.err
echo "vl rc=$?"
timeout 600 python bench.py --model qwen3-vl-moe-30b --steps 4 --warmup 2 --no-cpu-baseline > gpurun_out/final_vlmoe.json 2> gpurun_out/final_vlmoe.err
echo "vlmoe rc=$?"
echo "=== bench lines ==="
for f in final_30b final_llama final_vl final_vlmoe; do echo "-- $f"; cat gpurun_out/$f.json; echo; done
du -sh gpurun_out
