#!/bin/bash
# post-TR step kernel stats (csv) + first successful PMC on nk256s
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd /tmp && export TMPDIR=/tmp
for m in qwen3-moe-30b llama3-8b qwen25-vl-7b; do
  timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d $R/gpurun_out/prof_$m -o step -- \
    python $R/bench.py --model $m --steps 2 --warmup 1 --no-cpu-baseline \
    > $R/gpurun_out/prof_$m.log 2>&1
  echo "$m rc=$?"
  find $R/gpurun_out/prof_$m -type f ! -name '*kernel_stats.csv' -delete
done
timeout 300 rocprofv3 --pmc "SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_MFMA SQ_ACTIVE_INST_ANY SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT" \
  --output-format csv -d $R/gpurun_out/pmc_gg -o p \
  -- python $R/tests/gpu_gg_traffic.py > $R/gpurun_out/pmc_gg.log 2>&1
echo "gg pmc rc=$?"
du -sh $R/gpurun_out
for m in qwen3-moe-30b llama3-8b qwen25-vl-7b; do
  echo "=== $m ==="; head -6 $R/gpurun_out/prof_$m/*kernel_stats.csv | cut -c1-170
done
