#!/bin/bash
# PMC passes on the CURRENT dkv v6 kernel (post-shfl): where do wave cycles
# go. Counter names per MI355X_MICROARCH.md (SQ 8 slots; no trace domains).
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd /tmp && export TMPDIR=/tmp
run_pmc () {
  timeout 300 rocprofv3 --pmc $1 --output-format csv -d $R/gpurun_out/pmc_$2 -o p \
    -- python $R/tests/gpu_attn_pmc.py > $R/gpurun_out/pmc_$2.log 2>&1
  echo "$2 rc=$?"
}
run_pmc "SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAIT_INST_LDS SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_MFMA SQ_BUSY_CYCLES" sq
run_pmc "SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE FETCH_SIZE" lds
for t in sq lds; do
  echo "=== $t ==="
  f=$(ls $R/gpurun_out/pmc_$t/*.csv 2>/dev/null | head -1)
  head -1 "$f"
  grep -a "k_attn_bwd_dkv_g" "$f" | head -4
done
du -sh $R/gpurun_out
