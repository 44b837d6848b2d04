#!/bin/bash
# PMC counter collection for the kbench kernels (run on a GPU box).
export TMPDIR=/tmp
cd /tmp
exec rocprofv3 --pmc VALUBusy MeanOccupancyPerCU FetchSize WriteSize \
  LDSBankConflict SQ_VALU_MFMA_BUSY_CYCLES \
  -d /root/repo/gpurun_out/pmc_mfma -- bash /root/repo/tools/kb.sh "${1:-5000}" kfb0
