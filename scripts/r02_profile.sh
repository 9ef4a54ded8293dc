#!/bin/bash
# Round-2 profiling evidence: per-kernel stats for the flagship bench and
# PMC counters for the forward-FFT column passes (separate runs; counters
# never combined with trace domains per pool rules).
set -x
export TMPDIR=/tmp
cd /tmp
REPO=/root/repo
mkdir -p $REPO/gpurun_out/prof

# 1. kernel stats over the honest flagship bench
timeout 600 rocprofv3 --stats -d $REPO/gpurun_out/prof/bench -- \
  python $REPO/bench.py --steps 3 --warmup 1 --blocks-per-step 2 \
  > $REPO/gpurun_out/prof/bench_stats.log 2>&1
tail -3 $REPO/gpurun_out/prof/bench_stats.log

# 2. PMC pass A: fetch + wave/wait cycles on the fwd 2^29 FFT
cat > /tmp/fftbench.py <<'PY'
import sys
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch
C = native(); torch.cuda.set_device(0)
print("fwd29", C.bench_fft(1 << 29, 1, -1, 10, "native"), "ms")
print("bwd18", C.bench_fft(1 << 18, 2048, 1, 10, "native"), "ms")
PY
timeout 600 rocprofv3 --pmc FETCH_SIZE SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT \
  --kernel-trace -d $REPO/gpurun_out/prof/pmcA -- python /tmp/fftbench.py \
  > $REPO/gpurun_out/prof/pmcA.log 2>&1
tail -3 $REPO/gpurun_out/prof/pmcA.log

# 3. PMC pass B: write size
timeout 600 rocprofv3 --pmc WRITE_SIZE SQ_BUSY_CYCLES \
  --kernel-trace -d $REPO/gpurun_out/prof/pmcB -- python /tmp/fftbench.py \
  > $REPO/gpurun_out/prof/pmcB.log 2>&1
tail -3 $REPO/gpurun_out/prof/pmcB.log

ls -R $REPO/gpurun_out/prof | head -40
