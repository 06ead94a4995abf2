#!/bin/bash
# PMC counter capture for the decode GEMM (counters-only run: never
# combine --pmc with trace domains per pool policy)
set -e
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 500 rocprofv3 --pmc SQ_BUSY_CYCLES,SQ_WAIT_ANY,TCC_HIT_sum,TCC_MISS_sum \
  -d gpurun_out/pmc -o gemm_pmc -- \
  python - <<'PYEOF' > gpurun_out/pmc_run.log 2>&1
import torch, time
from mlrun_amd import ops
for name, m, n, k, ks, var in [("wo", 16, 4096, 4096, 2, 2),
                               ("wgu", 16, 28672, 4096, 1, 2)]:
    a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda") * 0.1
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda") * 0.1
    out = torch.empty(m, n, dtype=torch.bfloat16, device="cuda")
    sc = torch.empty(max(ks*m*n,1), dtype=torch.float32, device="cuda")
    for _ in range(10):
        ops.skinny_gemm(a, w, out=out, c_f32=sc, ksplit=ks, variant=var)
    torch.cuda.synchronize()
print("pmc run done")
PYEOF
ls gpurun_out/pmc
