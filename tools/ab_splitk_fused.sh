#!/bin/bash
# A/B the fused split-K reduce (SCANNER_SPLITK_FUSED) on GPU:
# 1. full GPU test suite with the fused default ON (numerics + hipgraph
#    run-to-run determinism cover the new path)
# 2. flagship + resnet bench, fused vs unfused, same box
set -x
mkdir -p gpurun_out
cd /root/repo

timeout 300 python -m pytest tests/test_resnet_gpu.py tests/test_gpu.py -x -q \
  > gpurun_out/ab_fused_tests.log 2>&1
echo "tests rc=$?" >> gpurun_out/ab_fused_tests.log

run_bench() {  # $1=pipeline $2=tag $3=fused
  SCANNER_SPLITK_FUSED=$3 timeout 240 python bench.py --pipeline $1 \
    --steps 3 --warmup 1 > gpurun_out/ab_${2}.json 2> gpurun_out/ab_${2}.err
}
run_bench full   full_fused1 1
run_bench full   full_fused0 0
run_bench full   full_fused1b 1
run_bench resnet rn_fused1 1
run_bench resnet rn_fused0 0
tail -1 gpurun_out/ab_full_fused1.json gpurun_out/ab_full_fused0.json \
  gpurun_out/ab_full_fused1b.json gpurun_out/ab_rn_fused1.json \
  gpurun_out/ab_rn_fused0.json
tail -3 gpurun_out/ab_fused_tests.log
