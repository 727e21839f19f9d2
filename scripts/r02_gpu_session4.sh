#!/bin/bash
# r02 GPU session 4: validate auto long-row dispatch + bench headline;
# cgemm wide-tile A/B (cfg5 rmatvec); MDC chain after the twin-scale
# cancellation; PMC traffic for the rolling kernel (weak-scaling dims).
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
OUT=gpurun_out/r02d
mkdir -p $OUT

echo "== 1. parity with the auto dispatch (rolling kicks in on big m) =="
timeout 900 python -m pytest tests/test_gpu_parity.py tests/test_gpu_fullsize.py \
    -x -q -m gpu 2>&1 | tail -3

echo "== 2. shapes under the new default =="
SW="timeout 120 python scripts/gpu_fd_shape_sweep.py"
DIMS=512x4096x256  $SW 2>&1 | tail -1
DIMS=2048x2048x128 $SW 2>&1 | tail -1
DIMS=1024x4096x256 $SW 2>&1 | tail -1
DIMS=4096x4096x256 $SW 2>&1 | tail -1

echo "== 3. bench.py N=1 defaults =="
timeout 600 python bench.py --gpus 1 --steps 50 --warmup 5 \
    > $OUT/bench_n1.json 2> $OUT/bench_n1.err
tail -2 $OUT/bench_n1.json

echo "== 4. cgemm wide-tile A/B at cfg5 =="
timeout 300 python scripts/gpu_fred_probe.py 2>&1 | tail -3
PAM_CGEMM_TILE=2 timeout 300 python scripts/gpu_fred_probe.py 2>&1 | tail -3

echo "== 5. MDC chain =="
timeout 300 python scripts/gpu_mdc_probe.py 2>&1 | tail -2

echo "== 6. PMC traffic, rolling kernel at the long-row shape =="
cd /tmp && export TMPDIR=/tmp
R=/root/repo
DIMS=512x4096x256 timeout 300 rocprofv3 --pmc FETCH_SIZE \
    --output-format csv -d $R/$OUT/pmc_fetch -o fetch -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
DIMS=512x4096x256 timeout 300 rocprofv3 --pmc WRITE_SIZE \
    --output-format csv -d $R/$OUT/pmc_write -o write -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
cd $R
python scripts/extract_traffic.py \
    $OUT/pmc_fetch/fetch_counter_collection.csv \
    $OUT/pmc_write/write_counter_collection.csv 2>&1 | tail -8
echo "== 7. kernel stats trace at long shape (for profiles/) =="
cd /tmp
DIMS=512x4096x256 timeout 300 rocprofv3 --kernel-trace --stats \
    --output-format csv -d $R/$OUT/ktrace -o ktrace -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -3
cd $R
echo DONE
