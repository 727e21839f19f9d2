#!/bin/bash
# r02 GPU session 1: validate new kernels (complex matmult / ctranspose /
# cgemm accumulate / RSWAP), sweep the long-row stencil shape, collect
# PMC traffic for the N>1 weak-scaling dims.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
OUT=gpurun_out/r02a
mkdir -p $OUT

echo "== 1. kernel-change validation =="
timeout 900 python -m pytest tests/test_gpu_matmult.py \
    tests/test_gpu_fredholm.py -x -q -m gpu 2>&1 | tail -4

echo "== 2. RSWAP kernel parity =="
PAM_FD_RSWAP=1 timeout 600 python -m pytest tests/test_gpu_parity.py \
    -x -q -m gpu -k "centered or forward or backward" 2>&1 | tail -3

echo "== 3. long-row sweep (512x4096x256 fp64 matvec) =="
SW="timeout 120 python scripts/gpu_fd_shape_sweep.py"
export DIMS=512x4096x256
$SW                                        2>&1 | tail -1
PAM_FD_CAP=65536   $SW                     2>&1 | tail -1
PAM_FD_CAP=131072  $SW                     2>&1 | tail -1
PAM_FD_CAP=524288  $SW                     2>&1 | tail -1
PAM_FD_GY=64  $SW                          2>&1 | tail -1
PAM_FD_GY=128 $SW                          2>&1 | tail -1
PAM_FD_GY=256 $SW                          2>&1 | tail -1
PAM_FD_NT=1 $SW                            2>&1 | tail -1
PAM_FD_VEC=4 $SW                           2>&1 | tail -1
PAM_FD_RSWAP=1 $SW                         2>&1 | tail -1
PAM_FD_RSWAP=1 PAM_FD_CAP=131072 $SW       2>&1 | tail -1
PAM_FD_RSWAP=1 PAM_FD_CAP=524288 $SW       2>&1 | tail -1
PAM_FD_ROLL=1 PAM_FD_ROLL_TGT=2048 $SW     2>&1 | tail -1
PAM_FD_ROLL=1 PAM_FD_ROLL_TGT=8192 $SW     2>&1 | tail -1
DIMS=2048x2048x128 $SW                     2>&1 | tail -1
DIMS=2048x2048x128 PAM_FD_RSWAP=1 $SW      2>&1 | tail -1

echo "== 4. PMC traffic at the long-row shape (default knobs) =="
cd /tmp && export TMPDIR=/tmp
R=/root/repo
DIMS=512x4096x256 timeout 300 rocprofv3 --pmc FETCH_SIZE \
    -d $R/$OUT/pmc_fetch_long -o fetch -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -2
DIMS=512x4096x256 timeout 300 rocprofv3 --pmc WRITE_SIZE \
    -d $R/$OUT/pmc_write_long -o write -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -2
cd $R
ls -la $OUT/pmc_fetch_long $OUT/pmc_write_long 2>/dev/null | head
python scripts/extract_traffic.py $OUT/pmc_fetch_long/*fetch*.csv \
    $OUT/pmc_write_long/*write*.csv 2>&1 | tail -12
echo DONE
