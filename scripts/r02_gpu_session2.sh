#!/bin/bash
# r02 GPU session 2: roll2 (statically-rotated rolling stencil) parity +
# variant sweep at the long-row shape; PMC traffic (CSV) for the
# weak-scaling dims calibration.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
OUT=gpurun_out/r02b
mkdir -p $OUT

echo "== 1. roll2 parity =="
PAM_FD_ROLL=2 timeout 600 python -m pytest tests/test_gpu_parity.py \
    -x -q -m gpu -k "centered or forward or backward" 2>&1 | tail -2
PAM_FD_ROLL=2 PAM_FD_ROLL_CV=8 PAM_FD_VEC=4 timeout 600 python -m pytest \
    tests/test_gpu_parity.py -x -q -m gpu -k "centered" 2>&1 | tail -2
PAM_FD_ROLL=2 PAM_FD_ROLL_CV=2 timeout 300 python -m pytest \
    tests/test_gpu_parity.py -x -q -m gpu -k "second" 2>&1 | tail -2

echo "== 2. rolling sweep at 512x4096x256 =="
SW="timeout 120 python scripts/gpu_fd_shape_sweep.py"
export DIMS=512x4096x256
for ROLL in 1 2; do
  for CV in 2 4 8; do
    for VEC in 2 4; do
      echo "ROLL=$ROLL CV=$CV VEC=$VEC"
      PAM_FD_ROLL=$ROLL PAM_FD_ROLL_CV=$CV PAM_FD_VEC=$VEC $SW 2>&1 | tail -1
    done
  done
done
echo "== 2b. best candidates at the bench shape =="
DIMS=2048x2048x128 PAM_FD_ROLL=2 PAM_FD_ROLL_CV=4 $SW 2>&1 | tail -1
DIMS=2048x2048x128 PAM_FD_ROLL=2 PAM_FD_ROLL_CV=8 $SW 2>&1 | tail -1
DIMS=2048x2048x128 PAM_FD_ROLL=2 PAM_FD_ROLL_CV=4 PAM_FD_VEC=4 $SW 2>&1 | tail -1
echo "== 2c. roll2 TGT sweep at long shape (best CV/VEC from above used later) =="
PAM_FD_ROLL=2 PAM_FD_ROLL_TGT=2048 $SW 2>&1 | tail -1
PAM_FD_ROLL=2 PAM_FD_ROLL_TGT=16384 $SW 2>&1 | tail -1

echo "== 3. PMC traffic (CSV) at long shape, roll2 default-CV =="
cd /tmp && export TMPDIR=/tmp
R=/root/repo
DIMS=512x4096x256 PAM_FD_ROLL=2 timeout 300 rocprofv3 --pmc FETCH_SIZE \
    --output-format csv -d $R/$OUT/pmc_fetch_roll2 -o fetch -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
DIMS=512x4096x256 PAM_FD_ROLL=2 timeout 300 rocprofv3 --pmc WRITE_SIZE \
    --output-format csv -d $R/$OUT/pmc_write_roll2 -o write -- \
    python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
cd $R
find $OUT -name "*.csv" | head
python scripts/extract_traffic.py $(find $OUT -name "*fetch*.csv") \
    $(find $OUT -name "*write*.csv") 2>&1 | tail -10
echo DONE
