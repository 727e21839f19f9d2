#!/bin/bash
# r02 GPU session 3: tall-chunk rolling sweep (preload amortization) +
# NT stores; pick the long-row winner.
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
OUT=gpurun_out/r02c
mkdir -p $OUT

SW="timeout 120 python scripts/gpu_fd_shape_sweep.py"
export DIMS=512x4096x256
export PAM_FD_ROLL=1 PAM_FD_VEC=4
for CV in 4 8; do
  for TGT in 512 1024 2048 4096; do
    echo "CV=$CV TGT=$TGT"
    PAM_FD_ROLL_CV=$CV PAM_FD_ROLL_TGT=$TGT $SW 2>&1 | tail -1
  done
done
echo "== NT stores =="
PAM_FD_ROLL_CV=8 PAM_FD_ROLL_TGT=1024 PAM_FD_NT=1 $SW 2>&1 | tail -1
PAM_FD_ROLL_CV=8 PAM_FD_NT=1 $SW 2>&1 | tail -1
PAM_FD_ROLL_CV=4 PAM_FD_ROLL_TGT=1024 PAM_FD_NT=1 $SW 2>&1 | tail -1
echo "== winner candidates at bench shape =="
DIMS=2048x2048x128 PAM_FD_ROLL_CV=8 PAM_FD_ROLL_TGT=1024 $SW 2>&1 | tail -1
DIMS=2048x2048x128 PAM_FD_ROLL_CV=8 PAM_FD_ROLL_TGT=1024 PAM_FD_NT=1 $SW 2>&1 | tail -1
echo "== row-parallel NT at long shape (for completeness) =="
PAM_FD_ROLL=0 PAM_FD_CAP=131072 $SW 2>&1 | tail -1
echo DONE
