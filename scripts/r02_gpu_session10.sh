# r02 session 10: auto-roll crossover sweep — row-parallel (-1) vs
# rolling (1) at row sizes bracketing the 4 MiB dispatch threshold,
# matvec and rmatvec.
set -u
mkdir -p gpurun_out/r02s10
{
  for dims in 2048x2048x128 1536x1536x256 1024x4096x128 1024x1536x512 512x4096x256; do
    for roll in -1 1; do
      for rmv in 0 1; do
        DIMS=$dims PAM_FD_ROLL=$roll RMATVEC=$rmv \
          timeout 180 python scripts/gpu_fd_shape_sweep.py 2>&1 \
          | tail -1 | sed "s/^/roll=$roll /"
      done
    done
  done
} > gpurun_out/r02s10/s10.log 2>&1
cat gpurun_out/r02s10/s10.log
