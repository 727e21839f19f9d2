# r02 session 11: denser auto-roll crossover sweep — power-of-two vs
# non-power-of-two row sizes from 1 to 8 MiB (matvec only; rmv mirrors
# per s10).
set -u
mkdir -p gpurun_out/r02s11
{
  for dims in 2048x1024x128 2048x1536x128 1536x2560x128 1024x3584x128 \
              1024x2560x256 1024x1280x512 512x2048x512; do
    for roll in -1 1; do
      DIMS=$dims PAM_FD_ROLL=$roll \
        timeout 180 python scripts/gpu_fd_shape_sweep.py 2>&1 \
        | tail -1 | sed "s/^/roll=$roll /"
    done
  done
} > gpurun_out/r02s11/s11.log 2>&1
cat gpurun_out/r02s11/s11.log
