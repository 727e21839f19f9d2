# r02 session 13: rolling-kernel TGT/CV knob sweep at the mid-size
# shapes the refined dispatch now routes to it.
set -u
mkdir -p gpurun_out/r02s13
{
  for dims in 1536x1536x256 1536x2560x128 1024x2560x256; do
    for tgt in 1024 2048 4096; do
      for cv in 4 8; do
        DIMS=$dims PAM_FD_ROLL=1 PAM_FD_ROLL_TGT=$tgt PAM_FD_ROLL_CV=$cv \
          timeout 180 python scripts/gpu_fd_shape_sweep.py 2>&1 \
          | tail -1 | sed "s/^/tgt=$tgt cv=$cv /"
      done
    done
  done
} > gpurun_out/r02s13/s13.log 2>&1
cat gpurun_out/r02s13/s13.log
