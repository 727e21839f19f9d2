# r02 session 12: validate the refined auto-roll rule — auto (no force)
# must match the per-shape winner at all 12 sweep shapes; suite + bench.
set -u
mkdir -p gpurun_out/r02s12
{
  echo "== auto dispatch across the 12 sweep shapes =="
  for dims in 2048x2048x128 1536x1536x256 1024x4096x128 1024x1536x512 \
              512x4096x256 2048x1024x128 2048x1536x128 1536x2560x128 \
              1024x3584x128 1024x2560x256 1024x1280x512 512x2048x512; do
    DIMS=$dims timeout 180 python scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
  done
  echo "== pytest -m gpu =="
  timeout 1200 python -m pytest tests/ -x -q -m gpu 2>&1 | tail -1
  echo "== bench =="
  timeout 600 python bench.py --gpus 1 --steps 50 --warmup 5 --skip-cpu-baseline 2>/dev/null | tail -1
} > gpurun_out/r02s12/s12.log 2>&1
cat gpurun_out/r02s12/s12.log
