# r02 session 8: round-final validation rep — full GPU suite, smoke,
# full bench (with cpu_baseline), rocprofv3 kernel stats of the bench.
set -u
mkdir -p gpurun_out/r02s8
{
  echo "== pytest -m gpu =="
  timeout 1200 python -m pytest tests/ -x -q -m gpu 2>&1 | tail -1
  echo "== smoke =="
  timeout 600 python -c "import __graft_entry__ as g; g.smoke(); print('smoke ok')" 2>&1 | tail -1
  echo "== bench (full) =="
  timeout 900 python bench.py --gpus 1 --steps 50 --warmup 5 2>/dev/null | tail -1
  echo "== rocprof kernel stats (bench, short) =="
  cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
  ( cd /tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv \
      -d "$GRAFT_REPO_ROOT/gpurun_out/r02s8/prof" -- \
      python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 20 --warmup 3 --skip-cpu-baseline \
      > /tmp/prof.log 2>&1; tail -1 /tmp/prof.log )
} > gpurun_out/r02s8/s8.log 2>&1
tail -8 gpurun_out/r02s8/s8.log
ls gpurun_out/r02s8/prof 2>/dev/null | head -5
