# r02 session 6: validate the BK auto rule (K<=64 -> BK=8) — full GPU
# suite, fred A/B new-default vs forced BK=16, MDC probe, bench rep.
set -u
mkdir -p gpurun_out/r02s6
{
  echo "== pytest -m gpu =="
  timeout 1200 python -m pytest tests/ -x -q -m gpu 2>&1 | tail -1
  echo "== fred probe: auto (K<=64 -> BK=8) =="
  timeout 240 python scripts/gpu_fred_probe.py 2>&1 | tail -2
  echo "== fred probe: forced BK=16 (old default) =="
  PAM_CGEMM_BK=16 timeout 240 python scripts/gpu_fred_probe.py 2>&1 | tail -2
  echo "== MDC probe (new default) =="
  timeout 240 python scripts/gpu_mdc_probe.py 2>&1 | tail -2
  echo "== bench (full, with cpu_baseline) =="
  timeout 900 python bench.py --gpus 1 --steps 50 --warmup 5 2>/dev/null | tail -1
} > gpurun_out/r02s6/s6.log 2>&1
tail -12 gpurun_out/r02s6/s6.log
