# r02 session 5: cgemm pipeline-knob sweep at the cfg5 Fredholm shapes
# (PAM_CGEMM_TILE/NBUF/BK), plus an MDC re-check under the best combo.
set -u
mkdir -p gpurun_out/r02s5
{
  echo "== cgemm knob sweep (fred probe: nf=513 ns=64 nr=256 nv=256) =="
  for env in "" "PAM_CGEMM_TILE=2" "PAM_CGEMM_NBUF=1" "PAM_CGEMM_BK=8"; do
    echo "-- env: ${env:-default}"
    env $env timeout 240 python scripts/gpu_fred_probe.py 2>&1 | tail -2
  done
  echo "== MDC probe (default knobs) =="
  timeout 240 python scripts/gpu_mdc_probe.py 2>&1 | tail -4
} > gpurun_out/r02s5/sweep.log 2>&1
tail -20 gpurun_out/r02s5/sweep.log
