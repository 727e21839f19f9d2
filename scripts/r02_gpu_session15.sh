# r02 session 15: PMC traffic at the ragged mid-size shape
# (1536,1536,256): row-parallel vs rolling FETCH/WRITE — is the
# row-parallel loss excess HBM traffic (L2 absorption failure on the
# ragged gy grid) or a rate effect?
set -u
OUT=gpurun_out/r02s15
mkdir -p $OUT
R=/root/repo
cd /tmp && export TMPDIR=/tmp
{
  for roll in -1 1; do
    DIMS=1536x1536x256 PAM_FD_ROLL=$roll timeout 300 rocprofv3 \
      --pmc FETCH_SIZE --output-format csv \
      -d $R/$OUT/fetch_roll$roll -o fetch -- \
      python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
    DIMS=1536x1536x256 PAM_FD_ROLL=$roll timeout 300 rocprofv3 \
      --pmc WRITE_SIZE --output-format csv \
      -d $R/$OUT/write_roll$roll -o write -- \
      python $R/scripts/gpu_fd_shape_sweep.py 2>&1 | tail -1
  done
  cd $R
  for roll in -1 1; do
    echo "-- roll=$roll"
    python scripts/extract_traffic.py $OUT/fetch_roll$roll/*fetch*.csv \
        $OUT/write_roll$roll/*write*.csv 2>&1 | tail -6
  done
} > $R/$OUT/s15.log 2>&1
cd $R
cat $OUT/s15.log
