"""Environment feature gates — the analogue of the reference's
pylops_mpi/utils/deps.py:58-66 (NCCL_PYLOPS_MPI / PYLOPS_MPI_CUDA_AWARE):

  PAM_DISABLE_OVERLAP=1   serialize the halo exchange before the stencil
                          kernel instead of overlapping it with the
                          interior rows (debug aid)
  PAM_FD_VEC={1,2,4}      force the stencil kernel's per-lane vector width
                          (default: measured optimum, 16 B/lane)
  PAM_FD_NT=1             nt cache hint on the stencil's output stores
                          (measured no-change on the production kernel —
                          documented negative, csrc/pam.hip)
  PAM_FD_GY / PAM_FD_CAP  stencil launch-shape A/B knobs (defaults: one
                          block-row per row, ~4 vector iterations per
                          block — the measured optimum, csrc/pam.hip)
  PAM_FD_ROLL             rolling-window stencil select: unset = AUTO
                          (rows >= PAM_FD_LONGROW bytes, default 4 MiB,
                          use the 1-load/pt rolling kernel — the r02
                          long-row fix, 4.85 -> 5.35 TB/s at the N=8
                          per-rank shape); -1 = force row-parallel;
                          1/2 = force rolling (2 = statically-rotated
                          variant, measured slower at CV8xV4)
  PAM_FD_ROLL_CV / PAM_FD_ROLL_TGT  rolling chains per thread (default
                          8) and target grid size (default 2048) — the
                          r02 swept optima
  PAM_CGEMM_TILE/_NBUF/_BK  cgemm pipeline A/B knobs (defaults measured
                          best at the cfg5 shapes; 128x64 tile, NBUF=1
                          and BK=8 all measured neutral-to-negative,
                          csrc/gemm.hip)
  PAM_EW_CAP              1-D elementwise grid cap (default: none — one
                          block per 256 vector items; the old 4096-block
                          grid-stride loop cost 28% of axpy bandwidth)
  PAM_DISABLE_DEVSCALARS=1  run CG/CGLS with host-side scalars (one
                          blocking readback per dot) instead of the
                          single-sync device-scalar iteration (debug aid;
                          both paths are bit-identical)

The RCCL data plane itself has no gate: it IS the backend (there is no
MPI in this stack to fall back to)."""
import os


def env_flag(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    return default if v is None else v not in ("0", "", "false", "False")


overlap_enabled = not env_flag("PAM_DISABLE_OVERLAP")
devscalars_enabled = not env_flag("PAM_DISABLE_DEVSCALARS")
