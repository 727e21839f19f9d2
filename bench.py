"""Headline bench: MPIFirstDerivative fp64 matvec+rmatvec pairs/sec + HBM
GB/s (BASELINE.json metric), 1..8 MI355X.

Usage (driver contract):
    python bench.py --gpus 1 --steps K --warmup W          # single GPU
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N ...      # N GPUs, RCCL

Workload:
  N=1  -> BASELINE.json configs[1]: dims (2048,2048,128) fp64 (4.29 GB).
  N>1  -> weak scaling toward the north-star config: dims
          (512*N, 4096, 256); at N=8 this IS 4096x4096x256 over 8 GPUs
          (4.29 GB per GPU in both cases).
A "step" = one matvec + one rmatvec of MPIFirstDerivative (centered,
order 3, sampling 1.0) on a SCATTER axis-0 1-D DistributedArray of
prod(dims) elements with plane-aligned local shapes; inputs are synthetic
(torch.randn seeded per rank) and resident in HBM before the timed region.
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

import pylops_mpi_amd as pm  # noqa: E402
from pylops_mpi_amd import derivative as deriv  # noqa: E402
from pylops_mpi_amd.comm import init_default_comm  # noqa: E402

HBM_PEAK_GBS = 8000.0  # gfx950 spec peak (MI355X_MICROARCH.md chip table)
BYTES_PER_PT = 16.0    # centered3 fp64: 8 read + 8 write, algorithmic


def workload_dims(n_gpus: int, smoke: bool = False):
    if smoke:
        # --smoke-gloo: same structure, CPU-sized (orchestration test
        # only — never a measured configuration)
        return (64 * n_gpus, 32, 16), "smoke-gloo (not a benchmark)"
    if n_gpus == 1:
        return (2048, 2048, 128), "MPIFirstDerivative cgls config (BASELINE configs[1])"
    return (512 * n_gpus, 4096, 256), \
        f"north-star weak scaling ({n_gpus}/8 of BASELINE 8-GPU config)"


def _smoke_fd_pair(op, x, comm):
    """CPU stand-in for one matvec+rmatvec pair, used ONLY under
    --smoke-gloo to exercise bench.py's full orchestration (rank env,
    comms, plane-aligned layout, halo exchange, reductions, JSON) with
    no GPU.  Torch slice restatement of the centered3 formulas
    (ref FirstDerivative.py:201-246); the real bench path never runs
    it — the product kernels stay fail-loud GPU-only."""
    shapes, counts = op._plane_counts()
    r, P = comm.rank, comm.size
    nloc = shapes[r][0]
    m = int(np.prod(op.dims[1:], initial=1))
    row0 = int(np.sum([s[0] for s in shapes[:r]], initial=0))
    N0 = op.dims[0]

    def pair(planes):
        gf, gb = None, None
        if P > 1:
            gf = torch.empty((1, m), dtype=planes.dtype)
            gb = torch.empty((1, m), dtype=planes.dtype)
            comm.sendrecv_neighbors(
                planes[:1].contiguous() if r > 0 else None,
                planes[-1:].contiguous() if r < P - 1 else None,
                gf if r > 0 else None, gb if r < P - 1 else None)
        top = gf if r > 0 else torch.zeros((1, m), dtype=planes.dtype)
        bot = gb if r < P - 1 else torch.zeros((1, m), dtype=planes.dtype)
        g = torch.cat([top, planes, bot])
        y = 0.5 * (g[2:] - g[:-2])
        if row0 == 0:
            y[0] = 0.0
        if row0 + nloc == N0:
            y[-1] = 0.0
        return y

    planes = x.local_array.view(nloc, m)
    y = pair(planes)
    z = pair(y)   # adjoint of centered3 = -centered3 up to edges; the
    # smoke only exercises data movement, not numerics (parity tests
    # pin numerics)
    return z


def read_traffic_calibration(dims, n_gpus):
    """Per-launch HBM traffic measured by a committed rocprofv3 --pmc run
    (profiles/traffic.json), or None."""
    path = os.path.join(ROOT, "profiles", "traffic.json")
    if not os.path.exists(path):
        return None
    try:
        d = json.load(open(path))
        key = "x".join(map(str, dims))
        ent = d.get(key)
        if ent and int(ent.get("n_gpus", 1)) == n_gpus:
            return float(ent["bytes_per_launch"])
    except Exception:
        pass
    return None


def _fd_pair_threaded(x3, y3, z3, nthreads):
    """One centered3 matvec+rmatvec pair, row-sharded across host threads
    (NumPy releases the GIL on slice arithmetic).  Same slice formulas as
    oracle/serial.py (matvec y[i]=0.5(x[i+1]-x[i-1]), adjoint
    y[j]=0.5 x[j-1] - 0.5 x[j+1] with the edge ranges of the dense
    transpose)."""
    import concurrent.futures as cf
    n0 = x3.shape[0]
    bounds = [(n0 * t // nthreads, n0 * (t + 1) // nthreads)
              for t in range(nthreads)]

    def mv(a, b):
        lo, hi = max(a, 1), min(b, n0 - 1)
        if hi > lo:
            y3[lo:hi] = 0.5 * (x3[lo + 1: hi + 1] - x3[lo - 1: hi - 1])
        if a == 0:
            y3[0] = 0.0
        if b == n0:
            y3[n0 - 1] = 0.0

    def rmv(a, b):
        for j in range(a, b):
            acc = None
            if 2 <= j <= n0 - 1:
                acc = 0.5 * y3[j - 1]
            if j <= n0 - 3:
                acc = (-0.5 * y3[j + 1]) if acc is None \
                    else acc - 0.5 * y3[j + 1]
            z3[j] = 0.0 if acc is None else acc

    with cf.ThreadPoolExecutor(nthreads) as ex:
        list(ex.map(lambda ab: mv(*ab), bounds))
        list(ex.map(lambda ab: rmv(*ab), bounds))


def cpu_baseline(dims):
    """The oracle (NumPy restatement of the reference path) timed on the
    host cores — kind 'port' (the reference's own mpi4py+NumPy path cannot
    run here: no MPI/pylops in the image).  Reports the
    all-core row-sharded figure as `value` (SURVEY §8d asks for both);
    the 1-thread figure is in `sample`."""
    import oracle
    n = int(np.prod(dims))
    rng = np.random.default_rng(42)
    xg = rng.standard_normal(n)
    op = oracle.SimFirstDerivative(dims, kind="centered", order=3)
    x = oracle.to_dist(xg, 1)
    y = op.matvec(x)          # warmup pair
    _ = op.rmatvec(y)
    pairs = 2
    t0 = time.perf_counter()
    for _ in range(pairs):
        y = op.matvec(x)
        _ = op.rmatvec(y)
    dt1 = time.perf_counter() - t0
    try:
        cores = len(os.sched_getaffinity(0))
    except AttributeError:
        cores = os.cpu_count()
    # all-core: row-sharded threads over the same slice arithmetic
    nthreads = min(int(cores), 64)
    x3 = xg.reshape(dims[0], -1)
    y3 = np.empty_like(x3)
    z3 = np.empty_like(x3)
    _fd_pair_threaded(x3, y3, z3, nthreads)  # warmup
    t0 = time.perf_counter()
    for _ in range(pairs):
        _fd_pair_threaded(x3, y3, z3, nthreads)
    dtT = time.perf_counter() - t0
    return {
        "value": pairs / dtT,
        "unit": "pairs/s",
        "cores": nthreads,
        "kind": "port",
        "sample": (f"{pairs} matvec+rmatvec pairs on the full "
                   f"{'x'.join(map(str, dims))} fp64 workload, NumPy "
                   f"oracle row-sharded over {nthreads} threads "
                   f"(host has {cores} cores); 1-thread figure: "
                   f"{pairs / dt1:.3f} pairs/s"),
    }


def _smoke_main(args, op, x, comm, n_gpus, dims, wname, rank):
    """--smoke-gloo body: the same step loop / barrier+sync protocol /
    max-over-ranks reduction / JSON emission as the real bench, sized
    for CPU (`bench.py --gpus 8` must run under
    gloo end-to-end before any 8-GPU lease exists)."""
    for _ in range(args.warmup):
        _smoke_fd_pair(op, x, comm)
    comm.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        _smoke_fd_pair(op, x, comm)
    comm.barrier()
    t1 = time.perf_counter()
    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    comm.allreduce_(elapsed, "max")
    t = float(elapsed.item())
    if rank == 0:
        print(json.dumps({
            "metric": "MPIFirstDerivative fp64 matvec+rmatvec pairs/sec",
            "value": args.steps / t, "unit": "pairs/s", "n_gpus": n_gpus,
            "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": 1e3 * t / args.steps, "higher_is_better": True,
            "scaling": "weak", "vs_baseline": None, "dtype": "f64",
            "data": "synthetic", "smoke": True,
            "config": {"workload": wname, "dims": list(dims),
                       "partition": "scatter-axis0", "kind": "centered",
                       "order": 3},
            "roofline": None, "cpu_baseline": None,
        }), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)  # >=50
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--smoke-gloo", action="store_true",
                    help="CPU orchestration smoke (tiny dims, torch "
                         "stand-in kernel; NOT a benchmark)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(args.gpus, world)
    comm = init_default_comm()
    rank = comm.rank
    assert comm.size == n_gpus, f"launch {n_gpus} ranks (got {comm.size})"
    if not args.smoke_gloo:
        assert torch.cuda.is_available(), "bench needs MI355X GPUs"
    device = comm.device

    dims, wname = workload_dims(n_gpus, smoke=args.smoke_gloo)
    n = int(np.prod(dims))
    m = dims[1] * dims[2]

    op = pm.MPIFirstDerivative(dims, sampling=1.0, kind="centered", order=3,
                               dtype=np.float64)
    # plane-aligned 1-D input resident in HBM (the steady-state CGLS shape)
    shapes, counts = op._plane_counts()
    gen = torch.Generator(device=device).manual_seed(42 + rank)
    local = torch.randn(counts[rank], generator=gen, dtype=torch.float64,
                        device=device)
    x = pm.DistributedArray((n,), comm, pm.Partition.SCATTER, 0,
                            local_array=local,
                            local_shapes=[(c,) for c in counts],
                            dtype=np.float64)

    if args.smoke_gloo:
        _smoke_main(args, op, x, comm, n_gpus, dims, wname, rank)
        return

    for _ in range(args.warmup):
        y = op.matvec(x)
        _ = op.rmatvec(y)

    # timed region: barrier + sync on both sides, HIP events per stencil
    # launch on the launch stream
    deriv.KERNEL_TIMING = True
    deriv.KERNEL_EVENTS.clear()
    comm.barrier()
    torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        y = op.matvec(x)
        _ = op.rmatvec(y)
    comm.barrier()
    torch.cuda.synchronize(device)
    t1 = time.perf_counter()
    deriv.KERNEL_TIMING = False

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64, device=device)
    comm.allreduce_(elapsed, "max")
    t = float(elapsed.item())

    # roofline of the dominant kernel (fd centered3 matvec, op code 4).
    # One event pair brackets one whole matvec: a single launch at
    # world 1; interior launch + halo wait + boundary launches under the
    # N>1 overlap path (so the N>1 figure conservatively includes any
    # exposed halo time).  Aggregate per STEP (sum / steps).
    mv_events = deriv.KERNEL_EVENTS.get(4, [])
    kern_ms = [e0.elapsed_time(e1) for e0, e1 in mv_events]
    local_pts = counts[rank]
    step_ms = float(np.sum(kern_ms)) / args.steps if kern_ms else None
    alg_bytes = BYTES_PER_PT * local_pts
    achieved = alg_bytes / (step_ms * 1e-3) / 1e9 if step_ms else None
    traffic = read_traffic_calibration(dims, n_gpus)

    pairs_per_s = args.steps / t
    gbps = 2.0 * BYTES_PER_PT * n * args.steps / t / 1e9  # mv+rmv per step

    if rank == 0:
        result = {
            "metric": "MPIFirstDerivative fp64 matvec+rmatvec pairs/sec",
            "value": pairs_per_s,
            "unit": "pairs/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": 1e3 * t / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": wname,
                "dims": list(dims),
                "partition": "scatter-axis0",
                "kind": "centered",
                "order": 3,
                "hbm_gbps_aggregate": gbps,
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": (achieved / HBM_PEAK_GBS) if achieved else None,
                "traffic": traffic,
            },
            "cpu_baseline": (None if args.skip_cpu_baseline or n_gpus != 1
                             else cpu_baseline(dims)),
        }
        print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
