"""bench.py orchestration under gloo on CPU: the
driver's exact torchrun launch shape at world 2 and 8 must run
end-to-end — rank env parsing, plane-aligned layout, halo exchange,
barrier+sync protocol, max-over-ranks reduction, rank-0 JSON line —
before any 8-GPU lease exists.  --smoke-gloo swaps the GPU kernel for a
tiny torch stand-in; it never touches the measured path."""
import json
import os
import socket
import subprocess
import sys

import pytest


def _free_port():
    # a fixed port flakes: TIME_WAIT from the previous world's
    # rendezvous can hold it for ~60 s
    with socket.socket() as sck:
        sck.bind(("127.0.0.1", 0))
        return sck.getsockname()[1]

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("world", [2, 8])
def test_bench_smoke_gloo(world):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", f"--nproc-per-node={world}",
           "--master-addr", "127.0.0.1",
           "--master-port", str(_free_port()),
           os.path.join(ROOT, "bench.py"),
           "--gpus", str(world), "--steps", "2", "--warmup", "1",
           "--smoke-gloo"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=300,
                         cwd=ROOT, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    res = json.loads(lines[0])
    assert res["smoke"] is True
    assert res["n_gpus"] == world
    assert res["value"] > 0
    assert res["steps"] == 2
    assert res["config"]["dims"][0] == 64 * world


def test_bench_smoke_gloo_world1():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    cmd = [sys.executable, os.path.join(ROOT, "bench.py"),
           "--gpus", "1", "--steps", "2", "--warmup", "1", "--smoke-gloo"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=180,
                         cwd=ROOT, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    res = json.loads([ln for ln in out.stdout.splitlines()
                      if ln.startswith("{")][0])
    assert res["smoke"] is True and res["n_gpus"] == 1
