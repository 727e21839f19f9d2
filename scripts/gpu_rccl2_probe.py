"""Can RCCL run world-2 with both ranks on the one visible GPU?  If yes,
the N>1 RCCL code paths (halo exchange, allreduce, bcast) become
testable on 1-GPU boxes."""
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    rank = int(os.environ["RANK"])
    dist.init_process_group("nccl", rank=rank,
                            world_size=int(os.environ["WORLD_SIZE"]))
    torch.cuda.set_device(0)
    t = torch.ones(8, device="cuda:0") * (rank + 1)
    dist.all_reduce(t)
    ok1 = bool((t == 3.0).all())
    # p2p pair (the halo pattern)
    s = torch.full((4,), float(rank), device="cuda:0")
    r = torch.empty(4, device="cuda:0")
    other = 1 - rank
    ops = [dist.P2POp(dist.irecv, r, other), dist.P2POp(dist.isend, s, other)]
    for w in dist.batch_isend_irecv(ops):
        w.wait()
    torch.cuda.synchronize()
    ok2 = bool((r == float(other)).all())
    print(f"rank {rank}: allreduce={ok1} p2p={ok2}", flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
