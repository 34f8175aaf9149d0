"""RCCL sanity on a real MI355X: world-1 init + collectives through our
topology, and (informational) a 2-rank-on-1-GPU attempt.

Run via: python -m torch.distributed.run --nnodes=1 --nproc-per-node N
         --master-addr 127.0.0.1 tools/check_rccl.py
"""
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    torch.cuda.set_device(rank % torch.cuda.device_count())
    # init_parallel_env skips process-group creation at world==1; this
    # check wants the RCCL communicator itself, so init explicitly
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=rank, world_size=world)
    from paddlenlp_amd.parallel.topology import init_parallel_env

    topo = init_parallel_env(dp_degree=world, backend="nccl")
    x = torch.ones(1024, device="cuda") * (rank + 1)
    dist.all_reduce(x)
    expect = world * (world + 1) / 2
    assert torch.allclose(x, torch.full_like(x, expect)), x[:4]
    # reduce_scatter/all_gather (the ZeRO bucket path) over RCCL
    buf = torch.arange(64 * world, device="cuda", dtype=torch.float32)
    shard = buf[rank * 64:(rank + 1) * 64]
    dist.reduce_scatter_tensor(shard, buf)
    dist.all_gather_into_tensor(buf, shard)
    torch.cuda.synchronize()
    print(f"[check_rccl] rank {rank}/{world}: all-reduce + RS/AG OK "
          f"(backend nccl -> RCCL), device {torch.cuda.current_device()}")
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
