"""RCCL world=2 on ONE GPU: exercises the REAL collective path
(`reduce_scatter_tensor` / `all_gather_into_tensor` on the nccl backend)
that round 1 never executed.  Both ranks map to
cuda:0; RCCL supports multiple ranks per device on ROCm.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 tools/rccl2_probe.py
"""
import os
import sys

import torch
import torch.distributed as dist


def main() -> None:
    torch.cuda.set_device(0)  # BOTH ranks on device 0
    dist.init_process_group("nccl")
    rank = dist.get_rank()
    world = dist.get_world_size()
    n = 1 << 20
    flat = torch.full((n,), float(rank + 1), device="cuda")
    shard = torch.empty(n // world, device="cuda")
    dist.reduce_scatter_tensor(shard, flat, op=dist.ReduceOp.AVG)
    expect = (1 + world) / 2.0
    assert torch.allclose(shard, torch.full_like(shard, expect)), shard[:4]
    out = torch.empty(n, device="cuda")
    shard.fill_(float(rank))
    dist.all_gather_into_tensor(out, shard)
    assert float(out[0]) == 0.0 and float(out[-1]) == world - 1, (out[0], out[-1])
    t = torch.tensor([float(rank)], device="cuda")
    dist.all_reduce(t)
    assert float(t) == sum(range(world))
    dist.barrier()
    if rank == 0:
        print("RCCL2_PROBE_OK world=%d" % world, flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    try:
        main()
    except Exception as e:  # surface the real error through torchrun
        import traceback

        print("RANK %s FAILED: %r" % (os.environ.get("RANK"), e), flush=True)
        traceback.print_exc()
        sys.exit(1)
