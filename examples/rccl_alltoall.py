#!/usr/bin/env python3
"""RCCL fan-out example: bootstrap an RcclMesh over starway tagged
messaging, then run an equal-chunk all-to-all. Launch one process per GPU:

  torchrun --nnodes=1 --nproc-per-node N examples/rccl_alltoall.py
"""
import asyncio
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


async def main() -> None:
    import torch
    import torch.distributed as dist

    import starway_amd as sw
    from starway_amd import rccl as swr

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        torch.cuda.set_device(rank % torch.cuda.device_count())
        dist.init_process_group("gloo")

    # Rendezvous of the 128-byte nccl id over starway itself: rank 0 runs a
    # server, everyone else connects and receives the id as a tagged msg.
    if rank == 0:
        server = sw.Server()
        blob = server.listen_address()
        if world > 1:
            blobs = [blob]
            dist.broadcast_object_list(blobs, src=0)
        eps = []
        while len(eps) < world - 1:
            eps = list(server.list_clients())
            await asyncio.sleep(0.01)
        mesh = await swr.bootstrap_from_messaging(
            "root", (server, eps), rank=rank, world=world,
            device=torch.cuda.current_device())
    else:
        blobs = [None]
        dist.broadcast_object_list(blobs, src=0)
        client = sw.Client()
        await client.aconnect_address(blobs[0])
        mesh = await swr.bootstrap_from_messaging(
            "peer", client, rank=rank, world=world,
            device=torch.cuda.current_device())

    n = 1 << 20
    send = torch.full((n * world,), rank, dtype=torch.uint8, device="cuda")
    recv = torch.zeros(n * world, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    mesh.all_to_all(send, recv)
    mesh.synchronize()
    expect = torch.cat([
        torch.full((n,), p, dtype=torch.uint8, device="cuda")
        for p in range(world)
    ])
    assert torch.equal(recv, expect)
    print(f"[rank {rank}] all-to-all OK ({world} ranks, {n} B chunks)")


if __name__ == "__main__":
    asyncio.run(main())
