"""Helpers for multi-process CPU (gloo) tests: run a function in N spawned
ranks on 127.0.0.1 and propagate failures."""

import os
import socket

import torch.multiprocessing as mp


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _entry(rank, world, port, fn, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        fn(rank, world, *args)
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world: int = 2, args: tuple = ()):  # gloo CPU
    port = free_port()
    mp.start_processes(_entry, args=(world, port, fn, args), nprocs=world,
                       join=True, start_method="spawn")
