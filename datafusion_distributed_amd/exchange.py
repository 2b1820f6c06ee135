"""exchange.py — torch.distributed bootstrap for the RCCL data plane.

torch.distributed is plumbing only (process bootstrap + barriers): the exchange itself is
dd_exchange_run (RCCL grouped send/recv over xGMI). The RCCL communicator is
our own (dd_comm_init), created from an ncclUniqueId broadcast over the torch process group
store — this mirrors the reference's ChannelResolver worker-address distribution
(src/protocol/channel_resolver.rs:27-41) at the contract level: rank = task.
"""

import os


def init_process_group(backend=None):
    import torch.distributed as dist

    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        if backend is None:
            import torch

            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    return rank, world


def create_comm(rank, world):
    """Broadcast the ncclUniqueId from rank 0 and open the dd_comm on every rank."""
    from .api import UNIQUE_ID_BYTES, Comm

    if world == 1:
        return Comm(Comm.unique_id(), 0, 1)
    import torch
    import torch.distributed as dist

    if rank == 0:
        uid = Comm.unique_id()
        t = torch.tensor(list(uid), dtype=torch.uint8)
    else:
        t = torch.zeros(UNIQUE_ID_BYTES, dtype=torch.uint8)
    # the nccl backend broadcasts device tensors only; gloo wants host tensors
    if dist.get_backend() == "nccl":
        t = t.cuda()
    dist.broadcast(t, src=0)
    return Comm(bytes(t.cpu().tolist()), rank, world)
