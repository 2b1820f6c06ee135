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
    """Broadcast the ncclUniqueId from rank 0 and open the dd_comm on every rank.

    Oversubscribed testing (more ranks than GPUs): RCCL refuses two ranks on one device
    (ncclCommInitRank "invalid usage", measured), so the data plane falls back to a gloo
    host exchange — window math and accounting identical, bytes cross host memory. The
    bench line marks it (config["data_plane"]) so it can never pass as an xGMI number.
    """
    from .api import UNIQUE_ID_BYTES, Comm

    if world == 1:
        return Comm(Comm.unique_id(), 0, 1)
    import torch
    import torch.distributed as dist

    if torch.cuda.device_count() < world:
        return GlooFallbackComm(rank, world)
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


class _GlooExchanged:
    def __init__(self, ms, egress, total_rows):
        self._ms = ms
        self._egress = egress
        self.total_rows = total_rows

    def stats(self):
        return self._ms, self._egress

    def destroy(self):
        pass


class GlooFallbackComm:
    """Testing-only data plane for oversubscribed runs (ranks > GPUs): the partition
    windows of `network_shuffle.rs:232-244` are exchanged host-side with
    all_gather_object. Fixed-width columns only (the bench workloads); never used when
    one GPU per rank is available."""

    data_plane = "gloo-fallback(oversubscribed)"

    def __init__(self, rank, world):
        self.rank = rank
        self.world = world

    def destroy(self):
        pass

    def exchange(self, part):
        import time

        import torch.distributed as dist

        t0 = time.perf_counter()
        world = self.world
        ppr = part.nparts // world
        off = part.row_offsets()
        cols = [part.col_out(i) for i in range(len(part.batch.cols))]
        for c in cols:
            if c["dtype"] == "utf8":
                raise NotImplementedError("gloo fallback covers fixed-width columns")
        per_dest = []
        egress = 0
        for j in range(world):
            lo, hi = int(off[j * ppr]), int(off[(j + 1) * ppr])
            sl = [c["data"][lo:hi] for c in cols]
            if j != self.rank:
                egress += sum(int(a.nbytes) for a in sl)
            per_dest.append(sl)
        gathered = [None] * world
        dist.all_gather_object(gathered, per_dest)
        mine = [g[self.rank] for g in gathered]  # producer-rank order
        total_rows = sum(len(sl[0]) for sl in mine)
        ms = (time.perf_counter() - t0) * 1e3
        return _GlooExchanged(ms, egress, total_rows)
