"""Native RCCL communicator (SURVEY.md §1b "Comm core (C++)").

The framework's own ncclComm over xGMI, independent of torch.distributed's
ProcessGroupNCCL: only the 128-byte unique id travels over the
torch.distributed rendezvous store (gloo or any backend works for that), then
every collective is a direct ncclAllReduce enqueued on the current torch
stream — stream-ordered, hipGraph-capturable, no Work-object bookkeeping.

Usage:
    comm = RcclComm.bootstrap()   # inside an initialized process group
    comm.all_reduce_sum(flat_bucket_tensor)   # in-place, current stream
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..ops import native


class RcclComm:
    def __init__(self, rank: int, world: int):
        self.rank = rank
        self.world = world
        self._alive = world >= 1

    @classmethod
    def bootstrap(cls, group=None, timeout_s: float = 120.0):
        """Create the communicator across the current torch.distributed
        group: rank 0 generates the RCCL unique id, the store broadcasts it.

        timeout_s > 0 arms the C++ watchdog: any collective that fails to
        complete within the timeout triggers ncclCommAbort (loud abort, no
        infinite hang on peer death — SURVEY.md §5 failure handling)."""
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        id_size = int(native.ext().comm_id_size())  # sizeof(ncclUniqueId)
        if rank == 0:
            idb = native.ext().comm_unique_id()
            assert idb.numel() == id_size
        else:
            idb = torch.zeros(id_size, dtype=torch.uint8)
        # broadcast the id bytes over whatever backend the group uses
        if dist.get_backend(group) == "nccl":
            dev_idb = idb.cuda()
            dist.broadcast(dev_idb, src=0, group=group)
            idb = dev_idb.cpu()
        else:
            dist.broadcast(idb, src=0, group=group)
        if timeout_s and timeout_s > 0:
            native.ext().comm_set_timeout(float(timeout_s))
        native.ext().comm_init(rank, world, idb.contiguous())
        return cls(rank, world)

    def all_reduce_sum(self, t: torch.Tensor):
        native.ext().comm_all_reduce_sum(t)

    @property
    def aborted(self) -> bool:
        return bool(native.ext().comm_aborted())

    def abort(self):
        native.ext().comm_abort()

    def destroy(self):
        if self._alive:
            native.ext().comm_destroy()
            self._alive = False
