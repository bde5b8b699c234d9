"""Per-process resources: device, streams, distributed context.

Reimplements the role of the reference ``Resources`` (include/resources.h:21-60,
src/resources.cu): one object owning the device, the compute / communication
HIP streams and the distributed communicator for this process. MI355X-native:
one process per GPU; torch.distributed (RCCL over xGMI) instead of MPI.
"""

from __future__ import annotations

import os
from typing import Optional

import torch


class Resources:
    def __init__(self, device: Optional[str] = None, distributed: bool = False):
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        if self.is_cuda and self.device.index is None:
            self.device = torch.device("cuda", torch.cuda.current_device())
        # Dedicated streams: main compute + halo-communication stream for
        # latency hiding (reference: m_bdy_stream,
        # include/distributed/distributed_manager.h:1000-1021).
        if self.is_cuda:
            torch.cuda.set_device(self.device)
            self.stream = torch.cuda.current_stream(self.device)
            self.comm_stream = torch.cuda.Stream(self.device)
        else:
            self.stream = None
            self.comm_stream = None
        self.distributed = distributed
        self.rank = 0
        self.world_size = 1
        if distributed:
            import torch.distributed as dist
            if not dist.is_initialized():
                backend = "nccl" if self.is_cuda else "gloo"
                os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
                os.environ.setdefault("MASTER_PORT", "29511")
                dist.init_process_group(backend=backend)
            self.rank = dist.get_rank()
            self.world_size = dist.get_world_size()

    def synchronize(self):
        if self.is_cuda:
            torch.cuda.synchronize(self.device)


_default: Optional[Resources] = None


def default_resources() -> Resources:
    global _default
    if _default is None:
        _default = Resources()
    return _default
