"""Engine context — the analog of the reference's global SparkSession
singleton (reference shared/spark.py:95-166).

``init_context()`` builds an ``AnovosContext`` holding the compute device
(the local MI355X when visible, else CPU), the distributed state, and the
side stream used to overlap RCCL collectives with compute. A module-level
``ctx`` mirrors the reference's module-level ``spark`` object so domain
functions can be called as ``f(ctx, idf, ...)``.
"""

from __future__ import annotations

import os

import torch

from anovos_amd.core import dist


class AnovosContext:
    def __init__(self, device=None):
        dist.init_from_env()
        if device is None:
            if torch.cuda.is_available():
                # modulo: tests may oversubscribe one device with 2 ranks
                device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0")) % max(torch.cuda.device_count(), 1))
            else:
                device = torch.device("cpu")
        self.device = torch.device(device)
        self.side_stream = dist.SideStream()

    @property
    def rank(self) -> int:
        return dist.rank()

    @property
    def world_size(self) -> int:
        return dist.world_size()

    def synchronize(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def __repr__(self):
        return f"AnovosContext(device={self.device}, rank={self.rank}/{self.world_size})"


_ctx = None


def init_context(device=None) -> AnovosContext:
    global _ctx
    if _ctx is None or device is not None:
        _ctx = AnovosContext(device)
    return _ctx


def get_context() -> AnovosContext:
    return init_context()
