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


def init_spark(*args, **kwargs):
    """Migration shim for the reference's shared/spark.py:26 entry point.

    This engine has no Spark: compute runs on the MI355X column store
    (one process per GPU, torch.distributed over RCCL). Returns the
    triple shape callers destructure — (ctx, None, None) — so
    `spark, sc, sqlContext = init_spark(...)` keeps working, with `ctx`
    standing in for the session everywhere the reference passed `spark`.
    All engine APIs accept this context as their first argument."""
    import warnings

    warnings.warn(
        "init_spark(): no Spark in anovos_amd — returning (AnovosContext, None, None); "
        "pass the context where the reference passed `spark`.",
        stacklevel=2,
    )
    ctx = init_context()
    return ctx, None, None
