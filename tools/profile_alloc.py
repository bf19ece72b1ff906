"""Measure torch caching-allocator cost for the bench's transform
allocation pattern (150 x 500 MB output columns per section) and compare
allocator configs. Run each config in its own process:
  python tools/profile_alloc.py            # default allocator
  PYTORCH_CUDA_ALLOC_CONF=expandable_segments:True python tools/profile_alloc.py
  PYTORCH_CUDA_ALLOC_CONF=backend:cudaMallocAsync python tools/profile_alloc.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    N = 125_000_000
    K = 150
    dev = "cuda"
    cfg = os.environ.get("PYTORCH_CUDA_ALLOC_CONF", "default")
    torch.empty(8, device=dev)

    res = {}
    # warm the pool with the exact block sizes
    bufs = [torch.empty(N, device=dev) for _ in range(K)]
    del bufs
    torch.cuda.synchronize()
    for rep in range(3):
        t0 = time.perf_counter()
        bufs = [torch.empty(N, device=dev) for _ in range(K)]
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        del bufs
        torch.cuda.synchronize()
        t2 = time.perf_counter()
        res[f"alloc150_rep{rep}_ms"] = round((t1 - t0) * 1000, 2)
        res[f"free150_rep{rep}_ms"] = round((t2 - t1) * 1000, 2)

    # fused fillnan wall with warm pool (the bench pattern: outputs die
    # right after the checksum)
    from anovos_amd.ops import elementwise

    cols = [torch.randn(N, device=dev) for _ in range(8)]  # 4 GB live set
    outs = elementwise.fill_nan_columns(cols, [0.0] * len(cols))
    del outs
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    outs = elementwise.fill_nan_columns(cols, [0.0] * len(cols))
    torch.cuda.synchronize()
    res["fillnan8_wall_ms"] = round((time.perf_counter() - t0) * 1000, 2)
    del outs

    print(cfg, res)
    with open("gpurun_out/alloc_probe.log", "a") as f:
        f.write(f"{cfg} {res}\n")


if __name__ == "__main__":
    main()
