"""PCIe/host-link probe for the host-pointer (drop-in plugin) path design:
measures pageable vs pinned H2D/D2H, duplex overlap, and CPU memcpy rates
so the staging strategy in ecx_*_chunks_host is chosen from data."""
import json
import time

import numpy as np
import torch

GIB = 1 << 30
N = 1 << 28  # 256 MiB


def rate(fn, reps=8):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return N * reps / GIB / (time.perf_counter() - t0)


def main():
    torch.cuda.set_device(0)
    dev = torch.empty(N, dtype=torch.uint8, device="cuda")
    dev2 = torch.empty(N, dtype=torch.uint8, device="cuda")
    pageable = torch.empty(N, dtype=torch.uint8)
    pinned = torch.empty(N, dtype=torch.uint8, pin_memory=True)
    out = {}
    out["h2d_pageable"] = rate(lambda: dev.copy_(pageable, non_blocking=True))
    out["h2d_pinned"] = rate(lambda: dev.copy_(pinned, non_blocking=True))
    out["d2h_pageable"] = rate(lambda: pageable.copy_(dev, non_blocking=True))
    out["d2h_pinned"] = rate(lambda: pinned.copy_(dev, non_blocking=True))
    # duplex: H2D on stream A, D2H on stream B concurrently
    sa, sb = torch.cuda.Stream(), torch.cuda.Stream()
    pin2 = torch.empty(N, dtype=torch.uint8, pin_memory=True)

    def duplex():
        with torch.cuda.stream(sa):
            dev.copy_(pinned, non_blocking=True)
        with torch.cuda.stream(sb):
            pin2.copy_(dev2, non_blocking=True)
    duplex(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(8):
        duplex()
    torch.cuda.synchronize()
    out["duplex_total"] = 2 * N * 8 / GIB / (time.perf_counter() - t0)
    # CPU memcpy pageable->pinned, single thread (numpy)
    src = np.frombuffer(pageable.numpy(), dtype=np.uint8)
    dst = np.frombuffer(pinned.numpy(), dtype=np.uint8)
    t0 = time.perf_counter()
    for _ in range(4):
        np.copyto(dst, src)
    out["cpu_memcpy_1t"] = N * 4 / GIB / (time.perf_counter() - t0)
    print(json.dumps({k: round(v, 2) for k, v in out.items()}))


if __name__ == "__main__":
    main()
