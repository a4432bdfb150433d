#!/usr/bin/env python3
"""PCIe/SDMA ceiling probe: raw pinned-host <-> HBM copy bandwidth.

Establishes the hardware ceiling the offload engine's staged path is
priced against (PCIe Gen5 x16 spec: 63 GB/s each way)."""
import sys
import time

import torch


def run(size_mb=256, iters=20, streams=4):
    assert torch.cuda.is_available()
    n = size_mb * 1024 * 1024
    dev = [torch.empty(n, dtype=torch.uint8, device="cuda") for _ in range(streams)]
    host = [torch.empty(n, dtype=torch.uint8, pin_memory=True) for _ in range(streams)]
    ss = [torch.cuda.Stream() for _ in range(streams)]

    def timed(fn):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return n * iters * streams / (time.perf_counter() - t0) / 1e9

    def d2h():
        for i in range(streams):
            with torch.cuda.stream(ss[i]):
                host[i].copy_(dev[i], non_blocking=True)

    def h2d():
        for i in range(streams):
            with torch.cuda.stream(ss[i]):
                dev[i].copy_(host[i], non_blocking=True)

    d2h()
    h2d()  # warmup
    print(f"D2H pinned ({streams} streams): {timed(d2h):.1f} GB/s")
    print(f"H2D pinned ({streams} streams): {timed(h2d):.1f} GB/s")

    # single stream for reference
    def d2h1():
        host[0].copy_(dev[0], non_blocking=True)

    def h2d1():
        dev[0].copy_(host[0], non_blocking=True)

    sd = n * iters / _time1(d2h1, iters) / 1e9
    sh = n * iters / _time1(h2d1, iters) / 1e9
    print(f"D2H pinned (1 stream): {sd:.1f} GB/s")
    print(f"H2D pinned (1 stream): {sh:.1f} GB/s")
    # HBM device-to-device
    def d2d():
        dev[1].copy_(dev[0], non_blocking=True)

    print(f"D2D HBM copy: {n * iters / _time1(d2d, iters) / 1e9:.1f} GB/s (r+w)")


def _time1(fn, iters):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return time.perf_counter() - t0


if __name__ == "__main__":
    run(*(int(a) for a in sys.argv[1:]))
