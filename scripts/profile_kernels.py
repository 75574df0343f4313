"""Bandwidth measurement for the CDNA4 cast/pack/preprocess kernels.

Run under rocprofv3 for per-kernel stats, or standalone for effective
GB/s (wall-clock around N iterations with stream sync):

    python scripts/profile_kernels.py
    rocprofv3 --kernel-trace --stats -- python scripts/profile_kernels.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from client_amd.ops import hip_runtime as hr

ITERS = 50


def bench(name, fn, bytes_moved):
    fn()  # warmup
    hr.device_sync()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        fn()
    hr.device_sync()
    dt = (time.perf_counter() - t0) / ITERS
    gbps = bytes_moved / dt / 1e9
    print(f"{name:28s} {dt*1e6:9.1f} us  {gbps:8.1f} GB/s")
    return gbps


def main():
    n = 256 * 1024 * 1024  # 256M elements: 1 GiB fp32
    a = hr.malloc(0, n * 4)
    b = hr.malloc(0, n * 4)
    print(f"n = {n} elements")

    results = {}
    results["cast_fp32_bf16"] = bench(
        "cast_fp32_bf16 (4B+2B/elem)",
        lambda: hr.cast_fp32_bf16(a, b, n, 0, True, 0), n * 6,
    )
    results["cast_bf16_fp32"] = bench(
        "cast_bf16_fp32 (2B+4B/elem)",
        lambda: hr.cast_bf16_fp32(a, b, n, 0, True, 0), n * 6,
    )
    results["cast_fp32_fp8"] = bench(
        "cast_fp32_fp8e4m3 (4B+1B)",
        lambda: hr.cast_fp32_fp8e4m3(a, b, n, 0, True, 0), n * 5,
    )
    results["cast_fp8_fp32"] = bench(
        "cast_fp8e4m3_fp32 (1B+4B)",
        lambda: hr.cast_fp8e4m3_fp32(a, b, n, 0, True, 0), n * 5,
    )
    results["memcpy_d2d"] = bench(
        "memcpy_d2d (4B+4B/elem)",
        lambda: hr.memcpy_d2d(b, a, n * 4, 0, True), n * 8,
    )
    # gather_pack: transpose-like strided read (worst case), 2-D
    rows, cols = 16384, 16384
    results["gather_pack_T"] = bench(
        "gather_pack fp32 16k^2 T",
        lambda: hr.gather_pack(a, b, 4, [rows, cols], [1, rows], 0, True),
        rows * cols * 8,
    )
    # bf16 transpose through the same tiled path (2B elements)
    results["gather_pack_T_bf16"] = bench(
        "gather_pack bf16 16k^2 T",
        lambda: hr.gather_pack(a, b, 2, [rows, cols], [1, rows], 0, True),
        rows * cols * 4,
    )
    # image preprocess: 720p -> 224x224
    ih, iw = 720, 1280
    img = hr.malloc(0, ih * iw * 3)
    out = hr.malloc(0, 3 * 224 * 224 * 4)
    results["image_preprocess"] = bench(
        "image_preprocess 720p->224",
        lambda: hr.image_preprocess(img, out, ih, iw, 224, 224, 1, False,
                                    [0, 0, 0], [1, 1, 1], 0, True),
        ih * iw * 3 + 3 * 224 * 224 * 4,
    )
    hr.free(img)
    # batched preprocess: 64 images per launch (launch-bound case)
    nimg = 64
    imgb = hr.malloc(0, nimg * ih * iw * 3)
    outb = hr.malloc(0, nimg * 3 * 224 * 224 * 4)
    results["image_preprocess_batched"] = bench(
        f"image_preprocess x{nimg}/launch",
        lambda: hr.image_preprocess_batched(imgb, outb, nimg, ih, iw, 224,
                                            224, 1, False, [0, 0, 0],
                                            [1, 1, 1], 0, True),
        nimg * (ih * iw * 3 + 3 * 224 * 224 * 4),
    )
    hr.free(imgb)
    hr.free(outb)
    hr.free(out)
    hr.free(a)
    hr.free(b)

    import json

    print(json.dumps(results))


if __name__ == "__main__":
    main()
