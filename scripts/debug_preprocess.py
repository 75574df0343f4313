"""Diagnose image_preprocess kernel vs numpy reference: mismatch count,
max diff, and the (c,y,x) locations of the worst offenders."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import numpy as np

from client_amd.ops import hip_runtime as hr
from test_hip_shm_gpu import _ref_preprocess  # noqa: E402

ih, iw, oh, ow = 300, 451, 224, 224
rng = np.random.default_rng(0)
img = rng.integers(0, 256, (ih, iw, 3), dtype=np.uint8)
mean = [104.0, 117.0, 123.0]
std = [1.0, 1.0, 1.0]
src = hr.malloc(0, img.nbytes)
dst = hr.malloc(0, 3 * oh * ow * 4)
hr.memcpy_h2d(src, img.reshape(-1), img.nbytes, 0, True)
for mode in (0, 1, 2):
    hr.image_preprocess(src, dst, ih, iw, oh, ow, mode, False, mean, std, 0, True)
    out = np.empty(3 * oh * ow, dtype=np.float32)
    hr.memcpy_d2h_into(dst, out.view(np.uint8), out.nbytes, 0)
    out = out.reshape(3, oh, ow)
    ref = _ref_preprocess(img, oh, ow, mode, mean, std)
    diff = np.abs(out - ref)
    bad = np.argwhere(diff > 1e-3)
    print(f"mode={mode} maxdiff={diff.max():.6f} n_bad={len(bad)}")
    for c, y, x in bad[:8]:
        fy = (y + 0.5) * ih / oh - 0.5
        fx = (x + 0.5) * iw / ow - 0.5
        print(f"  c={c} y={y} x={x} out={out[c,y,x]:.4f} ref={ref[c,y,x]:.4f} "
              f"fy={fy!r} fx={fx!r}")
hr.free(src)
hr.free(dst)
