#!/usr/bin/env python3
"""Image classification client (reference: image_client.py, 535 LoC;
C++ image_client.cc:86-190 does OpenCV preprocessing on the CPU).

MI355X-native difference: preprocessing (bilinear resize + scaling +
NCHW pack) runs as a CDNA4 HIP kernel (client_amd.ops image_preprocess)
writing straight into a HIP-IPC shared-memory region — the decoded
image is the only host->device copy, and the preprocessed tensor never
returns to the host. Falls back to a numpy implementation only when no
GPU is present (examples must run on CPU boxes; the GPU path raises if
the kernel extension is missing).

Accepts .ppm (P6), .npy (HWC uint8) or --synthetic images (no PIL in
this environment).
"""

import argparse
import sys

import numpy as np

import tritonclient.http as httpclient
import tritonclient.grpc as grpcclient


def load_image(path):
    if path.endswith(".npy"):
        img = np.load(path)
        assert img.dtype == np.uint8 and img.ndim == 3
        return img
    if path.endswith(".ppm"):
        with open(path, "rb") as f:
            data = f.read()
        # P6 header: magic, width height, maxval, raster
        parts = data.split(b"\n", 3)
        assert parts[0].strip() == b"P6", "only binary PPM supported"
        w, h = map(int, parts[1].split())
        raster = parts[3][-(w * h * 3):]
        return np.frombuffer(raster, dtype=np.uint8).reshape(h, w, 3).copy()
    raise SystemExit(f"unsupported image format: {path}")


def preprocess_cpu(img, size, scaling):
    ih, iw, _ = img.shape
    oh = ow = size
    sy, sx = ih / oh, iw / ow
    fy = (np.arange(oh) + 0.5) * sy - 0.5
    fx = (np.arange(ow) + 0.5) * sx - 0.5
    y0 = np.clip(np.floor(fy).astype(int), 0, ih - 1)
    x0 = np.clip(np.floor(fx).astype(int), 0, iw - 1)
    y1 = np.minimum(ih - 1, y0 + 1)
    x1 = np.minimum(iw - 1, x0 + 1)
    wy = np.where(fy < 0, 0.0, fy - np.floor(fy))[:, None]
    wx = np.where(fx < 0, 0.0, fx - np.floor(fx))[None, :]
    out = np.empty((3, oh, ow), dtype=np.float32)
    for c in range(3):
        p = img[:, :, c].astype(np.float32)
        out[c] = ((1 - wy) * ((1 - wx) * p[y0][:, x0] + wx * p[y0][:, x1])
                  + wy * ((1 - wx) * p[y1][:, x0] + wx * p[y1][:, x1]))
    if scaling == "INCEPTION":
        out = out / 127.5 - 1.0
    elif scaling == "VGG":
        means = np.array([104.0, 117.0, 123.0], dtype=np.float32)
        out = out - means[:, None, None]
    return out


def preprocess_gpu(img, size, scaling, region, offset=0):
    """CDNA4 preprocess kernel into a HIP-shm region (fp32 CHW)."""
    from client_amd.ops import hip_runtime as hr

    ih, iw, _ = img.shape
    mode = {"NONE": 0, "INCEPTION": 1, "VGG": 2}[scaling]
    mean = [104.0, 117.0, 123.0] if scaling == "VGG" else [0.0, 0.0, 0.0]
    src = hr.malloc(region._device_id, img.nbytes)
    try:
        hr.memcpy_h2d(src, img.reshape(-1), img.nbytes, region._device_id, False)
        hr.image_preprocess(
            src, region.ptr() + offset, ih, iw, size, size, mode, False,
            mean, [1.0, 1.0, 1.0], region._device_id, True,
        )
    finally:
        hr.free(src)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("image", nargs="?", default=None,
                        help=".ppm/.npy image; omit for --synthetic")
    parser.add_argument("-m", "--model-name", default="resnet50")
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    parser.add_argument("-i", "--protocol", default="http",
                        choices=["http", "grpc"])
    parser.add_argument("-s", "--scaling", default="INCEPTION",
                        choices=["NONE", "INCEPTION", "VGG"])
    parser.add_argument("-c", "--classes", type=int, default=3,
                        help="top-k classification results")
    parser.add_argument("--size", type=int, default=224)
    parser.add_argument("--synthetic", action="store_true")
    parser.add_argument("--gpu", action="store_true",
                        help="force the HIP preprocess + HIP-shm path")
    args = parser.parse_args()

    if args.synthetic or args.image is None:
        img = np.random.randint(0, 256, (480, 640, 3), dtype=np.uint8)
    else:
        img = load_image(args.image)

    from client_amd.ops import gpu_available

    use_gpu = args.gpu or gpu_available()

    if args.protocol == "http":
        client = httpclient.InferenceServerClient(args.url)
        mod = httpclient
    else:
        client = grpcclient.InferenceServerClient(args.url)
        mod = grpcclient

    try:
        shape = [1, 3, args.size, args.size]
        nbytes = int(np.prod(shape)) * 4
        inp = mod.InferInput("INPUT0", shape, "FP32")
        region = None
        if use_gpu:
            import client_amd.utils.hip_shared_memory as hipshm

            region = hipshm.create_shared_memory_region("img_in", nbytes, 0)
            preprocess_gpu(img, args.size, args.scaling, region)
            raw = (hipshm.get_raw_handle_bytes(region)
                   if args.protocol == "grpc" else hipshm.get_raw_handle(region))
            client.register_cuda_shared_memory("img_in", raw, 0, nbytes)
            inp.set_shared_memory("img_in", nbytes)
        else:
            tensor = preprocess_cpu(img, args.size, args.scaling)[None]
            inp.set_data_from_numpy(tensor.astype(np.float32))

        outputs = [mod.InferRequestedOutput("OUTPUT0",
                                            class_count=args.classes)
                   if args.protocol == "http"
                   else mod.InferRequestedOutput("OUTPUT0",
                                                 class_count=args.classes)]
        result = client.infer(args.model_name, [inp], outputs=outputs)
        classes = result.as_numpy("OUTPUT0")
        for row in classes.reshape(-1, args.classes):
            for entry in row:
                score, idx = entry.decode().split(":")
                print(f"    {float(score):.6f} ({idx})")
        if region is not None:
            client.unregister_cuda_shared_memory("img_in")
            import client_amd.utils.hip_shared_memory as hipshm

            hipshm.destroy_shared_memory_region(region)
        print("PASS: image client")
    finally:
        client.close()


if __name__ == "__main__":
    main()
