#!/usr/bin/env python3
"""Ensemble image classification: send the RAW u8 image to the server's
preprocess->classify ensemble — preprocessing runs server-side on the
MI355X with the CDNA4 kernel (reference: ensemble_image_client.cc sends
raw JPEG bytes to a DALI ensemble).

Server: python -m client_amd.server --http-port 8000 \
            --models resnet50,ensemble_image --device cuda:0
"""
import argparse

import numpy as np

import tritonclient.http as httpclient

if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("image", nargs="?", default=None, help=".npy u8 HWC")
    parser.add_argument("-u", "--url", default="127.0.0.1:8000")
    parser.add_argument("-c", "--classes", type=int, default=3)
    args = parser.parse_args()

    if args.image:
        img = np.load(args.image)
    else:
        img = np.random.randint(0, 256, (480, 640, 3), dtype=np.uint8)

    with httpclient.InferenceServerClient(args.url) as client:
        inp = httpclient.InferInput("IMAGE", list(img.shape), "UINT8")
        inp.set_data_from_numpy(img)
        outputs = [httpclient.InferRequestedOutput(
            "OUTPUT0", class_count=args.classes)]
        result = client.infer("ensemble_image", [inp], outputs=outputs)
        classes = result.as_numpy("OUTPUT0")
        for entry in classes.reshape(-1)[: args.classes]:
            score, idx = entry.decode().split(":")
            print(f"    {float(score):.6f} ({idx})")
        print("PASS: ensemble image client")
