"""Launch a client_amd KServe-v2 server process.

    python -m client_amd.server --grpc-port 8001 --http-port 8000 \
        --models identity_fp32,simple,resnet50 --device cuda:0

Used by the GPU tests and bench.py: HIP-IPC handles can only be opened
by a *different* process, so the server always runs out-of-process.
"""

import argparse
import signal
import threading


def build_core(model_names, device="cuda:0", dtype="bf16",
               decode_max_batch=8):
    from . import (
        AddSubModel,
        IdentityModel,
        InferenceCore,
        RepeatModel,
        SequenceModel,
        TorchModel,
    )

    core = InferenceCore()
    for name in model_names:
        if name == "identity_fp32":
            core.add_model(IdentityModel("identity_fp32", "FP32"))
        elif name == "identity_bf16":
            core.add_model(IdentityModel("identity_bf16", "BF16"))
        elif name == "identity_bytes":
            core.add_model(IdentityModel("identity_bytes", "BYTES"))
        elif name == "simple":
            core.add_model(AddSubModel("simple", "INT32", (-1, 16)))
        elif name == "simple_string":
            core.add_model(AddSubModel("simple_string", "BYTES", (-1, 16)))
        elif name == "sequence_accumulate":
            core.add_model(SequenceModel())
        elif name == "repeat_int32":
            core.add_model(RepeatModel())
        elif name == "resnet50":
            import torch

            from ..models import resnet50

            tdt = {"bf16": torch.bfloat16, "fp16": torch.float16,
                   "fp32": torch.float32}[dtype]
            io_dt = {"bf16": "BF16", "fp16": "FP16", "fp32": "FP32"}[dtype]
            core.add_model(
                TorchModel(
                    "resnet50",
                    resnet50(),
                    inputs=[("INPUT0", io_dt, [-1, 3, 224, 224])],
                    outputs=[("OUTPUT0", io_dt, [-1, 1000])],
                    device=device,
                    dtype=tdt,
                )
            )
        elif name == "densenet121":
            import torch

            from ..models import densenet121

            tdt = {"bf16": torch.bfloat16, "fp16": torch.float16,
                   "fp32": torch.float32}[dtype]
            io_dt = {"bf16": "BF16", "fp16": "FP16", "fp32": "FP32"}[dtype]
            core.add_model(
                TorchModel(
                    "densenet121",
                    densenet121(),
                    inputs=[("INPUT0", io_dt, [-1, 3, 224, 224])],
                    outputs=[("OUTPUT0", io_dt, [-1, 1000])],
                    device=device,
                    dtype=tdt,
                )
            )
        elif name in ("bert_large", "bert_tiny"):
            import torch

            from ..models.bert import bert_large, bert_tiny

            tdt = {"bf16": torch.bfloat16, "fp16": torch.float16,
                   "fp32": torch.float32}[dtype]
            io_dt = {"bf16": "BF16", "fp16": "FP16", "fp32": "FP32"}[dtype]
            module = bert_large() if name == "bert_large" else bert_tiny()
            hidden = 1024 if name == "bert_large" else 32
            core.add_model(
                TorchModel(
                    name,
                    module,
                    inputs=[("input_ids", "INT64", [-1, -1])],
                    outputs=[("pooled", io_dt, [-1, hidden])],
                    device=device,
                    dtype=tdt if device.startswith("cuda") else None,
                    # per-signature capture: each new (batch, seq) shape
                    # captures once (~0.5 s) then replays; serving
                    # traffic with few shapes amortizes immediately
                    use_graph=True,
                )
            )
        elif name in ("llama3_8b", "llama_tiny"):
            import torch

            from . import GenerateModel
            from ..models.llama import (
                LlamaModel,
                llama3_8b_config,
                llama_tiny_config,
            )

            cfg = (llama3_8b_config() if name == "llama3_8b"
                   else llama_tiny_config())
            tdt = {"bf16": torch.bfloat16, "fp16": torch.float16,
                   "fp32": torch.float32}[dtype]
            use_gpu = device.startswith("cuda")
            # construct directly on the target device: random-init of 8B
            # params on the host would take minutes
            with torch.device(device if use_gpu else "cpu"):
                module = LlamaModel(cfg)
            core.add_model(
                GenerateModel(name, module, device=device,
                              dtype=tdt if use_gpu else None,
                              max_batch=decode_max_batch)
            )
        elif name == "ensemble_image":
            # preprocess (HIP kernel on GPU) -> resnet50 pipeline; the
            # reference's ensemble_image_client analog. Requires
            # resnet50 earlier in --models.
            from . import EnsembleModel, PreprocessModel

            if "resnet50" not in core.models:
                raise SystemExit("ensemble_image requires resnet50 in --models")
            classifier = core.models["resnet50"]
            pre = PreprocessModel("preprocess_inception", device=device
                                  if device.startswith("cuda") else "cpu")
            core.add_model(pre)
            io_dt = {"bf16": "BF16", "fp16": "FP16", "fp32": "FP32"}[dtype]
            core.add_model(EnsembleModel(
                "ensemble_image",
                inputs=[("IMAGE", "UINT8", [-1, -1, 3])],
                outputs=[("OUTPUT0", io_dt if device.startswith("cuda")
                          else "FP32", [-1, 1000])],
                steps=[
                    (pre, {"IMAGE": "IMAGE"}, {"TENSOR": "preprocessed"}),
                    (classifier, {"INPUT0": "preprocessed"},
                     {"OUTPUT0": "OUTPUT0"}),
                ],
            ))
        elif name == "identity_gpu":
            # GPU identity via TorchModel (device fast path test target)
            import torch

            core.add_model(
                TorchModel(
                    "identity_gpu",
                    torch.nn.Identity(),
                    inputs=[("INPUT0", "FP32", [-1])],
                    outputs=[("OUTPUT0", "FP32", [-1])],
                    device=device,
                )
            )
        else:
            raise SystemExit(f"unknown model '{name}'")
    return core


def main(argv=None):
    import os

    # fast MIOpen kernel selection: skip exhaustive conv tuning at start
    os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
    # dmabuf IPC is the only mode the host driver supports; must be set
    # before the HIP runtime initializes or hipIpc* fails EINVAL
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    parser = argparse.ArgumentParser("client_amd.server")
    parser.add_argument("--http-port", type=int, default=0,
                        help="0 disables HTTP")
    parser.add_argument("--grpc-port", type=int, default=0,
                        help="0 disables gRPC; -1 picks an ephemeral port")
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--models", default="identity_fp32,simple")
    parser.add_argument("--device", default="cuda:0")
    parser.add_argument("--dtype", default="bf16",
                        choices=["bf16", "fp16", "fp32"])
    parser.add_argument("--grpc-workers", type=int, default=8)
    parser.add_argument("--dynamic-batching", action="store_true",
                        help="enable dynamic batching on torch models")
    parser.add_argument("--preferred-batch-size", type=int, default=32)
    parser.add_argument("--max-queue-delay-us", type=int, default=500)
    parser.add_argument("--decode-max-batch", type=int, default=8,
                        help="continuous-batching slots for generate models")
    parser.add_argument("--model-warmup", action="store_true",
                        help="pre-capture hipGraphs / prime MIOpen for the "
                             "serving batch sizes before READY is printed")
    args = parser.parse_args(argv)

    core = build_core(
        [m for m in args.models.split(",") if m], args.device, args.dtype,
        decode_max_batch=args.decode_max_batch,
    )
    if args.dynamic_batching:
        for model in core.models.values():
            if hasattr(model, "enable_dynamic_batching"):
                model.enable_dynamic_batching(
                    args.preferred_batch_size, args.max_queue_delay_us
                )
    if args.model_warmup:
        sizes = (8, args.preferred_batch_size) if args.dynamic_batching \
            else (8,)
        for model in core.models.values():
            if hasattr(model, "warmup"):
                model.warmup(sizes)

    stoppers = []
    if args.grpc_port != 0:
        from .grpc_server import GrpcServer

        port = max(args.grpc_port, 0)
        gs = GrpcServer(core, host=args.host, port=port,
                        max_workers=args.grpc_workers)
        gs.start()
        stoppers.append(lambda: gs.stop(grace=1))
        print(f"GRPC_READY {gs.port}", flush=True)
    if args.http_port != 0:
        from .http_server import HttpServer

        port = max(args.http_port, 0)
        hs = HttpServer(core, host=args.host, port=port)
        stop = hs.serve_forever_in_thread()
        stoppers.append(stop)
        print(f"HTTP_READY {hs.port}", flush=True)

    done = threading.Event()

    def _sig(*_):
        done.set()

    signal.signal(signal.SIGTERM, _sig)
    signal.signal(signal.SIGINT, _sig)
    done.wait()
    for stop in stoppers:
        try:
            stop()
        except Exception:
            pass


if __name__ == "__main__":
    main()
