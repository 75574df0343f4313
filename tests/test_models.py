"""Model-family tests (CPU, tiny configs): BERT encoder, Llama decoder
with KV cache, and decoupled token streaming end to end over gRPC."""

import queue

import numpy as np
import pytest

torch = pytest.importorskip("torch")


def test_resnet50_forward():
    from client_amd.models import resnet50

    m = resnet50()
    with torch.inference_mode():
        y = m(torch.randn(2, 3, 224, 224))
    assert y.shape == (2, 1000)


def test_bert_tiny_forward():
    from client_amd.models.bert import bert_tiny

    m = bert_tiny()
    with torch.inference_mode():
        y = m(torch.randint(0, 128, (3, 16)))
    assert y.shape == (3, 32)


def test_llama_kv_cache_matches_full_recompute():
    """Greedy decode with KV cache must produce the same tokens as
    recomputing the full sequence each step (numerics: fp32 tiny)."""
    from client_amd.models.llama import LlamaModel, llama_tiny_config

    torch.manual_seed(0)
    cfg = llama_tiny_config()
    m = LlamaModel(cfg).eval()
    prompt = torch.randint(0, cfg.vocab_size, (1, 8))
    n_new = 6

    cached_tokens = [int(t[0]) for t in m.generate(prompt, n_new)]

    # reference: full forward each step, no cache reuse across steps
    ref_tokens = []
    seq = prompt.clone()
    with torch.inference_mode():
        for _ in range(n_new):
            kv = m.make_kv_cache(1, seq.device, next(m.parameters()).dtype)
            logits = m.forward_step(seq, 0, kv)
            nxt = int(logits.argmax(-1)[0])
            ref_tokens.append(nxt)
            seq = torch.cat([seq, torch.tensor([[nxt]])], dim=1)
    assert cached_tokens == ref_tokens


def test_llama_generate_batch():
    from client_amd.models.llama import LlamaModel, llama_tiny_config

    torch.manual_seed(1)
    m = LlamaModel(llama_tiny_config()).eval()
    prompt = torch.randint(0, 256, (2, 4))
    toks = list(m.generate(prompt, 3))
    assert len(toks) == 3
    assert toks[0].shape == (2,)


@pytest.fixture(scope="module")
def llama_grpc_server():
    from client_amd.server.__main__ import build_core
    from client_amd.server.grpc_server import GrpcServer

    core = build_core(["llama_tiny"], device="cpu", dtype="fp32")
    server = GrpcServer(core, host="127.0.0.1", port=0)
    server.start()
    yield "127.0.0.1", server.port
    server.stop(grace=1)


def test_llama_decoupled_token_stream(llama_grpc_server):
    """BASELINE config 5 shape on CPU: decoupled gRPC stream delivers one
    response per generated token, final flagged."""
    import client_amd.grpc as grpcclient

    host, port = llama_grpc_server
    client = grpcclient.InferenceServerClient(f"{host}:{port}")
    results = queue.Queue()
    client.start_stream(callback=lambda result, error: results.put((result, error)))
    try:
        ids = np.random.randint(0, 256, 8).astype(np.int64)
        inputs = [
            grpcclient.InferInput("input_ids", [8], "INT64"),
            grpcclient.InferInput("max_tokens", [1], "INT32"),
        ]
        inputs[0].set_data_from_numpy(ids)
        inputs[1].set_data_from_numpy(np.array([5], dtype=np.int32))
        client.async_stream_infer(
            "llama_tiny", inputs, enable_empty_final_response=True
        )
        tokens = []
        while True:
            result, error = results.get(timeout=60)
            assert error is None
            if result.is_final_response():
                break
            tokens.append(int(result.as_numpy("token_id")[0]))
            idx = int(result.as_numpy("index")[0])
            assert idx == len(tokens) - 1
        assert len(tokens) == 5
    finally:
        client.stop_stream()
        client.close()


def test_bert_tiny_served(llama_grpc_server):
    """BERT over gRPC with host tensors (CPU)."""
    from client_amd.server.__main__ import build_core
    from client_amd.server.grpc_server import GrpcServer
    import client_amd.grpc as grpcclient

    core = build_core(["bert_tiny"], device="cpu", dtype="fp32")
    server = GrpcServer(core, host="127.0.0.1", port=0)
    server.start()
    try:
        client = grpcclient.InferenceServerClient(f"127.0.0.1:{server.port}")
        ids = np.random.randint(0, 128, (2, 16)).astype(np.int64)
        inp = grpcclient.InferInput("input_ids", [2, 16], "INT64")
        inp.set_data_from_numpy(ids)
        result = client.infer("bert_tiny", [inp])
        out = result.as_numpy("pooled")
        assert out.shape == (2, 32)
        assert np.isfinite(out).all()
        client.close()
    finally:
        server.stop(grace=1)


def test_decode_scheduler_matches_sequential():
    """Continuous-batched decode must produce exactly the tokens that
    per-stream sequential generate produces (row independence of
    forward_decode_batch)."""
    from client_amd.models.llama import LlamaModel, llama_tiny_config
    from client_amd.server.decode_scheduler import DecodeScheduler

    torch.manual_seed(3)
    cfg = llama_tiny_config()
    m = LlamaModel(cfg).eval()
    prompts = [
        torch.randint(0, cfg.vocab_size, (1, n)) for n in (5, 9, 3)
    ]
    expected = [
        [int(t[0]) for t in m.generate(p, 6)] for p in prompts
    ]

    sched = DecodeScheduler(m, max_batch=4, device="cpu",
                            dtype=torch.float32)
    try:
        queues = [
            sched.submit(p[0].numpy(), 6) for p in prompts
        ]
        got = []
        for q in queues:
            toks = []
            while True:
                t = q.get(timeout=60)
                if t is sched.END:
                    break
                toks.append(t)
            got.append(toks)
        assert got == expected
    finally:
        sched.shutdown()


def test_decode_scheduler_concurrent_submit():
    """Streams submitted while others are mid-decode join the batch and
    still produce the sequential-equivalent tokens."""
    import threading
    import time as _time

    from client_amd.models.llama import LlamaModel, llama_tiny_config
    from client_amd.server.decode_scheduler import DecodeScheduler

    torch.manual_seed(4)
    cfg = llama_tiny_config()
    m = LlamaModel(cfg).eval()
    prompts = [torch.randint(0, cfg.vocab_size, (1, 4 + i)) for i in range(3)]
    expected = [[int(t[0]) for t in m.generate(p, 8)] for p in prompts]

    sched = DecodeScheduler(m, max_batch=2, device="cpu",  # forces queuing
                            dtype=torch.float32)
    results = [None] * 3

    def run(i):
        if i:
            _time.sleep(0.05 * i)
        q = sched.submit(prompts[i][0].numpy(), 8)
        toks = []
        while True:
            t = q.get(timeout=60)
            if t is sched.END:
                break
            toks.append(t)
        results[i] = toks

    try:
        threads = [threading.Thread(target=run, args=(i,)) for i in range(3)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(60)
        assert results == expected
    finally:
        sched.shutdown()


def test_ensemble_preprocess_classify():
    """Ensemble pipeline: raw u8 image -> preprocess -> classifier, one
    request (reference ensemble_image_client shape), CPU."""
    from client_amd.server import (
        EnsembleModel,
        PreprocessModel,
        InferenceCore,
        TorchModel,
    )

    pre = PreprocessModel("pre", size=32)
    tiny_classifier = TorchModel(
        "cls", torch.nn.Sequential(torch.nn.Flatten(), torch.nn.Linear(3 * 32 * 32, 10)),
        inputs=[("INPUT0", "FP32", [-1, 3, 32, 32])],
        outputs=[("OUTPUT0", "FP32", [-1, 10])],
        device="cpu", use_graph=False,
    )
    ens = EnsembleModel(
        "ens",
        inputs=[("IMAGE", "UINT8", [-1, -1, 3])],
        outputs=[("OUTPUT0", "FP32", [-1, 10])],
        steps=[
            (pre, {"IMAGE": "IMAGE"}, {"TENSOR": "t"}),
            (tiny_classifier, {"INPUT0": "t"}, {"OUTPUT0": "OUTPUT0"}),
        ],
    )
    core = InferenceCore()
    core.add_model(ens)
    img = np.random.randint(0, 256, (48, 64, 3), dtype=np.uint8)
    request = {
        "inputs": [{
            "name": "IMAGE", "datatype": "UINT8", "shape": list(img.shape),
            "parameters": {"binary_data_size": img.nbytes},
        }],
        "parameters": {"binary_data_output": True},
    }
    response, parts = core.infer("ens", request, img.tobytes())
    assert response["outputs"][0]["shape"] == [1, 10]
    out = np.frombuffer(parts[0], dtype=np.float32)
    assert np.isfinite(out).all()


def test_decode_batch_padded_maxlen_equivalent():
    """forward_decode_batch with a padded (bucketed) max_len must equal
    the exact-length result — the property hipGraph bucketing relies on."""
    from client_amd.models.llama import LlamaModel, llama_tiny_config

    torch.manual_seed(5)
    cfg = llama_tiny_config()
    m = LlamaModel(cfg).eval()
    b = 3
    kv1 = m.make_kv_cache(b, "cpu", torch.float32)
    kv2 = m.make_kv_cache(b, "cpu", torch.float32)
    with torch.inference_mode():
        # seed some history at different lengths per row
        for row, plen in enumerate((4, 7, 2)):
            ids = torch.randint(0, cfg.vocab_size, (1, plen))
            row_kv1 = [(ck[row:row+1], cv[row:row+1]) for ck, cv in kv1]
            row_kv2 = [(ck[row:row+1], cv[row:row+1]) for ck, cv in kv2]
            m.forward_step(ids, 0, row_kv1)
            m.forward_step(ids, 0, row_kv2)
        tokens = torch.randint(0, cfg.vocab_size, (b, 1))
        pos = torch.tensor([4, 7, 2])
        exact = m.forward_decode_batch(tokens, pos, kv1)
        padded = m.forward_decode_batch(tokens, pos, kv2, max_len=64)
    torch.testing.assert_close(exact, padded, rtol=1e-5, atol=1e-5)


def test_chunked_prefill_matches_full():
    """Chunked prefill (continuation chunks with offset causal mask)
    must produce exactly the tokens full-prompt prefill produces."""
    from client_amd.models.llama import LlamaModel, llama_tiny_config
    from client_amd.server.decode_scheduler import DecodeScheduler

    torch.manual_seed(9)
    cfg = llama_tiny_config()
    m = LlamaModel(cfg).eval()
    prompts = [torch.randint(0, cfg.vocab_size, (1, n)) for n in (23, 7, 40)]
    expected = [[int(t[0]) for t in m.generate(p, 6)] for p in prompts]

    # tiny chunk forces multiple continuation chunks per prompt
    sched = DecodeScheduler(m, max_batch=4, device="cpu",
                            dtype=torch.float32, prefill_chunk=8)
    try:
        queues = [sched.submit(p[0].numpy(), 6) for p in prompts]
        got = []
        for q in queues:
            toks = []
            while True:
                t = q.get(timeout=60)
                if t is sched.END:
                    break
                toks.append(t)
            got.append(toks)
        assert got == expected
    finally:
        sched.shutdown()


def test_mid_prefill_rows_not_corrupted_by_decode():
    """Regression: a decode step scatters K/V for EVERY batch row
    (forward_decode_batch), and inactive rows used to be pointed at
    position 1 — overwriting cache position 1 of rows still in chunked
    prefill, silently corrupting their generations (diverged in 2/5
    seeds with this exact shape before the scratch-position fix). A
    short prompt becomes ACTIVE and decodes while long prompts are
    mid-prefill; every stream must still match sequential generate,
    across multiple seeds."""
    from client_amd.models.llama import LlamaModel, llama_tiny_config
    from client_amd.server.decode_scheduler import DecodeScheduler

    for seed in range(5):
        torch.manual_seed(seed)
        cfg = llama_tiny_config()
        m = LlamaModel(cfg).eval()
        # short prompt finishes prefill in one chunk and starts
        # decoding; 30/45-token prompts stay in PREFILL for several
        # decode steps (chunk=8)
        prompts = [torch.randint(0, cfg.vocab_size, (1, n))
                   for n in (4, 30, 45)]
        expected = [[int(t[0]) for t in m.generate(p, 8)] for p in prompts]

        sched = DecodeScheduler(m, max_batch=4, device="cpu",
                                dtype=torch.float32, prefill_chunk=8)
        try:
            queues = [sched.submit(p[0].numpy(), 8) for p in prompts]
            got = []
            for q in queues:
                toks = []
                while True:
                    t = q.get(timeout=60)
                    if t is sched.END:
                        break
                    toks.append(t)
                got.append(toks)
            assert got == expected, f"divergence at seed {seed}"
        finally:
            sched.shutdown()


def test_shm_offset_bounds_rejected():
    """Client-controlled shared_memory_offset/byte_size windows outside
    the registered region must be rejected before any copy (the
    reference server rejects out-of-range shm access; an unchecked
    offset would index arbitrary server memory)."""
    import multiprocessing.shared_memory as mpshm

    from client_amd.server.core import InferenceCore, InferenceError

    core = InferenceCore()
    seg = mpshm.SharedMemory(create=True, size=64)
    try:
        key = "/" + seg.name.lstrip("/")
        core.shm.register_system("r0", key, 0, 64)

        def infer(offset=0, byte_size=32):
            req = {
                "inputs": [{
                    "name": "IN", "datatype": "FP32", "shape": [8],
                    "parameters": {
                        "shared_memory_region": "r0",
                        "shared_memory_byte_size": byte_size,
                        "shared_memory_offset": offset,
                    },
                }],
            }
            core._input_array(req["inputs"][0], b"", 0)

        infer(offset=0, byte_size=32)       # in-bounds: fine
        infer(offset=32, byte_size=32)      # exactly fits: fine
        for off, size in ((33, 32), (0, 65), (-1, 32), (0, -1),
                          (2**40, 32)):
            with pytest.raises(InferenceError):
                infer(offset=off, byte_size=size)
    finally:
        core.shm.unregister_system()
        seg.close()
        seg.unlink()


def test_torchmodel_warmup_cpu_noop_and_cli():
    """warmup() is a no-op off-GPU and the --model-warmup CLI path wires
    through build_core without error."""
    import numpy as np
    import torch

    from client_amd.server import TorchModel

    model = TorchModel(
        "wident", torch.nn.Linear(16, 16), device="cpu",
        inputs=[("INPUT0", "FP32", [-1, 16])],
        outputs=[("OUTPUT0", "FP32", [-1, 16])],
    )
    model.warmup((2, 4))  # no-op on cpu, must not raise
    out = model.execute(
        {"INPUT0": np.zeros((2, 16), dtype=np.float32)}, {})
    assert out["OUTPUT0"].shape == (2, 16)


def test_resnet_fused_eltwise_matches_plain_fold():
    """The FusedBottleneck rewrite (bias+relu and bias+residual+relu as
    one-pass kernels, CPU fallback here) must match the plain BN-fold
    numerics, with no conv left carrying a bias for torch to run as a
    separate elementwise pass."""
    import copy

    import torch

    from client_amd.models.resnet import (
        FusedBottleneck,
        ResNet50,
        fold_batchnorm,
    )

    torch.manual_seed(1)
    m = ResNet50().eval()
    for mod in m.modules():
        if isinstance(mod, torch.nn.BatchNorm2d):
            mod.running_mean.uniform_(-0.5, 0.5)
            mod.running_var.uniform_(0.5, 2.0)
            mod.weight.data.uniform_(0.5, 1.5)
            mod.bias.data.uniform_(-0.3, 0.3)
    x = torch.randn(2, 3, 64, 64)
    with torch.inference_mode():
        plain = fold_batchnorm(copy.deepcopy(m), fuse_eltwise=False)
        fused = fold_batchnorm(copy.deepcopy(m), fuse_eltwise=True)
        assert torch.allclose(plain(x), fused(x), atol=1e-4, rtol=1e-4)
    blocks = [b for b in fused.modules() if isinstance(b, FusedBottleneck)]
    assert len(blocks) == 16  # 3 + 4 + 6 + 3
    for b in blocks:
        assert b.conv1.bias is None and b.conv2.bias is None
        assert b.conv3.bias is None
        if b.ds_conv is not None:
            assert b.ds_conv.bias is None
    assert fused.conv1.bias is None  # stem bias fused into BiasAct


def test_resnet_batchnorm_folding_exact():
    """fold_batchnorm is algebraically exact in eval mode and removes
    every BatchNorm2d (serving profile showed BN at ~29% of ResNet50
    kernel time)."""
    import copy

    import torch

    from client_amd.models.resnet import ResNet50, fold_batchnorm

    torch.manual_seed(0)
    m = ResNet50().eval()
    for mod in m.modules():
        if isinstance(mod, torch.nn.BatchNorm2d):
            mod.running_mean.uniform_(-0.5, 0.5)
            mod.running_var.uniform_(0.5, 2.0)
            mod.weight.data.uniform_(0.5, 1.5)
            mod.bias.data.uniform_(-0.3, 0.3)
    x = torch.randn(2, 3, 64, 64)
    with torch.inference_mode():
        ref = m(x)
        folded = fold_batchnorm(copy.deepcopy(m))
        out = folded(x)
    assert torch.allclose(ref, out, atol=1e-3, rtol=1e-4)
    assert not any(isinstance(mm, torch.nn.BatchNorm2d)
                   for mm in folded.modules())
