"""In-process llama decode micro-run for rocprofv3 --stats: submits N
concurrent generations to the DecodeScheduler and drains them (no
server, so the profiler sees the engine kernels directly)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

import numpy as np
import torch

from client_amd.models.llama import LlamaModel, llama3_8b_config
from client_amd.server.decode_scheduler import DecodeScheduler

cfg = llama3_8b_config()
with torch.device("cuda:0"):
    model = LlamaModel(cfg)
model = model.to(torch.bfloat16).eval()
sched = DecodeScheduler(model, max_batch=8, device="cuda:0")
sched.prewarm()

outs = []
for i in range(8):
    ids = np.random.randint(0, cfg.vocab_size, 128).astype(np.int64)
    outs.append(sched.submit(ids, 64))
total = 0
for q in outs:
    while True:
        tok = q.get(timeout=600)
        if tok is sched.END:
            break
        total += 1
sched.shutdown()
print("decoded tokens:", total)
