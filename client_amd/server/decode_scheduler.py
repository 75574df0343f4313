"""Continuous batched decoding for decoupled LLM serving.

Serialized per-stream decode loops waste the GPU: a decode step's cost
is dominated by reading the weights out of HBM3E, which is independent
of batch size until the batch is large. This scheduler keeps ONE decode
loop over a preallocated batched KV cache; concurrent streams claim a
slot, are prefilled individually into their cache rows, and then every
loop iteration advances ALL active rows one token (per-row positions +
masked attention — LlamaModel.forward_decode_batch). Streams join and
leave at token boundaries, so N concurrent streams cost ~1 stream of
wall-clock per token instead of N.

(Measured motivation, r01: serialized streams degrade inter-token
latency 14 ms -> 99 ms from concurrency 1 -> 4; profiles/genai_*.)
"""

import queue
import threading

import torch


class _Slot:
    __slots__ = ("state", "pos", "last_token", "remaining", "out_queue",
                 "prefill_ids", "prefill_pos")
    FREE, PREFILL, ACTIVE = 0, 1, 2

    def __init__(self):
        self.state = _Slot.FREE
        self.pos = 0
        self.last_token = 0
        self.remaining = 0
        self.out_queue = None
        self.prefill_ids = None
        self.prefill_pos = 0

    @property
    def active(self):
        return self.state == _Slot.ACTIVE


class DecodeScheduler:
    END = object()

    def __init__(self, model, max_batch=8, device="cuda:0", dtype=None,
                 use_graph=None, len_bucket=256, prefill_chunk=128,
                 prefill_rows_per_step=2):
        self.model = model
        self.device = device
        self.dtype = dtype if dtype is not None else next(
            model.parameters()
        ).dtype
        self.max_batch = max_batch
        self.kv_cache = model.make_kv_cache(max_batch, device, self.dtype)
        self.slots = [_Slot() for _ in range(max_batch)]
        # hipGraph capture of the decode step: pad max_len to fixed
        # buckets so every shape is static; the mask makes the padding
        # inert. Inputs/outputs are persistent device tensors that
        # replays read/write in place.
        self.use_graph = (device.startswith("cuda")
                          if use_graph is None else use_graph)
        self.len_bucket = len_bucket
        # chunked prefill: a new stream's prompt is consumed in chunks
        # interleaved with decode steps, so admission never stalls
        # in-flight streams for more than one chunk's worth of compute
        self.prefill_chunk = prefill_chunk
        self._tokens_dev = torch.zeros(max_batch, 1, dtype=torch.int64,
                                       device=device)
        # inactive rows scatter their (garbage) K/V at the model's
        # reserved scratch position, which decode never reads — NOT at
        # position 1, which would corrupt rows still in chunked prefill
        self._scratch = model.scratch_pos
        self._pos_dev = torch.full((max_batch,), self._scratch,
                                   dtype=torch.int64, device=device)
        self._graphs = {}  # bucket -> (graph, next_tokens_out)
        # batched prefill: one static-shape forward advances the
        # mid-prefill slots by one chunk (replacing the per-slot eager
        # loop whose 32-layer launch overhead stalled decode ~86 ms per
        # admission). Graphs are captured per (group_size, bucket) with
        # a row_map into the cache, so replay compute is proportional
        # to the rows actually prefilling — a single full-batch static
        # shape measured as a 60+ ms replay (8x512 positions of GEMM)
        # and made the stall WORSE.
        self._pf_bufs = {}    # group_size -> dict of persistent tensors
        self._pf_graphs = {}  # (group_size, bucket) -> (graph, out)
        # cap prefill rows advanced per loop iteration: a start burst
        # admitting max_batch prompts at once would otherwise run one
        # [8, C] replay (~45 ms measured) in a single decode gap;
        # spreading the rows keeps every ITL near one decode step while
        # the burst drains over a few iterations
        self.prefill_rows_per_step = prefill_rows_per_step
        self._pending = queue.Queue()
        self._cv = threading.Condition()
        self._alive = True
        self._worker = threading.Thread(target=self._run, daemon=True)
        self._worker.start()

    @property
    def kv_utilization(self):
        """Fraction of decode slots occupied (ORCA endpoint-load-metrics
        source; reference README.md:352-366)."""
        busy = sum(1 for s in self.slots if s.state != _Slot.FREE)
        return busy / max(1, len(self.slots))

    def prewarm(self, n_buckets=2):
        """Capture the first n decode-graph buckets up front. A cold
        capture costs ~0.5 s and otherwise lands in the first stream's
        TTFT (measured: TTFT p90 590 ms cold vs ~50 ms warm) — the
        MI355X analog of the reference's ModelWarmup."""
        if not self.use_graph:
            return
        for i in range(1, n_buckets + 1):
            bucket = min(i * self.len_bucket, self.model.cfg.max_seq)
            self._get_graph(bucket)
            # capture every group size the scheduler can actually issue
            # (1..prefill_rows_per_step, powers of two) — an uncaptured
            # (group, bucket) pair costs ~0.5 s in some stream's TTFT
            g = 1
            while g <= max(1, self.prefill_rows_per_step):
                self._get_prefill_graph(g, bucket)
                g *= 2
            if i <= 2:
                # idle-burst shape: with nothing decoding the cap lifts
                # to max_batch, so the first bursts replay (max_batch,
                # low-bucket) graphs
                self._get_prefill_graph(self._group_size(self.max_batch),
                                        bucket)
            if bucket >= self.model.cfg.max_seq:
                break

    def shutdown(self):
        with self._cv:
            self._alive = False
            self._cv.notify()

    def submit(self, input_ids, max_new_tokens):
        """Returns a queue yielding token ids (ints), then END."""
        out = queue.Queue()
        with self._cv:
            self._pending.put((input_ids, max_new_tokens, out))
            self._cv.notify()
        return out

    # ---- worker ----

    def _admit(self):
        """Claim free slots for pending requests (prefill happens in
        chunks from _prefill_step)."""
        while True:
            free = [i for i, s in enumerate(self.slots)
                    if s.state == _Slot.FREE]
            if not free:
                return
            try:
                input_ids, max_new, out = self._pending.get_nowait()
            except queue.Empty:
                return
            idx = free[0]
            slot = self.slots[idx]
            ids = torch.as_tensor(input_ids, dtype=torch.int64,
                                  device=self.device)[None]
            if ids.shape[1] >= self.model.cfg.max_seq:
                # prompt longer than the KV cache: reject the stream
                out.put(self.END)
                continue
            slot.state = _Slot.PREFILL
            slot.prefill_ids = ids
            slot.prefill_pos = 0
            slot.remaining = max_new
            slot.out_queue = out

    def _pf_buffers(self, group):
        bufs = self._pf_bufs.get(group)
        if bufs is None:
            dev = self.device
            bufs = {
                "tokens": torch.zeros(group, self.prefill_chunk,
                                      dtype=torch.int64, device=dev),
                "pos": torch.zeros(group, dtype=torch.int64, device=dev),
                "lens": torch.zeros(group, dtype=torch.int64, device=dev),
                "last": torch.zeros(group, dtype=torch.int64, device=dev),
                "rows": torch.zeros(group, dtype=torch.int64, device=dev),
            }
            self._pf_bufs[group] = bufs
        return bufs

    _EAGER = "eager"  # sentinel: capture failed, run this shape eager

    def _capture_graph(self, forward):
        """Capture forward().argmax(-1) into a hipGraph; returns
        (graph, out_tensor) or None if capture cannot complete.

        A capture can be INVALIDATED by concurrent work from other
        serving threads (co-served models replaying/copying while this
        model lazily captures a new shape —
        hipErrorStreamCaptureInvalidated, seen in the r02 mixed soak,
        where the raw exception killed the decode worker). One retry
        after a full sync, then the caller falls back to eager for
        that shape."""
        from .models import GRAPH_CAPTURE_LOCK

        for _ in range(2):
            try:
                # EXCLUSIVE vs every model execution for warmup AND
                # capture (see models._RWLock rationale)
                with GRAPH_CAPTURE_LOCK:
                    side = torch.cuda.Stream()
                    side.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(side):
                        with torch.inference_mode():
                            for _ in range(2):  # warmup (idempotent)
                                warm = forward()
                    torch.cuda.current_stream().wait_stream(side)
                    torch.cuda.synchronize()
                    graph = torch.cuda.CUDAGraph()  # hipGraph on ROCm
                    with torch.inference_mode():
                        with torch.cuda.graph(
                                graph,
                                capture_error_mode="thread_local"):
                            out = forward()
                return (graph, out)
            except Exception:
                try:
                    torch.cuda.synchronize()
                except Exception:
                    pass
        return None

    def _get_prefill_graph(self, group, bucket):
        entry = self._pf_graphs.get((group, bucket))
        if entry is not None:
            return entry
        bufs = self._pf_buffers(group)
        args = (bufs["tokens"], bufs["pos"], bufs["lens"], bufs["last"],
                self.kv_cache, bucket, bufs["rows"])
        entry = self._capture_graph(
            lambda: self.model.forward_prefill_chunk(*args).argmax(-1)
        ) or self._EAGER
        self._pf_graphs[(group, bucket)] = entry
        return entry

    @staticmethod
    def _group_size(n):
        g = 1
        while g < n:
            g *= 2
        return g

    def _prefill_step(self):
        """Advance every mid-prefill slot by one chunk. Slots are
        processed in power-of-two groups whose hipGraphs are captured
        per (group_size, bucket): replay compute stays proportional to
        the rows actually prefilling (usually one), so the decode-loop
        stall per admission is a few ms instead of a full eager
        32-layer forward (~86 ms)."""
        pf = [i for i, s in enumerate(self.slots)
              if s.state == _Slot.PREFILL]
        if not pf:
            return
        c = self.prefill_chunk
        # advance at most prefill_rows_per_step rows THIS iteration so
        # in-flight decodes never stall more than ~one decode step; the
        # rest progress on later iterations. When NOTHING is decoding
        # (e.g. a start burst) there is no ITL to protect — run every
        # prefilling row at once for the best TTFT.
        any_active = any(s.active for s in self.slots)
        cap = (max(1, self.prefill_rows_per_step) if any_active
               else self.max_batch)
        todo = pf[:cap]
        done = 0
        while done < len(todo):
            group_slots = todo[done:done + self.max_batch]
            group = min(self._group_size(len(group_slots)),
                        self.max_batch)
            group_slots = group_slots[:group]
            done += len(group_slots)
            tokens = torch.zeros(group, c, dtype=torch.int64)
            pos = torch.zeros(group, dtype=torch.int64)
            lens = torch.zeros(group, dtype=torch.int64)
            last = torch.zeros(group, dtype=torch.int64)
            rows = torch.zeros(group, dtype=torch.int64)
            ends = {}
            max_end = 1
            for g, i in enumerate(group_slots):
                slot = self.slots[i]
                ids = slot.prefill_ids
                total = ids.shape[1]
                start = slot.prefill_pos
                end = min(start + c, total)
                n_real = end - start
                tokens[g, :n_real] = ids[0, start:end]
                pos[g] = start
                lens[g] = n_real
                last[g] = n_real - 1
                rows[g] = i
                ends[i] = (end, total)
                max_end = max(max_end, end)
            bucket = self._bucket(max_end)
            bufs = self._pf_buffers(group)
            # resolve (or capture) the graph BEFORE taking the shared
            # execution guard — capture takes the exclusive side
            entry = (self._get_prefill_graph(group, bucket)
                     if self.use_graph else self._EAGER)
            from .models import GRAPH_EXEC_SHARED

            with GRAPH_EXEC_SHARED:
                bufs["tokens"].copy_(tokens)
                bufs["pos"].copy_(pos)
                bufs["lens"].copy_(lens)
                bufs["last"].copy_(last)
                bufs["rows"].copy_(rows)
                if entry is not self._EAGER:
                    graph, first_out = entry
                    graph.replay()
                    firsts = first_out.tolist()
                else:
                    with torch.inference_mode():
                        logits = self.model.forward_prefill_chunk(
                            bufs["tokens"], bufs["pos"], bufs["lens"],
                            bufs["last"], self.kv_cache, bucket,
                            bufs["rows"],
                        )
                        firsts = logits.argmax(-1).tolist()
            for g, i in enumerate(group_slots):
                slot = self.slots[i]
                end, total = ends[i]
                slot.prefill_pos = end
                if end < total:
                    continue
                first = int(firsts[g])
                slot.out_queue.put(first)
                slot.prefill_ids = None
                slot.pos = total  # position the NEXT token writes at
                slot.last_token = first
                slot.remaining -= 1
                if slot.remaining <= 0:
                    slot.state = _Slot.FREE
                    slot.out_queue.put(self.END)
                else:
                    slot.state = _Slot.ACTIVE

    def _bucket(self, max_len):
        b = ((max_len + self.len_bucket - 1) // self.len_bucket
             ) * self.len_bucket
        return min(b, self.model.cfg.max_seq)

    @property
    def _wide_graph(self):
        """CLIENT_AMD_WIDE_GRAPH=1: the decode graph also contains the
        H2D staging copies (reading the pinned host tensors, whose
        CONTENTS are re-read at every replay) and the D2H copy of the
        argmax tokens into a pinned output — per step the host only
        fills the pinned staging, replays, stream-syncs and reads the
        pinned result (no per-step copy launches, no tolist
        transfer). Measured +5.3% tok/s and ITL p99 18.9 -> 7.4 ms
        (profiles/genai_r02_c8.json); CLIENT_AMD_WIDE_GRAPH=0
        disables."""
        w = getattr(self, "_wide", None)
        if w is None:
            import os

            w = (os.environ.get("CLIENT_AMD_WIDE_GRAPH", "1") != "0"
                 and self.use_graph)
            self._wide = w
        return w

    def _wide_out(self):
        out = getattr(self, "_out_pinned", None)
        if out is None:
            out = torch.zeros(self.max_batch, dtype=torch.int64,
                              pin_memory=True)
            self._out_pinned = out
            self._out_np = out.numpy()
        return out

    def _get_graph(self, bucket):
        entry = self._graphs.get(bucket)
        if entry is not None:
            return entry
        if self._wide_graph:
            th, ph, _, _ = self._host_staging()
            out_pinned = self._wide_out()

            def fwd():
                self._tokens_dev.copy_(th, non_blocking=True)
                self._pos_dev.copy_(ph, non_blocking=True)
                out = self.model.forward_decode_batch(
                    self._tokens_dev, self._pos_dev, self.kv_cache,
                    max_len=bucket,
                ).argmax(-1)
                out_pinned.copy_(out, non_blocking=True)
                return out
        else:
            def fwd():
                return self.model.forward_decode_batch(
                    self._tokens_dev, self._pos_dev, self.kv_cache,
                    max_len=bucket,
                ).argmax(-1)
        entry = self._capture_graph(fwd) or self._EAGER
        if entry is self._EAGER and self._wide_graph:
            import sys

            print("[decode-scheduler] wide-graph capture failed for "
                  f"bucket {bucket}; running eager", file=sys.stderr,
                  flush=True)
        self._graphs[bucket] = entry
        return entry

    def _host_staging(self):
        st = getattr(self, "_staging", None)
        if st is None:
            pin = str(self.device).startswith("cuda")
            th = torch.zeros(self.max_batch, 1, dtype=torch.int64,
                             pin_memory=pin)
            ph = torch.zeros(self.max_batch, dtype=torch.int64,
                             pin_memory=pin)
            st = (th, ph, th.view(-1).numpy(), ph.numpy())
            self._staging = st
        return st

    def _decode_step(self):
        active = [i for i, s in enumerate(self.slots) if s.active]
        if not active:
            return False
        # run the FULL preallocated batch: decode cost is weight-read
        # bound, so inactive rows are free; masks keep rows independent.
        # Inputs stage through REUSED pinned host tensors + one async
        # H2D each (torch.tensor(..., device=cuda) per step was two
        # blocking H2D round-trips plus allocations).
        th, ph, tnp, pnp = self._host_staging()
        for i, s in enumerate(self.slots):
            tnp[i] = s.last_token
            pnp[i] = max(s.pos, 1) if s.active else self._scratch
        if self.use_graph:
            trace2 = getattr(self, "_trace2", None)
            if trace2 is None:
                import os as _os

                trace2 = _os.environ.get(
                    "CLIENT_AMD_DECODE_TRACE") == "2"
                self._trace2 = trace2
                if trace2:
                    self._t2_acc = [0.0, 0.0, 0.0, 0]
            if trace2:
                import time as _t

                a = _t.monotonic_ns()
            max_pos = max(s.pos for s in self.slots if s.active)
            # graph resolution (possible capture = exclusive) happens
            # BEFORE the shared execution guard
            entry = self._get_graph(self._bucket(max_pos + 1))
            from .models import GRAPH_EXEC_SHARED

            if trace2:
                b = _t.monotonic_ns()
            with GRAPH_EXEC_SHARED:
                wide = self._wide_graph and entry is not self._EAGER
                if not wide:
                    self._tokens_dev.copy_(th, non_blocking=True)
                    self._pos_dev.copy_(ph, non_blocking=True)
                if wide:
                    # everything (H2D, forward, argmax, D2H) is inside
                    # the graph; the pinned staging is already filled
                    graph, _ = entry
                    graph.replay()
                    if trace2:
                        c = _t.monotonic_ns()
                    torch.cuda.current_stream().synchronize()
                    next_tokens = self._out_np.tolist()
                elif entry is not self._EAGER:
                    graph, next_out = entry
                    graph.replay()
                    if trace2:
                        c = _t.monotonic_ns()
                    next_tokens = next_out.tolist()
                else:
                    with torch.inference_mode():
                        logits = self.model.forward_decode_batch(
                            self._tokens_dev, self._pos_dev,
                            self.kv_cache,
                            max_len=self._bucket(max_pos + 1),
                        )
                    if trace2:
                        c = _t.monotonic_ns()
                    next_tokens = logits.argmax(-1).tolist()
            if trace2:
                d = _t.monotonic_ns()
                acc = self._t2_acc
                acc[0] += b - a   # staging copies + graph lookup
                acc[1] += c - b   # replay submit
                acc[2] += d - c   # tolist = GPU wait + D2H
                acc[3] += 1
                if acc[3] % 50 == 0:
                    n = acc[3]
                    print(f"[decode-trace2] stage={acc[0]/n/1e6:.3f}ms "
                          f"submit={acc[1]/n/1e6:.3f}ms "
                          f"wait={acc[2]/n/1e6:.3f}ms", flush=True)
                    acc[0] = acc[1] = acc[2] = 0.0
                    acc[3] = 0
        else:
            self._tokens_dev.copy_(th)
            self._pos_dev.copy_(ph)
            with torch.inference_mode():
                logits = self.model.forward_decode_batch(
                    self._tokens_dev, self._pos_dev, self.kv_cache
                )
                next_tokens = logits.argmax(-1).tolist()
        for i in active:
            slot = self.slots[i]
            tok = int(next_tokens[i])
            slot.out_queue.put(tok)
            slot.last_token = tok
            slot.pos += 1
            slot.remaining -= 1
            if slot.remaining <= 0 or slot.pos >= self.model.cfg.max_seq - 1:
                slot.state = _Slot.FREE
                slot.out_queue.put(self.END)
        return True

    def _fail_all_streams(self, exc):
        """Recovery path for an unexpected error inside a scheduler
        step: terminate every in-flight stream (consumers see END) and
        reset the slots so new requests keep being served."""
        import sys
        import traceback

        print(f"[decode-scheduler] step failed, resetting slots: {exc}",
              file=sys.stderr, flush=True)
        traceback.print_exc()
        for slot in self.slots:
            if slot.state != _Slot.FREE and slot.out_queue is not None:
                slot.out_queue.put(self.END)
            slot.state = _Slot.FREE
            slot.prefill_ids = None
            slot.out_queue = None
        try:
            torch.cuda.synchronize()
        except Exception:
            pass

    def _run(self):
        import os
        import time as _time

        trace = os.environ.get("CLIENT_AMD_DECODE_TRACE") == "1"
        acc = [0.0, 0.0, 0.0, 0]  # admit+wait, prefill, decode, iters
        while True:
            with self._cv:
                while (self._alive and self._pending.empty()
                       and all(s.state == _Slot.FREE for s in self.slots)):
                    self._cv.wait()
                if not self._alive:
                    return
            if not trace:
                try:
                    self._admit()
                    self._prefill_step()
                    self._decode_step()
                except Exception as e:
                    # the worker must OUTLIVE any transient GPU error
                    # (a capture invalidation once killed this thread
                    # and hung every stream — r02 mixed soak): fail the
                    # in-flight streams, reset the slots, keep serving
                    self._fail_all_streams(e)
                continue
            t0 = _time.monotonic_ns()
            self._admit()
            t1 = _time.monotonic_ns()
            self._prefill_step()
            t2 = _time.monotonic_ns()
            self._decode_step()
            t3 = _time.monotonic_ns()
            acc[0] += t1 - t0
            acc[1] += t2 - t1
            acc[2] += t3 - t2
            acc[3] += 1
            if acc[3] % 200 == 0:
                n = acc[3]
                print(f"[decode-trace] iters={n} admit={acc[0]/n/1e6:.3f}ms "
                      f"prefill={acc[1]/n/1e6:.3f}ms "
                      f"decode={acc[2]/n/1e6:.3f}ms", flush=True)
                acc[0] = acc[1] = acc[2] = 0.0
                acc[3] = 0
