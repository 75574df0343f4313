"""perf_analyzer-class load generator.

The reference repo ships only a relocation stub for perf_analyzer
(src/c++/perf_analyzer/README.md:29-30) but the north-star metric is
perf_analyzer-style inferences/sec + p99 latency, so this
is a from-scratch implementation designed from the client-stat substrate
the reference exposes (RequestTimers/InferStat, common.h:93-114,
568-648): closed-loop concurrency driving, warmup, fixed measurement
windows with a stability check, latency percentiles from
REQUEST_START -> REQUEST_END nanosecond timestamps.

Input modes:
  - wire:   synthetic tensors serialized once, bytes reused per request
  - hipshm: per-concurrency-slot HIP-IPC region pairs; the CDNA4 pack
    kernel stages input once (or per request with --repack); requests
    carry only region references — no tensor bytes on the wire
"""

import threading
import time

import numpy as np

from ..utils import triton_to_np_dtype, TRITON_DTYPE_SIZES


class LatencyRecorder:
    def __init__(self):
        self._lock = threading.Lock()
        self.latencies_ns = []
        self.errors = 0

    def record(self, start_ns, end_ns, error=None):
        with self._lock:
            if error is not None:
                self.errors += 1
            else:
                self.latencies_ns.append(end_ns - start_ns)

    def snapshot_and_reset(self):
        with self._lock:
            lat, self.latencies_ns = self.latencies_ns, []
            err, self.errors = self.errors, 0
        return lat, err


def percentile(sorted_ns, q):
    if not sorted_ns:
        return 0.0
    idx = min(len(sorted_ns) - 1, int(round(q / 100.0 * (len(sorted_ns) - 1))))
    return sorted_ns[idx]


class ConcurrencyDriver:
    """Closed-loop load: ``concurrency`` worker threads, each owning one
    slot and re-issuing as soon as its previous request completes
    (perf_analyzer's concurrency model). ``issue_fn(slot)`` BLOCKS until
    the request completes — never issue new RPCs from completion
    callbacks (grpcio completion threads must stay free)."""

    def __init__(self, issue_fn, concurrency):
        self._issue_fn = issue_fn
        self._concurrency = concurrency
        self._recorder = LatencyRecorder()
        self._stop = threading.Event()
        self._done = threading.Event()
        self._active = 0
        self._lock = threading.Lock()

    def _worker(self, slot):
        while not self._stop.is_set():
            start = time.monotonic_ns()
            try:
                self._issue_fn(slot)
                self._recorder.record(start, time.monotonic_ns())
            except Exception as e:
                self._recorder.record(start, time.monotonic_ns(), e)
                time.sleep(0.01)
        with self._lock:
            self._active -= 1
            if self._active == 0:
                self._done.set()

    def run(self, warmup_s, window_s, max_windows, stability_pct=10.0,
            min_stable=3):
        """Returns (throughput_req_s, latencies_ns_sorted, errors, windows)."""
        self._active = self._concurrency
        for slot in range(self._concurrency):
            threading.Thread(target=self._worker, args=(slot,),
                             daemon=True).start()
        time.sleep(warmup_s)
        self._recorder.snapshot_and_reset()
        window_results = []
        all_lat = []
        stable = 0
        for _ in range(max_windows):
            time.sleep(window_s)
            lat, err = self._recorder.snapshot_and_reset()
            thr = len(lat) / window_s
            window_results.append((thr, lat, err))
            all_lat.extend(lat)
            if len(window_results) >= min_stable:
                recent = [w[0] for w in window_results[-min_stable:]]
                mean = sum(recent) / len(recent)
                if mean > 0 and all(
                    abs(r - mean) / mean * 100 <= stability_pct for r in recent
                ):
                    stable += 1
                    break
        self._stop.set()
        self._done.wait(timeout=60)
        total_s = window_s * len(window_results)
        total_req = sum(len(w[1]) for w in window_results)
        total_err = sum(w[2] for w in window_results)
        throughput = total_req / total_s if total_s else 0.0
        all_lat.sort()
        return throughput, all_lat, total_err, len(window_results)


class PeriodicConcurrencyDriver:
    """Ramped closed-loop load (perf_analyzer
    --periodic-concurrency-range): ONE run during which the number of
    concurrent workers grows from ``start`` to ``end`` by ``step``
    every ``period_s`` seconds, recording a separate measurement stage
    per concurrency level. Shows how the server's latency/throughput
    curve bends as load rises without tearing the connections down
    between levels (the ordinary sweep reconnects per level)."""

    def __init__(self, issue_fn, start, end, step, period_s):
        self._issue_fn = issue_fn
        self._start = start
        self._end = end
        self._step = max(1, step)
        self._period_s = period_s
        self._recorder = LatencyRecorder()
        self._stop = threading.Event()
        self._lock = threading.Lock()
        self._active = 0
        self._done = threading.Event()

    def _worker(self, slot):
        while not self._stop.is_set():
            start = time.monotonic_ns()
            try:
                self._issue_fn(slot)
                self._recorder.record(start, time.monotonic_ns())
            except Exception as e:
                self._recorder.record(start, time.monotonic_ns(), e)
                time.sleep(0.01)
        with self._lock:
            self._active -= 1
            if self._active == 0:
                self._done.set()

    def run(self, warmup_s=1.0):
        """Returns a list of per-stage dicts:
        (concurrency, throughput_req_s, latencies_ns_sorted, errors)."""
        stages = []
        level = self._start
        slot = 0
        with self._lock:
            self._active = 0
        while True:
            # bring up the delta workers for this level
            while slot < level:
                with self._lock:
                    self._active += 1
                threading.Thread(target=self._worker, args=(slot,),
                                 daemon=True).start()
                slot += 1
            if level == self._start and warmup_s > 0:
                time.sleep(warmup_s)
            self._recorder.snapshot_and_reset()
            time.sleep(self._period_s)
            lat, err = self._recorder.snapshot_and_reset()
            lat.sort()
            stages.append({
                "concurrency": level,
                "throughput": len(lat) / self._period_s,
                "latencies_ns": lat,
                "errors": err,
            })
            if level >= self._end:
                break
            level = min(level + self._step, self._end)
        self._stop.set()
        self._done.wait(timeout=60)
        return stages


class RequestRateDriver:
    """Open-loop load (perf_analyzer --request-rate-range): a request
    schedule at ``rate`` req/s with constant or Poisson gaps, executed
    by a pool of worker threads that each claim the next schedule slot,
    sleep until its deadline, then issue. Unlike the closed loop,
    server slowdown does not throttle the offered load — late issues
    are counted in ``delayed``."""

    def __init__(self, issue_fn, rate, num_threads, distribution="constant",
                 seed=0):
        self._issue_fn = issue_fn
        self._rate = rate
        self._num_threads = num_threads
        self._distribution = distribution
        self._seed = seed
        self._recorder = LatencyRecorder()
        self._stop = threading.Event()
        self._done = threading.Event()
        self._lock = threading.Lock()
        self._active = 0
        self._next_idx = 0
        self._start_ns = 0
        self.delayed = 0
        # Pre-generated gap schedule (regenerated lazily as needed).
        self._rng = np.random.default_rng(seed)
        self._deadlines = []

    def _deadline(self, idx):
        with self._lock:
            while len(self._deadlines) <= idx:
                if self._distribution == "poisson":
                    gap = float(self._rng.exponential(1.0 / self._rate))
                else:
                    gap = 1.0 / self._rate
                prev = self._deadlines[-1] if self._deadlines else 0.0
                self._deadlines.append(prev + gap)
            return self._deadlines[idx]

    def _worker(self, slot):
        while not self._stop.is_set():
            with self._lock:
                idx = self._next_idx
                self._next_idx += 1
            deadline_ns = self._start_ns + int(self._deadline(idx) * 1e9)
            now = time.monotonic_ns()
            if deadline_ns > now:
                if self._stop.wait((deadline_ns - now) / 1e9):
                    break
            else:
                with self._lock:
                    self.delayed += 1
            start = time.monotonic_ns()
            try:
                self._issue_fn(slot)
                self._recorder.record(start, time.monotonic_ns())
            except Exception as e:
                self._recorder.record(start, time.monotonic_ns(), e)
                time.sleep(0.01)
        with self._lock:
            self._active -= 1
            if self._active == 0:
                self._done.set()

    def run(self, warmup_s, window_s, max_windows, stability_pct=10.0,
            min_stable=3):
        self._start_ns = time.monotonic_ns()
        self._active = self._num_threads
        for slot in range(self._num_threads):
            threading.Thread(target=self._worker, args=(slot,),
                             daemon=True).start()
        time.sleep(warmup_s)
        self._recorder.snapshot_and_reset()
        window_results = []
        all_lat = []
        for _ in range(max_windows):
            time.sleep(window_s)
            lat, err = self._recorder.snapshot_and_reset()
            window_results.append((len(lat) / window_s, lat, err))
            all_lat.extend(lat)
            if len(window_results) >= min_stable:
                recent = [w[0] for w in window_results[-min_stable:]]
                mean = sum(recent) / len(recent)
                if mean > 0 and all(
                    abs(r - mean) / mean * 100 <= stability_pct
                    for r in recent
                ):
                    break
        self._stop.set()
        self._done.wait(timeout=60)
        total_s = window_s * len(window_results)
        total_req = sum(len(w[1]) for w in window_results)
        total_err = sum(w[2] for w in window_results)
        throughput = total_req / total_s if total_s else 0.0
        all_lat.sort()
        return throughput, all_lat, total_err, len(window_results)


def load_input_data(path):
    """Parse a perf_analyzer --input-data JSON file: {"data": [ {name:
    flat_list | {"content": [...], "shape": [...]} , ...}, ... ]}.
    Returns a list of per-request dicts name -> (array_values, shape|None)."""
    import json as _json

    with open(path) as f:
        doc = _json.load(f)
    entries = []
    for item in doc.get("data", []):
        entry = {}
        for name, val in item.items():
            if isinstance(val, dict):
                entry[name] = (val.get("content", []), val.get("shape"))
            else:
                entry[name] = (val, None)
        entries.append(entry)
    if not entries:
        raise ValueError(f"no 'data' entries in {path}")
    return entries


class PerfAnalyzer:
    def __init__(self, url, protocol="grpc", model_name="identity_fp32",
                 batch_size=1, shared_memory="none", input_dtype=None,
                 repack=False, verbose=False, device_id=0, shapes=None,
                 int_range=(0, 127), input_data=None, percentile_q=None):
        self.url = url
        self.protocol = protocol
        self.model_name = model_name
        self.batch_size = batch_size
        self.shared_memory = shared_memory
        self.repack = repack
        self.verbose = verbose
        self.device_id = device_id
        self.input_dtype = input_dtype
        # perf_analyzer-style --shape NAME:d1,d2 overrides for dynamic dims
        self.shapes = shapes or {}
        self.int_range = int_range
        # --input-data file entries (round-robined across slots);
        # None = synthetic random tensors
        self.input_data = (load_input_data(input_data)
                           if isinstance(input_data, str) else input_data)
        self._input_data_idx = 0
        self.percentile_q = percentile_q  # extra reported percentile
        self._client = None
        self._slots = []

    # ---- setup ----

    def _make_client(self, concurrency):
        if self.protocol == "grpc":
            import client_amd.grpc as grpcclient

            return grpcclient.InferenceServerClient(self.url), grpcclient
        else:
            import client_amd.http as httpclient

            return (
                httpclient.InferenceServerClient(
                    self.url, concurrency=concurrency
                ),
                httpclient,
            )

    def _model_io(self, client):
        meta = client.get_model_metadata(self.model_name)
        if self.protocol == "grpc":
            inputs = [(i.name, i.datatype, list(i.shape)) for i in meta.inputs]
            outputs = [(o.name, o.datatype, list(o.shape)) for o in meta.outputs]
        else:
            inputs = [(i["name"], i["datatype"], list(i["shape"]))
                      for i in meta["inputs"]]
            outputs = [(o["name"], o["datatype"], list(o["shape"]))
                       for o in meta["outputs"]]

        def concrete(name, shape):
            if name in self.shapes:
                return [self.batch_size] + list(self.shapes[name]) \
                    if len(self.shapes[name]) == len(shape) - 1 \
                    else list(self.shapes[name])
            out = [d if d > 0 else (self.batch_size if i == 0 else 16)
                   for i, d in enumerate(shape)]
            if any(d <= 0 for d in shape[1:]):
                # a silently-defaulted dim makes results incomparable
                # across runs (a BERT "seq128" sweep without
                # --shape input_ids:128 actually measured seq16 once) —
                # ALWAYS say what was measured
                import sys

                print(
                    f"perf: input '{name}' has wildcard dims {list(shape)}"
                    f" -> measuring {out}; pass --shape {name}:... to"
                    " control it",
                    file=sys.stderr,
                )
            return out

        inputs = [(n, d, concrete(n, s)) for n, d, s in inputs]
        outputs = [(n, d, concrete(n, s)) for n, d, s in outputs]
        return inputs, outputs

    def _synth_array(self, datatype, shape, name=None):
        np_dt = triton_to_np_dtype(datatype)
        if self.input_data is not None and name is not None:
            entry = self.input_data[self._input_data_idx % len(self.input_data)]
            if name in entry:
                content, dshape = entry[name]
                if np_dt == np.object_:
                    arr = np.array(
                        [c.encode() if isinstance(c, str) else c
                         for c in content], dtype=np.object_)
                else:
                    arr = np.array(content, dtype=np_dt)
                return arr.reshape(dshape if dshape else shape)
        if np_dt == np.object_:
            return np.array(
                [b"x" * 8] * int(np.prod(shape)), dtype=np.object_
            ).reshape(shape)
        if np_dt in (np.float16, np.float32, np.float64):
            return np.random.rand(*shape).astype(np_dt)
        lo, hi = self.int_range
        return np.random.randint(lo, hi, size=shape).astype(np_dt)

    def _setup_wire_slots(self, mod, inputs, outputs, concurrency):
        """Pre-serialize request objects, one set per slot (reused —
        mirrors the C++ client's protobuf recycling)."""
        slots = []
        for _ in range(concurrency):
            infer_inputs = []
            for name, datatype, shape in inputs:
                arr = self._synth_array(datatype, shape, name)
                ii = mod.InferInput(name, list(arr.shape), datatype)
                ii.set_data_from_numpy(arr)
                infer_inputs.append(ii)
            self._input_data_idx += 1
            infer_outputs = [mod.InferRequestedOutput(n) for n, _, _ in outputs]
            slots.append((infer_inputs, infer_outputs, []))
        return slots

    def _setup_hipshm_slots(self, client, mod, inputs, outputs, concurrency):
        import client_amd.utils.hip_shared_memory as hipshm

        slots = []
        self._regions = []
        for slot in range(concurrency):
            infer_inputs = []
            regions = []
            staged = []
            for name, datatype, shape in inputs:
                elem = TRITON_DTYPE_SIZES.get(datatype, 4)
                nbytes = int(np.prod(shape)) * elem
                rname = f"pa_{self.model_name}_in_{slot}_{name}"
                region = hipshm.create_shared_memory_region(
                    rname, nbytes, self.device_id
                )
                client.register_cuda_shared_memory(
                    rname,
                    hipshm.get_raw_handle_bytes(region)
                    if self.protocol == "grpc"
                    else hipshm.get_raw_handle(region),
                    self.device_id,
                    nbytes,
                )
                data = self._synth_array(
                    "FP32" if datatype in ("BF16", "FP16") else datatype, shape
                )
                if datatype in ("BF16",):
                    hipshm.set_shared_memory_region_cast(region, data, "BF16")
                else:
                    hipshm.set_shared_memory_region(region, [data])
                ii = mod.InferInput(name, shape, datatype)
                ii.set_shared_memory(rname, nbytes)
                infer_inputs.append(ii)
                regions.append(region)
                staged.append((region, data, datatype))
            infer_outputs = []
            for name, datatype, shape in outputs:
                elem = TRITON_DTYPE_SIZES.get(datatype, 4)
                nbytes = int(np.prod(shape)) * elem
                rname = f"pa_{self.model_name}_out_{slot}_{name}"
                region = hipshm.create_shared_memory_region(
                    rname, nbytes, self.device_id
                )
                client.register_cuda_shared_memory(
                    rname,
                    hipshm.get_raw_handle_bytes(region)
                    if self.protocol == "grpc"
                    else hipshm.get_raw_handle(region),
                    self.device_id,
                    nbytes,
                )
                io = mod.InferRequestedOutput(name)
                io.set_shared_memory(rname, nbytes)
                infer_outputs.append(io)
                regions.append(region)
            self._regions.extend(regions)
            slots.append((infer_inputs, infer_outputs, staged))
        return slots

    def _teardown_hipshm(self, client):
        import client_amd.utils.hip_shared_memory as hipshm

        for region in getattr(self, "_regions", []):
            try:
                # unregister by name: co-serving load generators must not
                # blow away each other's regions (model-scoped names)
                client.unregister_cuda_shared_memory(region._triton_shm_name)
            except Exception:
                pass
            try:
                hipshm.destroy_shared_memory_region(region)
            except Exception:
                pass
        self._regions = []

    # ---- measurement ----

    def _build_issue(self, client, mod, n_slots):
        """Build per-slot request objects + the blocking issue(slot) fn."""
        inputs, outputs = self._model_io(client)
        if self.shared_memory in ("cuda", "hip"):
            slots = self._setup_hipshm_slots(
                client, mod, inputs, outputs, n_slots
            )
        else:
            slots = self._setup_wire_slots(mod, inputs, outputs, n_slots)

        def repack_slot(staged):
            if self.repack and staged:
                import client_amd.utils.hip_shared_memory as hs

                for region, data, datatype in staged:
                    if datatype == "BF16":
                        hs.set_shared_memory_region_cast(region, data, "BF16")
                    else:
                        hs.set_shared_memory_region(region, [data])

        if self.protocol == "grpc":
            def issue(slot_idx):
                infer_inputs, infer_outputs, staged = slots[slot_idx]
                repack_slot(staged)
                done = threading.Event()
                box = {}

                def cb(result, error):
                    box["error"] = error
                    done.set()

                client.async_infer(
                    self.model_name, infer_inputs, callback=cb,
                    outputs=infer_outputs,
                )
                if not done.wait(timeout=120):
                    raise TimeoutError("request timed out")
                if box["error"] is not None:
                    raise box["error"]
        else:
            def issue(slot_idx):
                infer_inputs, infer_outputs, staged = slots[slot_idx]
                repack_slot(staged)
                client.infer(
                    self.model_name, infer_inputs, outputs=infer_outputs,
                )
        return issue

    def _server_stats_snapshot(self, client):
        """Cumulative server-side durations for the model (KServe-v2
        statistics; perf_analyzer's Server Queue / Compute breakdown)."""
        try:
            stats = client.get_inference_statistics(self.model_name)
            if self.protocol == "grpc":
                m = stats.model_stats[0]
                inf = m.inference_stats
                return {
                    "count": inf.success.count,
                    "queue_ns": inf.queue.ns,
                    "compute_input_ns": inf.compute_input.ns,
                    "compute_infer_ns": inf.compute_infer.ns,
                    "compute_output_ns": inf.compute_output.ns,
                }
            m = stats["model_stats"][0]
            inf = m["inference_stats"]
            return {
                "count": inf["success"]["count"],
                "queue_ns": inf["queue"]["ns"],
                "compute_input_ns": inf["compute_input"]["ns"],
                "compute_infer_ns": inf["compute_infer"]["ns"],
                "compute_output_ns": inf["compute_output"]["ns"],
            }
        except Exception:
            return None

    @staticmethod
    def _server_breakdown(before, after):
        """Delta two snapshots -> avg per-request server-side times."""
        if not before or not after:
            return None
        n = after["count"] - before["count"]
        if n <= 0:
            return None
        return {
            "requests": n,
            "avg_queue_us": int((after["queue_ns"] - before["queue_ns"])
                                / n / 1000),
            "avg_compute_input_us": int(
                (after["compute_input_ns"] - before["compute_input_ns"])
                / n / 1000),
            "avg_compute_infer_us": int(
                (after["compute_infer_ns"] - before["compute_infer_ns"])
                / n / 1000),
            "avg_compute_output_us": int(
                (after["compute_output_ns"] - before["compute_output_ns"])
                / n / 1000),
        }

    def _result_dict(self, throughput, lat, errors, windows, **extra):
        latency_us = {
            "avg": int(np.mean(lat) / 1000) if lat else 0,
            "p50": int(percentile(lat, 50) / 1000),
            "p90": int(percentile(lat, 90) / 1000),
            "p95": int(percentile(lat, 95) / 1000),
            "p99": int(percentile(lat, 99) / 1000),
        }
        if self.percentile_q is not None:
            latency_us[f"p{self.percentile_q:g}"] = int(
                percentile(lat, self.percentile_q) / 1000
            )
        result = {
            "batch_size": self.batch_size,
            "request_rate_per_sec": round(throughput, 2),
            "inferences_per_sec": round(throughput * self.batch_size, 2),
            "latency_us": latency_us,
            "errors": errors,
            "measurement_windows": windows,
            "shared_memory": self.shared_memory,
            "protocol": self.protocol,
        }
        result.update(extra)
        return result

    def run(self, concurrency_list, warmup_s=1.0, window_s=2.0, max_windows=6,
            stability_pct=10.0):
        """Closed-loop concurrency sweep; returns list of result dicts."""
        results = []
        for concurrency in concurrency_list:
            client, mod = self._make_client(concurrency)
            try:
                issue = self._build_issue(client, mod, concurrency)
                driver = ConcurrencyDriver(issue, concurrency)
                stats_before = self._server_stats_snapshot(client)
                throughput, lat, errors, windows = driver.run(
                    warmup_s, window_s, max_windows, stability_pct
                )
                server = self._server_breakdown(
                    stats_before, self._server_stats_snapshot(client))
                result = self._result_dict(
                    throughput, lat, errors, windows, concurrency=concurrency,
                    **({"server": server} if server else {}),
                )
                results.append(result)
                if self.verbose:
                    print(result)
                if self.shared_memory in ("cuda", "hip"):
                    self._teardown_hipshm(client)
            finally:
                client.close()
        return results

    def run_periodic(self, start, end, step, period_s=2.0, warmup_s=1.0):
        """Ramped single-run concurrency profile (perf_analyzer
        --periodic-concurrency-range start:end:step with a time-based
        request period); returns one result dict per stage."""
        client, mod = self._make_client(end)
        results = []
        try:
            issue = self._build_issue(client, mod, end)
            driver = PeriodicConcurrencyDriver(issue, start, end, step,
                                               period_s)
            stats_before = self._server_stats_snapshot(client)
            stages = driver.run(warmup_s=warmup_s)
            server = self._server_breakdown(
                stats_before, self._server_stats_snapshot(client))
            for st in stages:
                result = self._result_dict(
                    st["throughput"], st["latencies_ns"], st["errors"], 1,
                    concurrency=st["concurrency"], ramped=True,
                    **({"server": server} if server else {}),
                )
                results.append(result)
                if self.verbose:
                    print(result)
            if self.shared_memory in ("cuda", "hip"):
                self._teardown_hipshm(client)
        finally:
            client.close()
        return results

    def run_request_rate(self, rate_list, warmup_s=1.0, window_s=2.0,
                         max_windows=6, stability_pct=10.0,
                         distribution="constant", max_threads=16):
        """Open-loop request-rate sweep (perf_analyzer
        --request-rate-range / --request-distribution)."""
        results = []
        for rate in rate_list:
            client, mod = self._make_client(max_threads)
            try:
                issue = self._build_issue(client, mod, max_threads)
                driver = RequestRateDriver(
                    issue, rate, max_threads, distribution
                )
                stats_before = self._server_stats_snapshot(client)
                throughput, lat, errors, windows = driver.run(
                    warmup_s, window_s, max_windows, stability_pct
                )
                server = self._server_breakdown(
                    stats_before, self._server_stats_snapshot(client))
                result = self._result_dict(
                    throughput, lat, errors, windows,
                    target_request_rate=rate,
                    request_distribution=distribution,
                    delayed_requests=driver.delayed,
                    **({"server": server} if server else {}),
                )
                results.append(result)
                if self.verbose:
                    print(result)
                if self.shared_memory in ("cuda", "hip"):
                    self._teardown_hipshm(client)
            finally:
                client.close()
        return results
