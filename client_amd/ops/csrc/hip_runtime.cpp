// client_amd._hip_c — MI355X-native HIP runtime + CDNA4 kernels.
//
// The GPU surface of the client stack: HIP-IPC shared
// memory (hipMalloc + hipIpcGetMemHandle / hipIpcOpenMemHandle), async
// copies on cached per-device streams, and hand-written gfx950 kernels
// for the work the reference does on the CPU (per-element BF16 loops,
// OpenCV preprocess):
//   - cast_fp32_bf16 / cast_bf16_fp32   (wire-exact truncate / zero-pad)
//   - cast_fp32_fp8e4m3 / cast_fp8e4m3_fp32 (OCP e4m3fn, CDNA4 native)
//   - gather_pack (strided -> contiguous, lifts the reference's DLPack
//     contiguity restriction, cuda_shared_memory/__init__.py:328-388)
//   - image_preprocess (u8 HWC -> resize bilinear -> normalize -> CHW,
//     replaces image_client.cc:86-190's OpenCV path)
//
// All kernels are memory-bound streaming kernels: 16 B/lane vectorized
// access, 256-thread blocks (4 waves of 64), grid-stride with the grid
// capped at 2048 workgroups (cdna_hip_programming.md §6 G11/G13).
// Pure HIP — no CUDA headers, no torch dependency; torch interop is via
// DLPack in Python.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string(#expr) + " failed: " +          \
                               hipGetErrorString(_e));                     \
    }                                                                      \
  } while (0)

// ---------------------------------------------------------------------------
// CDNA4 kernels + C-ABI launchers. kernels.hip is the single source of
// truth, shared with the C++ client library; it is INCLUDED here (one
// translation unit, one device image) — a separate-TU link was observed
// to abort at first launch on the GPU box.
// ---------------------------------------------------------------------------
#include "kernels.hip"

// ---------------------------------------------------------------------------
// Launch helpers
// ---------------------------------------------------------------------------

static inline int grid_for(long work_items) {
  // memory-bound streaming grid: cap at 2048 workgroups, grid-stride the
  // rest (G11). 256 CUs x 8 blocks/CU.
  long blocks = (work_items + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// Per-device cached streams: [0] = copy/pack stream, [1] = side stream for
// collective overlap (SURVEY.md §2.9 row 5 mandates the second stream).
static std::mutex g_stream_mu;
static std::unordered_map<int, std::vector<hipStream_t>> g_streams;

static hipStream_t get_stream(int device, int idx = 0) {
  std::lock_guard<std::mutex> lock(g_stream_mu);
  auto& vec = g_streams[device];
  while ((int)vec.size() <= idx) {
    int prev;
    HIP_CHECK(hipGetDevice(&prev));
    HIP_CHECK(hipSetDevice(device));
    hipStream_t s;
    HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    vec.push_back(s);
    HIP_CHECK(hipSetDevice(prev));
  }
  return vec[idx];
}

// ---------------------------------------------------------------------------
// Python bindings
// ---------------------------------------------------------------------------

static std::pair<size_t, size_t> mem_info(int device) {
  int prev;
  HIP_CHECK(hipGetDevice(&prev));
  HIP_CHECK(hipSetDevice(device));
  size_t free_b = 0, total_b = 0;
  HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
  HIP_CHECK(hipSetDevice(prev));
  return {free_b, total_b};
}

static int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

static uintptr_t hip_malloc(int device, size_t byte_size) {
  int prev;
  HIP_CHECK(hipGetDevice(&prev));
  HIP_CHECK(hipSetDevice(device));
  void* ptr = nullptr;
  HIP_CHECK(hipMalloc(&ptr, byte_size));
  HIP_CHECK(hipSetDevice(prev));
  return reinterpret_cast<uintptr_t>(ptr);
}

static void hip_free(uintptr_t ptr) {
  HIP_CHECK(hipFree(reinterpret_cast<void*>(ptr)));
}

static py::bytes ipc_get_mem_handle(uintptr_t ptr) {
  hipIpcMemHandle_t handle;
  HIP_CHECK(hipIpcGetMemHandle(&handle, reinterpret_cast<void*>(ptr)));
  return py::bytes(reinterpret_cast<const char*>(&handle), sizeof(handle));
}

static uintptr_t ipc_open_mem_handle(py::bytes raw) {
  std::string s = raw;
  if (s.size() < sizeof(hipIpcMemHandle_t)) {
    s.resize(sizeof(hipIpcMemHandle_t), '\0');
  }
  hipIpcMemHandle_t handle;
  std::memcpy(&handle, s.data(), sizeof(handle));
  void* ptr = nullptr;
  HIP_CHECK(hipIpcOpenMemHandle(&ptr, handle, hipIpcMemLazyEnablePeerAccess));
  return reinterpret_cast<uintptr_t>(ptr);
}

static void ipc_close_mem_handle(uintptr_t ptr) {
  HIP_CHECK(hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr)));
}

static void memcpy_h2d(uintptr_t dst, py::buffer src, size_t byte_size,
                       int device, bool sync) {
  py::buffer_info info = src.request();
  if ((size_t)(info.size * info.itemsize) < byte_size)
    throw std::runtime_error("source buffer smaller than byte_size");
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst), info.ptr, byte_size,
                             hipMemcpyHostToDevice, s));
    if (sync) HIP_CHECK(hipStreamSynchronize(s));
  }
}

static py::bytes memcpy_d2h(uintptr_t src, size_t byte_size, int device) {
  std::string out(byte_size, '\0');
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(&out[0], reinterpret_cast<void*>(src), byte_size,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
  return py::bytes(out);
}

static void memcpy_d2h_into(uintptr_t src, py::buffer dst, size_t byte_size,
                            int device) {
  py::buffer_info info = dst.request(true);
  if ((size_t)(info.size * info.itemsize) < byte_size)
    throw std::runtime_error("destination buffer smaller than byte_size");
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(info.ptr, reinterpret_cast<void*>(src), byte_size,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void memcpy_d2d(uintptr_t dst, uintptr_t src, size_t byte_size,
                       int device, bool sync) {
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                             reinterpret_cast<void*>(src), byte_size,
                             hipMemcpyDeviceToDevice, s));
    if (sync) HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void stream_sync(int device, int idx) {
  hipStream_t s = get_stream(device, idx);
  py::gil_scoped_release release;
  HIP_CHECK(hipStreamSynchronize(s));
}

static uintptr_t stream_handle(int device, int idx) {
  // raw hipStream_t of a cached stream, for torch.cuda.ExternalStream
  // interop (event-ordered overlap of pack kernels with RCCL
  // collectives — SURVEY.md §2.8 final row)
  return reinterpret_cast<uintptr_t>(get_stream(device, idx));
}

static void memcpy_peer_async(uintptr_t dst, int dst_device, uintptr_t src,
                              int src_device, size_t byte_size,
                              int stream_idx) {
  // direct xGMI point-to-point copy; with one call per destination GPU
  // on its own stream, the 7 copies of an 8-way scatter ride 7 distinct
  // xGMI links concurrently (ring bcast is per-link
  // bound; direct scatter is not)
  hipStream_t s = get_stream(src_device, stream_idx);
  py::gil_scoped_release release;
  HIP_CHECK(hipMemcpyPeerAsync(reinterpret_cast<void*>(dst), dst_device,
                               reinterpret_cast<void*>(src), src_device,
                               byte_size, s));
}

static void stream_fence(int device, int from_idx, std::vector<int> to_idxs) {
  // device-side ordering: record an event on stream from_idx and make
  // every stream in to_idxs wait on it — no host block. Used to order
  // the 7-link peer scatter after the pack kernel.
  hipStream_t from = get_stream(device, from_idx);
  hipEvent_t ev;
  HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  HIP_CHECK(hipEventRecord(ev, from));
  for (int idx : to_idxs) {
    HIP_CHECK(hipStreamWaitEvent(get_stream(device, idx), ev, 0));
  }
  // destruction is deferred by the runtime until the event completes
  HIP_CHECK(hipEventDestroy(ev));
}

static void device_enable_peer_access(int device, int peer) {
  int prev;
  HIP_CHECK(hipGetDevice(&prev));
  HIP_CHECK(hipSetDevice(device));
  hipError_t e = hipDeviceEnablePeerAccess(peer, 0);
  if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) {
    HIP_CHECK(hipSetDevice(prev));
    throw std::runtime_error(std::string("hipDeviceEnablePeerAccess: ") +
                             hipGetErrorString(e));
  }
  HIP_CHECK(hipSetDevice(prev));
}

static void device_sync() {
  py::gil_scoped_release release;
  HIP_CHECK(hipDeviceSynchronize());
}

// ---- kernel wrappers (operate on raw device pointers) ----

static void cast_fp32_bf16(uintptr_t src, uintptr_t dst, long n, int device,
                           bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  HIP_CHECK(ca_cast_fp32_bf16(reinterpret_cast<const void*>(src),
                 reinterpret_cast<void*>(dst), n, s));
  
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_bf16_fp32(uintptr_t src, uintptr_t dst, long n, int device,
                           bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  HIP_CHECK(ca_cast_bf16_fp32(reinterpret_cast<const void*>(src),
                 reinterpret_cast<void*>(dst), n, s));
  
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_fp32_fp8e4m3(uintptr_t src, uintptr_t dst, long n, int device,
                              bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  HIP_CHECK(ca_cast_fp32_fp8e4m3(reinterpret_cast<const void*>(src),
                 reinterpret_cast<void*>(dst), n, s));
  
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_fp8e4m3_fp32(uintptr_t src, uintptr_t dst, long n, int device,
                              bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  HIP_CHECK(ca_cast_fp8e4m3_fp32(reinterpret_cast<const void*>(src),
                 reinterpret_cast<void*>(dst), n, s));
  
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void gather_pack(uintptr_t src, uintptr_t dst, int elem_size,
                        std::vector<long> shape, std::vector<long> strides,
                        int device, bool sync) {
  // normalize to 4-D (pad leading dims with 1 / stride 0)
  while (shape.size() < 4) {
    shape.insert(shape.begin(), 1);
    strides.insert(strides.begin(), 0);
  }
  if (shape.size() > 4) throw std::runtime_error("gather_pack: >4D unsupported");
  hipStream_t s = get_stream(device);
  HIP_CHECK(ca_gather_pack(reinterpret_cast<const void*>(src),
                           reinterpret_cast<void*>(dst), elem_size,
                           shape.data(), strides.data(), s));
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void bias_act_bf16(uintptr_t x, uintptr_t bias, uintptr_t out,
                          long n_planes, long plane, int channels,
                          bool relu, uintptr_t stream_handle) {
  HIP_CHECK(ca_bias_act_bf16(reinterpret_cast<const void*>(x),
                             reinterpret_cast<const void*>(bias),
                             reinterpret_cast<void*>(out), n_planes, plane,
                             channels, relu ? 1 : 0,
                             reinterpret_cast<hipStream_t>(stream_handle)));
}

static void decode_gemm_bf16(uintptr_t x, uintptr_t w, uintptr_t y, int n,
                             int k, uintptr_t stream_handle) {
  HIP_CHECK(ca_decode_gemm_bf16(
      reinterpret_cast<const void*>(x), reinterpret_cast<const void*>(w),
      reinterpret_cast<void*>(y), n, k,
      reinterpret_cast<hipStream_t>(stream_handle)));
}

static void bias_res_act_cl_bf16(uintptr_t x, uintptr_t res, uintptr_t bias,
                                 uintptr_t out, long n, int channels,
                                 bool relu, uintptr_t stream_handle) {
  HIP_CHECK(ca_bias_res_act_cl_bf16(
      reinterpret_cast<const void*>(x), reinterpret_cast<const void*>(res),
      reinterpret_cast<const void*>(bias), reinterpret_cast<void*>(out), n,
      channels, relu ? 1 : 0,
      reinterpret_cast<hipStream_t>(stream_handle)));
}

static void bias_res_act_bf16(uintptr_t x, uintptr_t res, uintptr_t bias,
                              uintptr_t out, long n_planes, long plane,
                              int channels, bool relu,
                              uintptr_t stream_handle) {
  HIP_CHECK(ca_bias_res_act_bf16(
      reinterpret_cast<const void*>(x), reinterpret_cast<const void*>(res),
      reinterpret_cast<const void*>(bias), reinterpret_cast<void*>(out),
      n_planes, plane, channels, relu ? 1 : 0,
      reinterpret_cast<hipStream_t>(stream_handle)));
}

static void rmsnorm_bf16(uintptr_t x, uintptr_t w, uintptr_t out,
                         long rows, int dim, double eps,
                         uintptr_t stream_handle) {
  // launches on the CALLER's stream (torch's current stream) so it
  // composes with torch ops and hipGraph capture
  hipStream_t s = reinterpret_cast<hipStream_t>(stream_handle);
  HIP_CHECK(ca_rmsnorm_bf16(reinterpret_cast<const void*>(x),
                            reinterpret_cast<const void*>(w),
                            reinterpret_cast<void*>(out), rows, dim,
                            (float)eps, s));
}

static void rope_decode_bf16(uintptr_t q, uintptr_t k, uintptr_t cos_tab,
                             uintptr_t sin_tab, uintptr_t pos, int b, int hq,
                             int hk, int d, uintptr_t stream_handle) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream_handle);
  HIP_CHECK(ca_rope_decode_bf16(reinterpret_cast<void*>(q),
                                reinterpret_cast<void*>(k),
                                reinterpret_cast<const void*>(cos_tab),
                                reinterpret_cast<const void*>(sin_tab),
                                reinterpret_cast<const void*>(pos), b, hq, hk,
                                d, s));
}

static void rope_scatter_decode_bf16(uintptr_t q, uintptr_t k, uintptr_t v,
                                     uintptr_t ck, uintptr_t cv,
                                     uintptr_t cos_tab, uintptr_t sin_tab,
                                     uintptr_t pos, int b, int hq, int hk,
                                     int d, long cache_len, float q_scale,
                                     uintptr_t stream_handle) {
  HIP_CHECK(ca_rope_scatter_decode_bf16(
      reinterpret_cast<void*>(q), reinterpret_cast<const void*>(k),
      reinterpret_cast<const void*>(v), reinterpret_cast<void*>(ck),
      reinterpret_cast<void*>(cv), reinterpret_cast<const void*>(cos_tab),
      reinterpret_cast<const void*>(sin_tab),
      reinterpret_cast<const void*>(pos), b, hq, hk, d, cache_len, q_scale,
      reinterpret_cast<hipStream_t>(stream_handle)));
}

static void image_preprocess(uintptr_t src, uintptr_t dst, int ih, int iw,
                             int oh, int ow, int mode, bool out_bf16,
                             std::vector<float> mean, std::vector<float> stdev,
                             int device, bool sync) {
  if (mean.size() != 3 || stdev.size() != 3)
    throw std::runtime_error("mean/std must have 3 channels");
  hipStream_t s = get_stream(device);
  HIP_CHECK(ca_image_preprocess(reinterpret_cast<const void*>(src),
                                reinterpret_cast<void*>(dst), ih, iw, oh, ow,
                                mode, out_bf16 ? 1 : 0, mean.data(),
                                stdev.data(), s));
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void image_preprocess_batched(uintptr_t src, uintptr_t dst,
                                     int n_images, int ih, int iw, int oh,
                                     int ow, int mode, bool out_bf16,
                                     std::vector<float> mean,
                                     std::vector<float> stdev, int device,
                                     bool sync) {
  if (mean.size() != 3 || stdev.size() != 3)
    throw std::runtime_error("mean/std must have 3 channels");
  hipStream_t s = get_stream(device);
  HIP_CHECK(ca_image_preprocess_batched(
      reinterpret_cast<const void*>(src), reinterpret_cast<void*>(dst),
      n_images, ih, iw, oh, ow, mode, out_bf16 ? 1 : 0, mean.data(),
      stdev.data(), s));
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

PYBIND11_MODULE(_hip_c, m) {
  m.doc() = "client_amd MI355X HIP runtime + CDNA4 kernels";
  m.def("device_count", &device_count);
  m.def("mem_info", &mem_info, py::arg("device") = 0);
  m.def("malloc", &hip_malloc, py::arg("device"), py::arg("byte_size"));
  m.def("free", &hip_free, py::arg("ptr"));
  m.def("ipc_get_mem_handle", &ipc_get_mem_handle, py::arg("ptr"));
  m.def("ipc_open_mem_handle", &ipc_open_mem_handle, py::arg("raw_handle"));
  m.def("ipc_close_mem_handle", &ipc_close_mem_handle, py::arg("ptr"));
  m.def("memcpy_h2d", &memcpy_h2d, py::arg("dst"), py::arg("src"),
        py::arg("byte_size"), py::arg("device") = 0, py::arg("sync") = true);
  m.def("memcpy_d2h", &memcpy_d2h, py::arg("src"), py::arg("byte_size"),
        py::arg("device") = 0);
  m.def("memcpy_d2h_into", &memcpy_d2h_into, py::arg("src"), py::arg("dst"),
        py::arg("byte_size"), py::arg("device") = 0);
  m.def("memcpy_d2d", &memcpy_d2d, py::arg("dst"), py::arg("src"),
        py::arg("byte_size"), py::arg("device") = 0, py::arg("sync") = true);
  m.def("stream_sync", &stream_sync, py::arg("device") = 0,
        py::arg("stream_idx") = 0);
  m.def("stream_handle", &stream_handle, py::arg("device") = 0,
        py::arg("stream_idx") = 0);
  m.def("memcpy_peer_async", &memcpy_peer_async, py::arg("dst"),
        py::arg("dst_device"), py::arg("src"), py::arg("src_device"),
        py::arg("byte_size"), py::arg("stream_idx") = 0);
  m.def("device_enable_peer_access", &device_enable_peer_access,
        py::arg("device"), py::arg("peer"));
  m.def("stream_fence", &stream_fence, py::arg("device"),
        py::arg("from_idx"), py::arg("to_idxs"));
  m.def("device_sync", &device_sync);
  m.def("cast_fp32_bf16", &cast_fp32_bf16, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("cast_bf16_fp32", &cast_bf16_fp32, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("cast_fp32_fp8e4m3", &cast_fp32_fp8e4m3, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("cast_fp8e4m3_fp32", &cast_fp8e4m3_fp32, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("bias_act_bf16", &bias_act_bf16, py::arg("x"), py::arg("bias"),
        py::arg("out"), py::arg("n_planes"), py::arg("plane"),
        py::arg("channels"), py::arg("relu"), py::arg("stream_handle"));
  m.def("bias_res_act_bf16", &bias_res_act_bf16, py::arg("x"), py::arg("res"),
        py::arg("bias"), py::arg("out"), py::arg("n_planes"), py::arg("plane"),
        py::arg("channels"), py::arg("relu"), py::arg("stream_handle"));
  m.def("decode_gemm_bf16", &decode_gemm_bf16, py::arg("x"), py::arg("w"),
        py::arg("y"), py::arg("n"), py::arg("k"), py::arg("stream_handle"));
  m.def("bias_res_act_cl_bf16", &bias_res_act_cl_bf16, py::arg("x"),
        py::arg("res"), py::arg("bias"), py::arg("out"), py::arg("n"),
        py::arg("channels"), py::arg("relu"), py::arg("stream_handle"));
  m.def("rmsnorm_bf16", &rmsnorm_bf16, py::arg("x"), py::arg("w"),
        py::arg("out"), py::arg("rows"), py::arg("dim"), py::arg("eps"),
        py::arg("stream_handle"));
  m.def("rope_decode_bf16", &rope_decode_bf16, py::arg("q"), py::arg("k"),
        py::arg("cos_tab"), py::arg("sin_tab"), py::arg("pos"), py::arg("b"),
        py::arg("hq"), py::arg("hk"), py::arg("d"),
        py::arg("stream_handle"));
  m.def("rope_scatter_decode_bf16", &rope_scatter_decode_bf16, py::arg("q"),
        py::arg("k"), py::arg("v"), py::arg("ck"), py::arg("cv"),
        py::arg("cos_tab"), py::arg("sin_tab"), py::arg("pos"), py::arg("b"),
        py::arg("hq"), py::arg("hk"), py::arg("d"), py::arg("cache_len"),
        py::arg("q_scale") = 1.0f, py::arg("stream_handle") = 0);
  m.def("gather_pack", &gather_pack, py::arg("src"), py::arg("dst"),
        py::arg("elem_size"), py::arg("shape"), py::arg("strides"),
        py::arg("device") = 0, py::arg("sync") = true);
  m.def("image_preprocess", &image_preprocess, py::arg("src"), py::arg("dst"),
        py::arg("ih"), py::arg("iw"), py::arg("oh"), py::arg("ow"),
        py::arg("mode") = 0, py::arg("out_bf16") = false,
        py::arg("mean") = std::vector<float>{0.f, 0.f, 0.f},
        py::arg("std") = std::vector<float>{1.f, 1.f, 1.f},
        py::arg("device") = 0, py::arg("sync") = true);
  m.def("image_preprocess_batched", &image_preprocess_batched, py::arg("src"),
        py::arg("dst"), py::arg("n_images"), py::arg("ih"), py::arg("iw"),
        py::arg("oh"), py::arg("ow"), py::arg("mode") = 0,
        py::arg("out_bf16") = false,
        py::arg("mean") = std::vector<float>{0.f, 0.f, 0.f},
        py::arg("std") = std::vector<float>{1.f, 1.f, 1.f},
        py::arg("device") = 0, py::arg("sync") = true);
}
