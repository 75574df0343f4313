// C++ perf_analyzer-class load generator over the native HTTP client.
// Closed-loop concurrency sweep with p50/p90/p99 from RequestTimers-
// style client timestamps (the reference relocated perf_analyzer out of
// its repo; this is the native-core equivalent of client_amd.perf).
//
// Usage: perf_client -u host:port -m model [-b batch]
//        [--concurrency-range start:end:step] [--measurement-interval s]
//        [--shm] (system shared-memory input/output mode)
#include <algorithm>
#include <atomic>
#include <chrono>
#include <cstring>
#include <cmath>
#include <numeric>
#include <iostream>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "client_amd/common.h"
#include "client_amd/http_client.h"
#include "client_amd/json.h"
#include "client_amd/shm_utils.h"

namespace ca = client_amd;
using Clock = std::chrono::steady_clock;

struct Args {
  std::string url = "127.0.0.1:8000";
  std::string model = "simple";
  int batch = 1;
  int c_start = 1, c_end = 4, c_step = 1;
  double window_s = 2.0;
  int max_windows = 4;
  bool shm = false;
};

int main(int argc, char** argv) {
  Args args;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() { return std::string(argv[++i]); };
    if (a == "-u") args.url = next();
    else if (a == "-m") args.model = next();
    else if (a == "-b") args.batch = atoi(next().c_str());
    else if (a == "--concurrency-range") {
      sscanf(next().c_str(), "%d:%d:%d", &args.c_start, &args.c_end,
             &args.c_step);
    } else if (a == "--measurement-interval")
      args.window_s = atof(next().c_str());
    else if (a == "--max-windows") args.max_windows = atoi(next().c_str());
    else if (a == "--shm") args.shm = true;
  }

  for (int concurrency = args.c_start; concurrency <= args.c_end;
       concurrency += args.c_step) {
    std::atomic<bool> stop{false};
    std::mutex lat_mu;
    std::vector<uint64_t> latencies_ns;
    std::atomic<uint64_t> errors{0};

    auto worker = [&](int slot) {
      std::unique_ptr<ca::InferenceServerHttpClient> client;
      if (!ca::InferenceServerHttpClient::Create(&client, args.url).IsOk()) {
        errors++;
        return;
      }
      // fetch metadata to size inputs
      std::string meta_str;
      if (!client->ModelMetadata(&meta_str, args.model).IsOk()) {
        errors++;
        return;
      }
      ca::Json meta = ca::Json::Parse(meta_str);
      std::vector<std::unique_ptr<ca::InferInput>> inputs;
      std::vector<ca::InferInput*> input_ptrs;
      std::vector<std::unique_ptr<ca::InferRequestedOutput>> outputs;
      std::vector<const ca::InferRequestedOutput*> output_ptrs;
      std::vector<std::vector<uint8_t>> buffers;
      // optional system-shm input/output mode: one region per worker,
      // inputs packed once, requests carry only region references
      int shm_fd = -1;
      void* shm_base = nullptr;
      std::string shm_key = "/perf_client_" + std::to_string(slot);
      std::string shm_name = "pc_slot_" + std::to_string(slot);
      size_t shm_cursor = 0;
      const size_t kShmBytes = 1 << 22;
      if (args.shm) {
        if (!ca::CreateSharedMemoryRegion(shm_key, kShmBytes, &shm_fd)
                 .IsOk() ||
            !ca::MapSharedMemory(shm_fd, 0, kShmBytes, &shm_base).IsOk() ||
            !client->RegisterSystemSharedMemory(shm_name, shm_key, kShmBytes)
                 .IsOk()) {
          errors++;
          return;
        }
      }
      for (const auto& in : meta["inputs"].AsArray()) {
        std::vector<int64_t> shape;
        int64_t elems = 1;
        for (const auto& d : in["shape"].AsArray()) {
          int64_t v = d.AsInt() > 0 ? d.AsInt() : (shape.empty() ? args.batch
                                                                 : 16);
          shape.push_back(v);
          elems *= v;
        }
        ca::InferInput* input;
        ca::InferInput::Create(&input, in["name"].AsString(), shape,
                               in["datatype"].AsString());
        size_t elem_size = 4;
        size_t nbytes = elems * elem_size;
        if (args.shm) {
          memset((char*)shm_base + shm_cursor, 1, nbytes);
          input->SetSharedMemory(shm_name, nbytes, shm_cursor);
          shm_cursor += nbytes;
        } else {
          buffers.emplace_back(nbytes, 1);
          input->AppendRaw(buffers.back().data(), buffers.back().size());
        }
        inputs.emplace_back(input);
        input_ptrs.push_back(input);
      }
      if (args.shm) {
        for (const auto& out : meta["outputs"].AsArray()) {
          std::vector<int64_t> shape;
          int64_t elems = 1;
          for (const auto& d : out["shape"].AsArray()) {
            int64_t v = d.AsInt() > 0 ? d.AsInt()
                                      : (shape.empty() ? args.batch : 16);
            elems *= v;
            shape.push_back(v);
          }
          size_t nbytes = elems * 4;
          if (shm_cursor + nbytes > kShmBytes) break;
          ca::InferRequestedOutput* o;
          ca::InferRequestedOutput::Create(&o, out["name"].AsString());
          o->SetSharedMemory(shm_name, nbytes, shm_cursor);
          shm_cursor += nbytes;
          outputs.emplace_back(o);
          output_ptrs.push_back(o);
        }
      }
      ca::InferOptions options(args.model);
      while (!stop.load(std::memory_order_relaxed)) {
        auto t0 = Clock::now();
        ca::InferResult* result = nullptr;
        ca::Error err = client->Infer(&result, options, input_ptrs,
                                      output_ptrs);
        auto t1 = Clock::now();
        if (err.IsOk() && result != nullptr &&
            result->RequestStatus().IsOk()) {
          std::lock_guard<std::mutex> lock(lat_mu);
          latencies_ns.push_back(
              std::chrono::duration_cast<std::chrono::nanoseconds>(t1 - t0)
                  .count());
        } else {
          errors++;
        }
        delete result;
      }
      if (args.shm) {
        client->UnregisterSystemSharedMemory(shm_name);
        ca::UnmapSharedMemory(shm_base, kShmBytes);
        ca::CloseSharedMemory(shm_fd);
        ca::UnlinkSharedMemoryRegion(shm_key);
      }
    };

    std::vector<std::thread> threads;
    for (int slot = 0; slot < concurrency; ++slot)
      threads.emplace_back(worker, slot);

    // warmup then measure
    std::this_thread::sleep_for(std::chrono::milliseconds(500));
    {
      std::lock_guard<std::mutex> lock(lat_mu);
      latencies_ns.clear();
    }
    auto m0 = Clock::now();
    std::this_thread::sleep_for(
        std::chrono::duration<double>(args.window_s * args.max_windows));
    stop = true;
    auto m1 = Clock::now();
    for (auto& t : threads) t.join();

    double elapsed =
        std::chrono::duration_cast<std::chrono::duration<double>>(m1 - m0)
            .count();
    std::vector<uint64_t> lat;
    {
      std::lock_guard<std::mutex> lock(lat_mu);
      lat = latencies_ns;
    }
    std::sort(lat.begin(), lat.end());
    auto pct = [&](double q) -> double {
      if (lat.empty()) return 0;
      size_t idx = std::min(
          lat.size() - 1, (size_t)llround(q / 100.0 * (lat.size() - 1)));
      return lat[idx] / 1000.0;  // usec
    };
    double thr = lat.size() / elapsed;
    std::cout << "Concurrency: " << concurrency
              << ", throughput: " << thr * args.batch
              << " infer/sec, latency avg: "
              << (lat.empty() ? 0
                              : std::accumulate(lat.begin(), lat.end(), 0.0) /
                                    lat.size() / 1000.0)
              << " usec, p50: " << pct(50) << " usec, p90: " << pct(90)
              << " usec, p99: " << pct(99) << " usec, errors: " << errors
              << std::endl;
  }
  return 0;
}
