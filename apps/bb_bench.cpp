// bb_bench — native benchmark client: iterated put/get with latency
// percentiles and throughput summary.
// Parity: reference clients/benchmark_client.cpp (1 GiB × 10 iters harness
// shape, avg latency + MB/s); adds p50/p99 and batch mode.
#include <algorithm>
#include <chrono>
#include <iostream>
#include <random>
#include <vector>

#include "blackbird/client/client.h"
#include "blackbird/client/gpu_client.h"

using namespace blackbird;
using Clock = std::chrono::steady_clock;

static double ms_since(Clock::time_point t0) {
  return std::chrono::duration<double, std::milli>(Clock::now() - t0).count();
}

static double pct(std::vector<double> v, double p) {
  if (v.empty()) return 0;
  std::sort(v.begin(), v.end());
  size_t i = static_cast<size_t>(p * (v.size() - 1));
  return v[i];
}

int main(int argc, char** argv) {
  ClientOptions opts;
  uint64_t size = 1ull << 20;
  int iters = 32;
  int batch = 0;   // 0 = single-object mode
  int gpu = -1;    // ≥0 = GPU mode: device buffers + GpuClient fused path
  bool upsert = true;  // steady-state mode: in-place upserts + batch sessions
  PlacementConfig pcfg;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> std::string { return i + 1 < argc ? argv[++i] : ""; };
    if (a == "--keystone") opts.keystone_endpoint = next();
    else if (a == "--size") size = strtoull(next().c_str(), nullptr, 10);
    else if (a == "--iters") iters = atoi(next().c_str());
    else if (a == "--batch") batch = atoi(next().c_str());
    else if (a == "--replication") pcfg.replication = atoi(next().c_str());
    else if (a == "--stripe") pcfg.max_workers_per_copy = atoi(next().c_str());
    else if (a == "--gpu") gpu = atoi(next().c_str());
    else if (a == "--no-upsert") upsert = false;
    else if (a == "--help" || a == "-h") {
      std::cout << "bb_bench [--keystone H:P] [--size BYTES] [--iters N]\n"
                   "         [--batch N] [--replication N] [--stripe N]\n"
                   "         [--gpu DEV]   device buffers + fused GPU data plane\n"
                   "         [--no-upsert] fresh keys + removes each iter (no\n"
                   "                       session fast path)\n";
      return 0;
    }
  }

  Client client(opts);
  if (auto r = client.connect(); !r.ok()) {
    std::cerr << "connect failed: " << r.message() << "\n";
    return 1;
  }

  std::vector<uint8_t> payload(size);
  std::mt19937_64 rng(42);
  for (auto& b : payload) b = static_cast<uint8_t>(rng());

  std::vector<double> put_ms, get_ms;
  double put_bytes = 0, get_bytes = 0;
  auto wall0 = Clock::now();

  if (gpu >= 0) {
    // GPU data plane: source/destination buffers in HBM, batches through the
    // fused put (copy+digest one launch) and fused get kernels.
    int n = batch > 0 ? batch : 1;
    GpuClient gc(client, gpu);
    if (auto r = gc.init(); !r.ok()) {
      std::cerr << "gpu init failed: " << r.message() << "\n";
      return 1;
    }
    gc.set_placement_cache(true);  // digest-verified RPC-free gets
    if (upsert && pcfg.replication <= 1) pcfg.replace = true;
    GpuClient::BatchPutSession psess;
    GpuClient::BatchGetSession gsess;
    auto src = gpu::device_malloc(size * n, gpu);
    auto dst = gpu::device_malloc(size * n, gpu);
    if (!src.ok() || !dst.ok()) {
      std::cerr << "device_malloc failed\n";
      return 1;
    }
    for (int b = 0; b < n; ++b)
      (void)gpu::fill_pattern(reinterpret_cast<void*>(src.value() + b * size),
                              size, 0x5eedULL + b, nullptr);
    (void)gpu::sync();
    std::vector<GpuClient::DevPutItem> puts;
    std::vector<GpuClient::DevGetItem> gets;
    std::vector<std::string> keys;
    for (int b = 0; b < n; ++b) {
      keys.push_back("bench/g" + std::to_string(b));
      puts.push_back({keys[b], reinterpret_cast<void*>(src.value() + b * size), size});
      gets.push_back({keys[b], reinterpret_cast<void*>(dst.value() + b * size), size});
    }
    for (int it = 0; it < iters; ++it) {
      auto t0 = Clock::now();
      auto pr = gc.batch_put_device(puts, pcfg, pcfg.replace ? &psess : nullptr);
      if (!pr.ok()) {
        std::cerr << "gpu batch_put failed: " << pr.message() << "\n";
        return 1;
      }
      for (size_t i = 0; i < pr.value().size(); ++i)
        if (pr.value()[i] != 0) {
          std::cerr << "gpu put status[" << i << "]=" << pr.value()[i] << "\n";
          return 1;
        }
      put_ms.push_back(ms_since(t0));
      put_bytes += static_cast<double>(size) * n;
      t0 = Clock::now();
      auto gr = gc.batch_get_device(gets, false, pcfg.replace ? &gsess : nullptr);
      if (!gr.ok()) {
        std::cerr << "gpu batch_get failed: " << gr.message() << "\n";
        return 1;
      }
      for (size_t i = 0; i < gr.value().size(); ++i)
        if (gr.value()[i] != 0) {
          std::cerr << "gpu get status[" << i << "]=" << gr.value()[i] << "\n";
          return 1;
        }
      get_ms.push_back(ms_since(t0));
      get_bytes += static_cast<double>(size) * n;
      if (!pcfg.replace) client.batch_remove(keys);
    }
    if (pcfg.replace) client.batch_remove(keys);
    // spot-verify last round's payloads against the known fill pattern
    for (int b = 0; b < n && b < 4; ++b) {
      auto bad = gpu::verify_pattern(
          reinterpret_cast<void*>(dst.value() + b * size), size,
          0x5eedULL + b, nullptr);
      if (!bad.ok() || bad.value() != 0) {
        std::cerr << "VERIFY FAILED on object " << b << ": "
                  << (bad.ok() ? std::to_string(bad.value()) + " bad u64s"
                               : bad.message())
                  << "\n";
        uint8_t s8[16] = {}, d8[16] = {};
        (void)gpu::download(s8, src.value() + b * size, sizeof(s8));
        (void)gpu::download(d8, dst.value() + b * size, sizeof(d8));
        auto hex = [](const uint8_t* p) {
          char out[40];
          for (int i = 0; i < 16; ++i) snprintf(out + 2 * i, 3, "%02x", p[i]);
          return std::string(out, 32);
        };
        std::cerr << "  src[0:16]=" << hex(s8) << "\n  dst[0:16]=" << hex(d8)
                  << "\n";
        return 1;
      }
    }
    (void)gpu::device_free(src.value());
    (void)gpu::device_free(dst.value());
  } else if (batch > 0) {
    std::vector<Client::PutItem> items;
    std::vector<std::string> keys;
    for (int b = 0; b < batch; ++b) keys.push_back("bench/b" + std::to_string(b));
    for (int b = 0; b < batch; ++b)
      items.push_back({keys[b], payload.data(), size});
    // steady-state mode (default): in-place upserts + host batch session —
    // two tiny RPCs per step around direct memcpys into the mapped pool
    if (upsert && pcfg.replication <= 1) pcfg.replace = true;
    Client::HostPutSession hsess;
    for (int it = 0; it < iters; ++it) {
      auto t0 = Clock::now();
      auto pr = client.batch_put(items, pcfg,
                                 pcfg.replace ? &hsess : nullptr);
      if (!pr.ok()) {
        std::cerr << "batch_put failed: " << pr.message() << "\n";
        return 1;
      }
      put_ms.push_back(ms_since(t0));
      put_bytes += static_cast<double>(size) * batch;
      t0 = Clock::now();
      auto gr = client.batch_get(keys);
      if (!gr.ok()) {
        std::cerr << "batch_get failed: " << gr.message() << "\n";
        return 1;
      }
      get_ms.push_back(ms_since(t0));
      get_bytes += static_cast<double>(size) * batch;
      if (!pcfg.replace) client.batch_remove(keys);
    }
    if (pcfg.replace) client.batch_remove(keys);
  } else {
    for (int it = 0; it < iters; ++it) {
      std::string key = "bench/k" + std::to_string(it);
      auto t0 = Clock::now();
      auto pr = client.put(key, payload.data(), size, pcfg);
      if (!pr.ok()) {
        std::cerr << "put failed: " << pr.message() << "\n";
        return 1;
      }
      put_ms.push_back(ms_since(t0));
      put_bytes += static_cast<double>(size);
      t0 = Clock::now();
      auto gr = client.get(key);
      if (!gr.ok()) {
        std::cerr << "get failed: " << gr.message() << "\n";
        return 1;
      }
      get_ms.push_back(ms_since(t0));
      get_bytes += static_cast<double>(size);
      client.remove(key);
    }
  }

  double wall_s = ms_since(wall0) / 1e3;
  auto sum = [](const std::vector<double>& v) {
    double s = 0;
    for (double x : v) s += x;
    return s;
  };
  std::cout.setf(std::ios::fixed);
  std::cout.precision(2);
  std::cout << "WRITE: avg " << sum(put_ms) / put_ms.size() << " ms, p50 "
            << pct(put_ms, 0.5) << " ms, p99 " << pct(put_ms, 0.99) << " ms, "
            << put_bytes / (sum(put_ms) / 1e3) / 1e6 << " MB/s\n";
  std::cout << "READ:  avg " << sum(get_ms) / get_ms.size() << " ms, p50 "
            << pct(get_ms, 0.5) << " ms, p99 " << pct(get_ms, 0.99) << " ms, "
            << get_bytes / (sum(get_ms) / 1e3) / 1e6 << " MB/s\n";
  std::cout << "TOTAL: " << (put_bytes + get_bytes) / wall_s / 1e6 << " MB/s over "
            << wall_s << " s\n";
  return 0;
}
