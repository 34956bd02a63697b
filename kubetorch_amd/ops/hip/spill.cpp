// Native checkpoint spill/restore engine for MI355X (host-side HIP).
//
// Spills GPU tensors to a file through a double-buffered pinned-host ring:
// hipMemcpyAsync D2H into pinned staging overlaps with fwrite of the
// previous chunk (and the reverse for restore), so a 100+ GB checkpoint
// moves at min(PCIe, disk) bandwidth without serializing through Python.
// This is the "pinned hipMemcpyAsync spill sized for 288 GB HBM/GPU" path
// named in BASELINE.json; python-side wrapper:
// kubetorch_amd/utils/checkpoint.py (save/load_engine_checkpoint_fast).

#include <torch/extension.h>

#include <hip/hip_runtime.h>

#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <cstdio>
#include <stdexcept>
#include <string>
#include <vector>

namespace {

constexpr size_t CHUNK = 64ull << 20;  // 64 MiB pinned chunks

#define HIP_CHECK(expr)                                                \
  do {                                                                 \
    hipError_t _e = (expr);                                            \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

struct PinnedRing {
  void* buf[2] = {nullptr, nullptr};
  hipEvent_t ev[2] = {nullptr, nullptr};
  PinnedRing() {
    for (int i = 0; i < 2; ++i) {
      HIP_CHECK(hipHostMalloc(&buf[i], CHUNK, hipHostMallocDefault));
      HIP_CHECK(hipEventCreateWithFlags(&ev[i], hipEventDisableTiming));
    }
  }
  ~PinnedRing() {
    for (int i = 0; i < 2; ++i) {
      if (buf[i]) (void)hipHostFree(buf[i]);
      if (ev[i]) (void)hipEventDestroy(ev[i]);
    }
  }
};

struct Chunk {
  const char* dev_ptr;
  size_t bytes;
};

std::vector<Chunk> plan_chunks(const std::vector<at::Tensor>& tensors) {
  std::vector<Chunk> chunks;
  for (const auto& t : tensors) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "spill needs contiguous GPU tensors");
    const char* p = (const char*)t.data_ptr();
    size_t left = t.numel() * t.element_size();
    while (left > 0) {
      size_t n = left < CHUNK ? left : CHUNK;
      chunks.push_back({p, n});
      p += n;
      left -= n;
    }
  }
  return chunks;
}

double spill_to_file(const std::vector<at::Tensor>& tensors,
                     const std::string& path) {
  if (tensors.empty()) return 0.0;
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(tensors[0].device());
  hipStream_t stream =
      c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  FILE* f = fopen(path.c_str(), "wb");
  TORCH_CHECK(f, "cannot open ", path);
  PinnedRing ring;
  auto chunks = plan_chunks(tensors);
  // wait for any producer work on the current stream before copying
  HIP_CHECK(hipStreamSynchronize(stream));
  auto t0 = std::chrono::steady_clock::now();
  size_t total = 0;
  for (size_t i = 0; i < chunks.size(); ++i) {
    const int s = i & 1;
    // issue D2H for chunk i into slot s
    HIP_CHECK(hipMemcpyAsync(ring.buf[s], chunks[i].dev_ptr, chunks[i].bytes,
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipEventRecord(ring.ev[s], stream));
    // while it flies, flush the previous chunk (other slot) to disk
    if (i > 0) {
      const int p = (i - 1) & 1;
      HIP_CHECK(hipEventSynchronize(ring.ev[p]));
      if (fwrite(ring.buf[p], 1, chunks[i - 1].bytes, f) !=
          chunks[i - 1].bytes) {
        fclose(f);
        TORCH_CHECK(false, "short write to ", path);
      }
    }
    total += chunks[i].bytes;
  }
  const int last = (chunks.size() - 1) & 1;
  HIP_CHECK(hipEventSynchronize(ring.ev[last]));
  TORCH_CHECK(fwrite(ring.buf[last], 1, chunks.back().bytes, f) ==
                  chunks.back().bytes,
              "short write to ", path);
  fclose(f);
  auto dt = std::chrono::duration<double>(std::chrono::steady_clock::now() -
                                          t0).count();
  return total / dt / 1e9;  // GB/s
}

double restore_from_file(const std::string& path,
                         std::vector<at::Tensor> tensors) {
  if (tensors.empty()) return 0.0;
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(tensors[0].device());
  hipStream_t stream =
      c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  FILE* f = fopen(path.c_str(), "rb");
  TORCH_CHECK(f, "cannot open ", path);
  PinnedRing ring;
  std::vector<Chunk> chunks = plan_chunks(tensors);
  auto t0 = std::chrono::steady_clock::now();
  size_t total = 0;
  for (size_t i = 0; i < chunks.size(); ++i) {
    const int s = i & 1;
    // slot s must have drained its previous H2D before we overwrite it
    if (i >= 2) HIP_CHECK(hipEventSynchronize(ring.ev[s]));
    if (fread(ring.buf[s], 1, chunks[i].bytes, f) != chunks[i].bytes) {
      fclose(f);
      TORCH_CHECK(false, "short read from ", path);
    }
    HIP_CHECK(hipMemcpyAsync((void*)chunks[i].dev_ptr, ring.buf[s],
                             chunks[i].bytes, hipMemcpyHostToDevice, stream));
    HIP_CHECK(hipEventRecord(ring.ev[s], stream));
    total += chunks[i].bytes;
  }
  HIP_CHECK(hipStreamSynchronize(stream));
  fclose(f);
  auto dt = std::chrono::duration<double>(std::chrono::steady_clock::now() -
                                          t0).count();
  return total / dt / 1e9;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("spill_to_file", &spill_to_file,
          "Write GPU tensors to a file via double-buffered pinned D2H; "
          "returns GB/s");
  mod.def("restore_from_file", &restore_from_file,
          "Read a file into GPU tensors via pinned H2D; returns GB/s");
}
