// hipsnap: HBM arena + pinned double-buffered streaming + GPU snapshot.
//
// The MI355X cold-start device path (BASELINE.json north star): one reserved
// HBM slab per GPU process, filled by a pinned-staging hipMemcpyAsync
// pipeline (file -> pinned ring -> HBM, two streams so disk reads, host
// memcpy and DMA overlap), and drained the same way for GPU snapshots.
// This replaces the reference's CRIU+cuda-checkpoint GPU state capture
// (reference: pkg/worker/criu_nvidia.go, docker/Dockerfile.worker CRIU
// stage) with a driver-independent HIP-level snapshot.
//
// Sized for MI355X: 288 GB HBM3E per GPU, host link PCIe Gen5 x16
// (~63 GB/s) — the staging ring uses 2 x 128 MiB pinned buffers by default,
// large enough to hit peak PCIe bandwidth, small enough to stay resident.

#include <hip/hip_runtime.h>

#include <torch/extension.h>

#include <atomic>
#include <cstdio>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " __FILE__ ":" +   \
                               std::to_string(__LINE__));                      \
    }                                                                          \
  } while (0)

namespace {

class Arena {
 public:
  Arena(int device, int64_t size_bytes, int64_t staging_bytes)
      : device_(device), size_(size_bytes) {
    HIP_CHECK(hipSetDevice(device_));
    HIP_CHECK(hipMalloc(&base_, size_));
    staging_half_ = staging_bytes / 2;
    if (staging_half_ < (1 << 20)) staging_half_ = (1 << 20);
    HIP_CHECK(hipHostMalloc(&pinned_[0], staging_half_, hipHostMallocDefault));
    HIP_CHECK(hipHostMalloc(&pinned_[1], staging_half_, hipHostMallocDefault));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_[0], hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_[1], hipStreamNonBlocking));
  }

  ~Arena() {
    hipSetDevice(device_);
    hipStreamDestroy(stream_[0]);
    hipStreamDestroy(stream_[1]);
    hipHostFree(pinned_[0]);
    hipHostFree(pinned_[1]);
    hipFree(base_);
  }

  int64_t base_ptr() const { return reinterpret_cast<int64_t>(base_); }
  int64_t size() const { return size_; }

  // H2D: host buffer -> arena[offset], pinned double-buffered. Accepts any
  // buffer-protocol object without copying; GIL released for the transfer.
  void upload(int64_t offset, py::buffer data) {
    py::buffer_info info = data.request();
    const char* src = (const char*)info.ptr;
    int64_t nbytes = (int64_t)(info.size * info.itemsize);
    py::gil_scoped_release release;
    upload_raw(offset, src, nbytes);
  }

  void upload_raw(int64_t offset, const char* src, int64_t nbytes) {
    check_range(offset, nbytes);
    HIP_CHECK(hipSetDevice(device_));
    int64_t done = 0;
    int buf = 0;
    // Pipeline: memcpy chunk i into pinned[buf] while pinned[buf^1]'s DMA
    // is in flight; wait only for the buffer we are about to reuse.
    while (done < nbytes) {
      int64_t chunk = std::min(staging_half_, nbytes - done);
      HIP_CHECK(hipStreamSynchronize(stream_[buf]));  // buffer free?
      std::memcpy(pinned_[buf], src + done, chunk);
      HIP_CHECK(hipMemcpyAsync((char*)base_ + offset + done, pinned_[buf],
                               chunk, hipMemcpyHostToDevice, stream_[buf]));
      done += chunk;
      buf ^= 1;
    }
    HIP_CHECK(hipStreamSynchronize(stream_[0]));
    HIP_CHECK(hipStreamSynchronize(stream_[1]));
    h2d_bytes_ += nbytes;
  }

  // Stream a file region straight into the arena (cold-start weight path):
  // fread into pinned ring overlaps the previous chunk's DMA.
  int64_t upload_file(int64_t offset, const std::string& path,
                      int64_t file_offset, int64_t nbytes) {
    check_range(offset, nbytes);
    HIP_CHECK(hipSetDevice(device_));
    FILE* f = std::fopen(path.c_str(), "rb");
    if (!f) throw std::runtime_error("cannot open " + path);
    if (file_offset > 0 && std::fseek(f, (long)file_offset, SEEK_SET) != 0) {
      std::fclose(f);
      throw std::runtime_error("seek failed in " + path);
    }
    int64_t done = 0;
    int buf = 0;
    while (done < nbytes) {
      int64_t chunk = std::min(staging_half_, nbytes - done);
      HIP_CHECK(hipStreamSynchronize(stream_[buf]));
      size_t got = std::fread(pinned_[buf], 1, (size_t)chunk, f);
      if ((int64_t)got != chunk) {
        std::fclose(f);
        throw std::runtime_error("short read from " + path);
      }
      HIP_CHECK(hipMemcpyAsync((char*)base_ + offset + done, pinned_[buf],
                               chunk, hipMemcpyHostToDevice, stream_[buf]));
      done += chunk;
      buf ^= 1;
    }
    HIP_CHECK(hipStreamSynchronize(stream_[0]));
    HIP_CHECK(hipStreamSynchronize(stream_[1]));
    std::fclose(f);
    h2d_bytes_ += nbytes;
    return nbytes;
  }

  // D2H into a bytes object (allocated once; GIL released for the copy).
  py::bytes download(int64_t offset, int64_t nbytes) {
    check_range(offset, nbytes);
    HIP_CHECK(hipSetDevice(device_));
    PyObject* obj = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)nbytes);
    if (!obj) throw std::bad_alloc();
    char* out = PyBytes_AS_STRING(obj);
    {
      py::gil_scoped_release release;
      download_raw(offset, out, nbytes);
    }
    return py::reinterpret_steal<py::bytes>(obj);
  }

  void download_raw(int64_t offset, char* out, int64_t nbytes) {
    int64_t done = 0;
    int buf = 0;
    int64_t pending_off[2] = {-1, -1};
    int64_t pending_len[2] = {0, 0};
    while (done < nbytes) {
      int64_t chunk = std::min(staging_half_, nbytes - done);
      if (pending_off[buf] >= 0) {
        HIP_CHECK(hipStreamSynchronize(stream_[buf]));
        std::memcpy(&out[pending_off[buf]], pinned_[buf],
                    (size_t)pending_len[buf]);
      }
      HIP_CHECK(hipMemcpyAsync(pinned_[buf], (char*)base_ + offset + done,
                               chunk, hipMemcpyDeviceToHost, stream_[buf]));
      pending_off[buf] = done;
      pending_len[buf] = chunk;
      done += chunk;
      buf ^= 1;
    }
    for (int b = 0; b < 2; b++) {
      if (pending_off[b] >= 0) {
        HIP_CHECK(hipStreamSynchronize(stream_[b]));
        std::memcpy(&out[pending_off[b]], pinned_[b], (size_t)pending_len[b]);
      }
    }
    d2h_bytes_ += nbytes;
  }

  // Stream an arena region to a file (snapshot path).
  int64_t download_file(int64_t offset, const std::string& path,
                        int64_t nbytes, bool append) {
    check_range(offset, nbytes);
    HIP_CHECK(hipSetDevice(device_));
    FILE* f = std::fopen(path.c_str(), append ? "ab" : "wb");
    if (!f) throw std::runtime_error("cannot open " + path);
    int64_t done = 0;
    int buf = 0;
    int64_t pending_len[2] = {0, 0};
    while (done < nbytes) {
      int64_t chunk = std::min(staging_half_, nbytes - done);
      if (pending_len[buf] > 0) {
        HIP_CHECK(hipStreamSynchronize(stream_[buf]));
        std::fwrite(pinned_[buf], 1, (size_t)pending_len[buf], f);
      }
      HIP_CHECK(hipMemcpyAsync(pinned_[buf], (char*)base_ + offset + done,
                               chunk, hipMemcpyDeviceToHost, stream_[buf]));
      pending_len[buf] = chunk;
      done += chunk;
      buf ^= 1;
    }
    // Drain in ring order (oldest pending first) so file bytes stay
    // sequential: `buf` is the next ring slot = the older pending.
    for (int i = 0; i < 2; i++) {
      int b = (buf + i) & 1;
      if (pending_len[b] > 0) {
        HIP_CHECK(hipStreamSynchronize(stream_[b]));
        std::fwrite(pinned_[b], 1, (size_t)pending_len[b], f);
        pending_len[b] = 0;
      }
    }
    std::fclose(f);
    d2h_bytes_ += nbytes;
    return nbytes;
  }

  // Zero-copy torch view over an arena region (uint8).
  torch::Tensor tensor_view(int64_t offset, int64_t nbytes) {
    check_range(offset, nbytes);
    auto options = torch::TensorOptions()
                       .dtype(torch::kUInt8)
                       .device(torch::kCUDA, device_);
    return torch::from_blob((char*)base_ + offset, {nbytes}, options);
  }

  py::dict stats() const {
    py::dict d;
    d["h2d_bytes"] = (int64_t)h2d_bytes_;
    d["d2h_bytes"] = (int64_t)d2h_bytes_;
    d["staging_half_bytes"] = staging_half_;
    return d;
  }

 private:
  void check_range(int64_t offset, int64_t nbytes) const {
    if (offset < 0 || nbytes < 0 || offset + nbytes > size_) {
      throw std::out_of_range("arena range [" + std::to_string(offset) + ", " +
                              std::to_string(offset + nbytes) + ") outside " +
                              std::to_string(size_));
    }
  }

  int device_;
  int64_t size_;
  void* base_ = nullptr;
  void* pinned_[2] = {nullptr, nullptr};
  int64_t staging_half_ = 0;
  hipStream_t stream_[2];
  std::atomic<int64_t> h2d_bytes_{0};
  std::atomic<int64_t> d2h_bytes_{0};
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "hipsnap: HBM arena + pinned streaming + GPU snapshot (gfx950)";
  py::class_<Arena>(m, "Arena")
      .def(py::init<int, int64_t, int64_t>(), py::arg("device") = 0,
           py::arg("size_bytes") = (int64_t)4 << 30,
           py::arg("staging_bytes") = (int64_t)256 << 20)
      .def("base_ptr", &Arena::base_ptr)
      .def("size", &Arena::size)
      .def("upload", &Arena::upload)
      .def("upload_file", &Arena::upload_file,
           py::call_guard<py::gil_scoped_release>())
      .def("download", &Arena::download)
      .def("download_file", &Arena::download_file, py::arg("offset"),
           py::arg("path"), py::arg("nbytes"), py::arg("append") = false,
           py::call_guard<py::gil_scoped_release>())
      .def("tensor_view", &Arena::tensor_view)
      .def("stats", &Arena::stats);
}
