// demodel_amd._hip — the MI355X landing-pipeline runtime.
//
// Native equivalent of the data plane the reference left inside goproxy's
// io.Copy loop (reference cmd/demodel/start.go:201-204; SURVEY.md §3.2 "HOT
// LOOP"), rebuilt for MI355X: pinned host ring slabs, hipMemcpyAsync on side
// streams, HIP events, socket->pinned drains that bypass Python, and DLPack
// export so landed HBM blobs become zero-copy torch tensors.
//
// Deliberately torch-header-free: every API takes raw pointers/handles, so
// the extension cross-compiles in seconds and has no ABI coupling.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <hip/hip_runtime.h>

#include <sys/socket.h>
#include <sys/types.h>
#include <unistd.h>
#include <cerrno>
#include <cstring>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error at " #expr ": ") +    \
                               hipGetErrorString(_e));                      \
    }                                                                       \
  } while (0)

// ------------------------------------------------------------------------
// kernels (defined in the .hip translation units)

extern "C" {
void launch_sha256_batch(const void* data, size_t nbytes, size_t chunk_bytes,
                         uint32_t* out_digests, int n_chunks,
                         hipStream_t stream);
void launch_sha256_chain_init(uint32_t* state, hipStream_t stream);
void launch_sha256_chain_update(uint32_t* state, const void* data,
                                size_t nblocks, hipStream_t stream);
void launch_scatter_ranges(const void* src, const uint64_t* desc,
                           int n_desc, hipStream_t stream);
void launch_cast_f32_to_bf16(const float* src, uint16_t* dst, size_t n,
                             hipStream_t stream);
void launch_gguf_dequant(int qtype, const void* src, uint16_t* dst_bf16,
                         int64_t n_blocks, hipStream_t stream);
void launch_inflate_streams(const uint64_t* desc, int n_streams,
                            int* status, hipStream_t stream);
void launch_zstd_frames(const uint64_t* desc, int n_frames, int* status,
                        hipStream_t stream, int window);
void launch_snappy_streams(const uint64_t* desc, int n_streams,
                           hipStream_t stream);
void launch_lz4_streams(const uint64_t* desc, int n_streams,
                        hipStream_t stream);
}

// ------------------------------------------------------------------------
// DLPack minimal ABI (dlpack.h v0.8 layout)

typedef struct {
  int32_t device_type;  // kDLROCM = 10
  int32_t device_id;
} DLDevice;

typedef struct {
  uint8_t code;  // kDLUInt = 1
  uint8_t bits;
  uint16_t lanes;
} DLDataType;

typedef struct {
  void* data;
  DLDevice device;
  int32_t ndim;
  DLDataType dtype;
  int64_t* shape;
  int64_t* strides;
  uint64_t byte_offset;
} DLTensor;

struct DLManagedTensor {
  DLTensor dl_tensor;
  void* manager_ctx;
  void (*deleter)(DLManagedTensor*);
};

// ------------------------------------------------------------------------

struct DeviceBufferImpl {
  void* ptr = nullptr;
  size_t nbytes = 0;
  int device = 0;
  ~DeviceBufferImpl() {
    if (ptr) hipFree(ptr);  // best-effort; errors unreportable in dtor
  }
};

struct DlpackCtx {
  std::shared_ptr<DeviceBufferImpl> buf;
  int64_t shape[1];
  DLManagedTensor tensor;
};

class DeviceBuffer {
 public:
  explicit DeviceBuffer(size_t nbytes) {
    impl_ = std::make_shared<DeviceBufferImpl>();
    HIP_CHECK(hipGetDevice(&impl_->device));
    HIP_CHECK(hipMalloc(&impl_->ptr, nbytes));
    impl_->nbytes = nbytes;
  }
  uintptr_t ptr() const { return (uintptr_t)impl_->ptr; }
  size_t nbytes() const { return impl_->nbytes; }

  // Export as a 1-D uint8 DLPack tensor; the capsule co-owns the buffer.
  py::capsule to_dlpack() const {
    auto* ctx = new DlpackCtx();
    ctx->buf = impl_;
    ctx->shape[0] = (int64_t)impl_->nbytes;
    DLTensor& t = ctx->tensor.dl_tensor;
    t.data = impl_->ptr;
    t.device = {10 /*kDLROCM*/, impl_->device};
    t.ndim = 1;
    t.dtype = {1 /*kDLUInt*/, 8, 1};
    t.shape = ctx->shape;
    t.strides = nullptr;
    t.byte_offset = 0;
    ctx->tensor.manager_ctx = ctx;
    ctx->tensor.deleter = [](DLManagedTensor* self) {
      delete static_cast<DlpackCtx*>(self->manager_ctx);
    };
    return py::capsule(&ctx->tensor, "dltensor", [](PyObject* cap) {
      // unconsumed capsule: free it ourselves
      if (PyCapsule_IsValid(cap, "dltensor")) {
        auto* t = static_cast<DLManagedTensor*>(
            PyCapsule_GetPointer(cap, "dltensor"));
        t->deleter(t);
      }
    });
  }

 private:
  std::shared_ptr<DeviceBufferImpl> impl_;
};

class PinnedPool {
 public:
  PinnedPool(size_t slab_bytes, int n_slabs) : slab_bytes_(slab_bytes) {
    slabs_.resize(n_slabs);
    for (auto& s : slabs_) HIP_CHECK(hipHostMalloc(&s, slab_bytes, 0));
  }
  ~PinnedPool() {
    for (auto& s : slabs_)
      if (s) hipHostFree(s);
  }
  int n_slabs() const { return (int)slabs_.size(); }
  size_t slab_bytes() const { return slab_bytes_; }
  uintptr_t slab_ptr(int i) const { return (uintptr_t)slabs_.at(i); }
  py::memoryview slab_view(int i) {
    return py::memoryview::from_memory(slabs_.at(i), slab_bytes_);
  }

 private:
  size_t slab_bytes_;
  std::vector<void*> slabs_;
};

class Stream {
 public:
  explicit Stream(int priority = 0) {
    HIP_CHECK(hipStreamCreateWithPriority(&s_, hipStreamNonBlocking,
                                          priority));
    owned_ = true;
  }
  ~Stream() {
    if (owned_ && s_) hipStreamDestroy(s_);
  }
  void sync() { HIP_CHECK(hipStreamSynchronize(s_)); }
  uintptr_t handle() const { return (uintptr_t)s_; }
  hipStream_t raw() const { return s_; }

 private:
  hipStream_t s_ = nullptr;
  bool owned_ = false;
};

class Event {
 public:
  Event() { HIP_CHECK(hipEventCreate(&e_)); }
  ~Event() {
    if (e_) hipEventDestroy(e_);
  }
  void record(uintptr_t stream) {
    HIP_CHECK(hipEventRecord(e_, (hipStream_t)stream));
  }
  void wait(uintptr_t stream) {
    HIP_CHECK(hipStreamWaitEvent((hipStream_t)stream, e_, 0));
  }
  void sync() { HIP_CHECK(hipEventSynchronize(e_)); }
  bool query() { return hipEventQuery(e_) == hipSuccess; }
  float elapsed_ms(const Event& start) {
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, start.e_, e_));
    return ms;
  }

 private:
  hipEvent_t e_ = nullptr;
};

// ------------------------------------------------------------------------
// socket -> pinned drains (no GIL, MSG_WAITALL = one syscall per slab)

static ssize_t drain_fd(int fd, uintptr_t dst, size_t want) {
  char* p = (char*)dst;
  size_t got = 0;
  while (got < want) {
    ssize_t n = recv(fd, p + got, want - got, MSG_WAITALL);
    if (n < 0) {
      if (errno == EINTR) continue;
      return -errno;
    }
    if (n == 0) break;  // EOF
    got += (size_t)n;
  }
  return (ssize_t)got;
}

static ssize_t read_file_into(int fd, uintptr_t dst, size_t want,
                              int64_t offset) {
  char* p = (char*)dst;
  size_t got = 0;
  while (got < want) {
    ssize_t n = offset >= 0
                    ? pread(fd, p + got, want - got, offset + (int64_t)got)
                    : read(fd, p + got, want - got);
    if (n < 0) {
      if (errno == EINTR) continue;
      return -errno;
    }
    if (n == 0) break;
    got += (size_t)n;
  }
  return (ssize_t)got;
}

// ------------------------------------------------------------------------

PYBIND11_MODULE(_hip, m) {
  m.doc() = "demodel-amd MI355X pipeline runtime + CDNA4 kernels";

  m.def("device_count", [] {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    return e == hipSuccess ? n : 0;
  });
  m.def("device_probe", [] {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    return py::make_tuple(n, std::string(hipGetErrorString(e)));
  });
  m.def("set_device", [](int d) { HIP_CHECK(hipSetDevice(d)); });
  m.def("device_sync", [] { HIP_CHECK(hipDeviceSynchronize()); },
        py::call_guard<py::gil_scoped_release>());

  py::class_<DeviceBuffer>(m, "DeviceBuffer")
      .def(py::init<size_t>(), py::arg("nbytes"),
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("ptr", &DeviceBuffer::ptr)
      .def_property_readonly("nbytes", &DeviceBuffer::nbytes)
      .def("to_dlpack", &DeviceBuffer::to_dlpack);

  py::class_<PinnedPool>(m, "PinnedPool")
      .def(py::init<size_t, int>(), py::arg("slab_bytes"),
           py::arg("n_slabs"), py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("n_slabs", &PinnedPool::n_slabs)
      .def_property_readonly("slab_bytes", &PinnedPool::slab_bytes)
      .def("slab_ptr", &PinnedPool::slab_ptr)
      .def("slab_view", &PinnedPool::slab_view);

  py::class_<Stream>(m, "Stream")
      .def(py::init<int>(), py::arg("priority") = 0)
      .def("sync", &Stream::sync,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("handle", &Stream::handle);

  py::class_<Event>(m, "Event")
      .def(py::init<>())
      .def("record", &Event::record)
      .def("wait", &Event::wait)
      .def("sync", &Event::sync, py::call_guard<py::gil_scoped_release>())
      .def("query", &Event::query)
      .def("elapsed_ms", &Event::elapsed_ms);

  m.def("h2d_async",
        [](uintptr_t dst, uintptr_t src, size_t n, uintptr_t stream) {
          HIP_CHECK(hipMemcpyAsync((void*)dst, (void*)src, n,
                                   hipMemcpyHostToDevice,
                                   (hipStream_t)stream));
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("d2h_async",
        [](uintptr_t dst, uintptr_t src, size_t n, uintptr_t stream) {
          HIP_CHECK(hipMemcpyAsync((void*)dst, (void*)src, n,
                                   hipMemcpyDeviceToHost,
                                   (hipStream_t)stream));
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("d2d_async",
        [](uintptr_t dst, uintptr_t src, size_t n, uintptr_t stream) {
          HIP_CHECK(hipMemcpyAsync((void*)dst, (void*)src, n,
                                   hipMemcpyDeviceToDevice,
                                   (hipStream_t)stream));
        },
        py::call_guard<py::gil_scoped_release>());

  m.def("drain_fd", &drain_fd, py::arg("fd"), py::arg("dst"),
        py::arg("want"), py::call_guard<py::gil_scoped_release>(),
        "recv() a TCP fd into a pinned slab without the GIL; returns bytes "
        "read (0..want; <0 = -errno)");
  m.def("read_file_into", &read_file_into, py::arg("fd"), py::arg("dst"),
        py::arg("want"), py::arg("offset") = -1,
        py::call_guard<py::gil_scoped_release>());

  // ---- kernel launches -------------------------------------------------
  m.def("sha256_batch",
        [](uintptr_t data, size_t nbytes, size_t chunk_bytes,
           uintptr_t out_digests, int n_chunks, uintptr_t stream) {
          launch_sha256_batch((const void*)data, nbytes, chunk_bytes,
                              (uint32_t*)out_digests, n_chunks,
                              (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>(),
        "SHA-256 of n_chunks independent chunks of `data` (last may be "
        "short); digests -> out (n_chunks x 8 u32, big-endian words)");
  m.def("sha256_chain_init",
        [](uintptr_t state, uintptr_t stream) {
          launch_sha256_chain_init((uint32_t*)state, (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("sha256_chain_update",
        [](uintptr_t state, uintptr_t data, size_t nblocks,
           uintptr_t stream) {
          launch_sha256_chain_update((uint32_t*)state, (const void*)data,
                                     nblocks, (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>(),
        "advance one whole-blob SHA-256 chain over nblocks 64-byte blocks "
        "already resident in HBM");
  m.def("scatter_ranges",
        [](uintptr_t src, uintptr_t desc, int n_desc, uintptr_t stream) {
          launch_scatter_ranges((const void*)src, (const uint64_t*)desc,
                                n_desc, (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>(),
        "copy n_desc (src_off,dst_ptr,len) descriptors from a landed blob "
        "into tensor storages");
  m.def("cast_f32_to_bf16",
        [](uintptr_t src, uintptr_t dst, size_t n, uintptr_t stream) {
          launch_cast_f32_to_bf16((const float*)src, (uint16_t*)dst, n,
                                  (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("gguf_dequant",
        [](int qtype, uintptr_t src, uintptr_t dst, int64_t n_blocks,
           uintptr_t stream) {
          launch_gguf_dequant(qtype, (const void*)src, (uint16_t*)dst,
                              n_blocks, (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("inflate_streams",
        [](uintptr_t desc, int n_streams, uintptr_t stream) {
          launch_inflate_streams((const uint64_t*)desc, n_streams, nullptr,
                                 (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>(),
        "DEFLATE-inflate n_streams descriptors (8 u64 each: src, src_len, "
        "dst, dst_cap, written, status, consumed, pad), wave per stream");
  m.def("zstd_frames",
        [](uintptr_t desc, int n_frames, uintptr_t stream, int window) {
          launch_zstd_frames((const uint64_t*)desc, n_frames, nullptr,
                             (hipStream_t)stream, window);
        },
        py::call_guard<py::gil_scoped_release>(), py::arg("desc"),
        py::arg("n_frames"), py::arg("stream"), py::arg("window") = 0,
        "zstd-decompress n_frames descriptors (8 u64 each: src, src_len, "
        "dst, dst_cap, written, status, consumed, ws), wave per frame; ws "
        "needs >= 144 KiB per frame; window 0=auto/16384/65536 selects "
        "the LDS window template");
  m.def("snappy_streams",
        [](uintptr_t desc, int n_streams, uintptr_t stream) {
          launch_snappy_streams((const uint64_t*)desc, n_streams,
                                (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>(),
        "snappy-decompress n_streams descriptors (8 u64 each: src, "
        "src_len, dst, dst_cap, written, status, consumed, pad), wave "
        "per stream (parquet's default page codec)");
  m.def("lz4_streams",
        [](uintptr_t desc, int n_streams, uintptr_t stream) {
          launch_lz4_streams((const uint64_t*)desc, n_streams,
                             (hipStream_t)stream);
        },
        py::call_guard<py::gil_scoped_release>(),
        "LZ4 raw-block decompress n_streams descriptors (8 u64 each: "
        "src, src_len, dst, dst_cap, written, status, consumed, pad), "
        "wave per stream (parquet LZ4/LZ4_RAW page codecs)");
}
