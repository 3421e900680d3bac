// hipIpc tensor transport — the GPU tier of the object store.
//
// Role parity: reference Ray Direct Transport / GPU objects
// (python/ray/experimental/gpu_object_manager/, doc
// doc/source/ray-core/direct-transport.rst:12) which moves tensors
// actor-to-actor via NCCL/GLOO/NIXL. MI355X-native design: intra-node
// handoff uses hipIpcMemHandle — the consumer maps the producer's HBM pages
// directly (zero host copies, zero device copies for read-only use).
// Requires HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC) on this driver stack.
//
// Torch's caching allocator hands out offsets into larger allocations, so the
// exported descriptor is (handle of base allocation, offset, nbytes); the
// base address/size come from hipPointerGetAttribute RANGE_START/RANGE_SIZE.
// Opened handles are cached per (handle bytes, device): hipIpcOpenMemHandle
// may only be opened once per process per allocation.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstring>
#include <map>
#include <mutex>
#include <string>

namespace {

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e),     \
                " at " #expr);                                              \
  } while (0)

struct OpenedHandle {
  void* base = nullptr;
  int64_t refcount = 0;
};

std::mutex g_mu;
std::map<std::string, OpenedHandle> g_opened;  // key: handle bytes + device

// Export: returns (handle_bytes, offset_in_allocation, alloc_size)
py::tuple export_handle(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda(), "export_handle needs a GPU tensor");
  TORCH_CHECK(t.is_contiguous(), "export_handle needs a contiguous tensor");
  void* ptr = t.data_ptr();
  // find the base of the caching-allocator block
  hipDeviceptr_t base = nullptr;
  size_t range_size = 0;
  HIP_CHECK(hipPointerGetAttribute(&base, HIP_POINTER_ATTRIBUTE_RANGE_START_ADDR,
                                   (hipDeviceptr_t)ptr));
  HIP_CHECK(hipPointerGetAttribute(&range_size, HIP_POINTER_ATTRIBUTE_RANGE_SIZE,
                                   (hipDeviceptr_t)ptr));
  hipIpcMemHandle_t handle;
  HIP_CHECK(hipIpcGetMemHandle(&handle, (void*)base));
  int64_t offset = (char*)ptr - (char*)base;
  return py::make_tuple(
      py::bytes(reinterpret_cast<const char*>(&handle), sizeof(handle)),
      offset, (int64_t)range_size);
}

void close_handle_key(const std::string& key) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto it = g_opened.find(key);
  if (it == g_opened.end()) return;
  if (--it->second.refcount <= 0) {
    hipIpcCloseMemHandle(it->second.base);
    g_opened.erase(it);
  }
}

// Import: maps the peer allocation and returns a tensor view (zero-copy).
// The returned tensor owns a refcount on the opened handle.
torch::Tensor import_handle(py::bytes handle_bytes, int64_t offset,
                            std::vector<int64_t> shape,
                            torch::ScalarType dtype, int64_t device) {
  std::string hb = handle_bytes;
  TORCH_CHECK(hb.size() == sizeof(hipIpcMemHandle_t), "bad handle size");
  std::string key = hb + "@" + std::to_string(device);
  HIP_CHECK(hipSetDevice((int)device));
  void* base = nullptr;
  {
    std::lock_guard<std::mutex> lk(g_mu);
    auto it = g_opened.find(key);
    if (it != g_opened.end()) {
      it->second.refcount++;
      base = it->second.base;
    }
  }
  if (base == nullptr) {
    hipIpcMemHandle_t handle;
    std::memcpy(&handle, hb.data(), sizeof(handle));
    HIP_CHECK(hipIpcOpenMemHandle(&base, handle, hipIpcMemLazyEnablePeerAccess));
    std::lock_guard<std::mutex> lk(g_mu);
    auto& oh = g_opened[key];
    oh.base = base;
    oh.refcount++;
  }
  void* ptr = (char*)base + offset;
  auto options = torch::TensorOptions().dtype(dtype).device(torch::kCUDA, device);
  auto deleter = [key](void*) { close_handle_key(key); };
  return torch::from_blob(ptr, shape, deleter, options);
}

int64_t num_opened() {
  std::lock_guard<std::mutex> lk(g_mu);
  return (int64_t)g_opened.size();
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("export_handle", &export_handle);
  m.def("import_handle", &import_handle, py::arg("handle"), py::arg("offset"),
        py::arg("shape"), py::arg("dtype"), py::arg("device") = 0);
  m.def("num_opened", &num_opened);
}
