// hipIpc-based cross-process GPU memory sharing for prefill->decode KV
// transfer over xGMI (the MI355X-native replacement for the reference's
// NIXL transfer engine; see SURVEY.md §2.4 "NIXL call sites").
//
// The KV pool of a prefill worker is allocated with hipMalloc (outside the
// torch caching allocator so the IPC handle maps exactly the pool) and
// exported; the decode worker opens the handle and pulls pages with the
// page-copy kernels (reads travel over the direct xGMI link).
//
// Requires HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC) on this driver stack.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <pybind11/pybind11.h>

namespace py = pybind11;

// Allocate `nbytes` with hipMalloc on `device` and wrap as a torch tensor
// (uint8). Freed when the tensor dies.
torch::Tensor ipc_alloc(int64_t nbytes, int64_t device) {
  hipError_t err = hipSetDevice((int)device);
  TORCH_CHECK(err == hipSuccess, "hipSetDevice failed: ", hipGetErrorString(err));
  void* ptr = nullptr;
  err = hipMalloc(&ptr, (size_t)nbytes);
  TORCH_CHECK(err == hipSuccess, "hipMalloc(", nbytes, ") failed: ",
              hipGetErrorString(err));
  auto options = torch::TensorOptions()
                     .dtype(torch::kUInt8)
                     .device(torch::kCUDA, (int)device);
  return torch::from_blob(ptr, {nbytes}, [](void* p) { hipFree(p); }, options);
}

py::bytes ipc_export(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda());
  hipIpcMemHandle_t handle;
  hipError_t err = hipIpcGetMemHandle(&handle, t.data_ptr());
  TORCH_CHECK(err == hipSuccess, "hipIpcGetMemHandle failed: ",
              hipGetErrorString(err));
  return py::bytes(reinterpret_cast<const char*>(&handle), sizeof(handle));
}

// Open a peer process's exported buffer; returns a tensor view of it on
// `device` (the local device through which the mapping is made).
torch::Tensor ipc_open(py::bytes handle_bytes, int64_t nbytes, int64_t device) {
  std::string s = handle_bytes;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t), "bad ipc handle size");
  hipIpcMemHandle_t handle;
  memcpy(&handle, s.data(), sizeof(handle));
  hipError_t err = hipSetDevice((int)device);
  TORCH_CHECK(err == hipSuccess);
  void* ptr = nullptr;
  err = hipIpcOpenMemHandle(&ptr, handle, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess, "hipIpcOpenMemHandle failed: ",
              hipGetErrorString(err));
  auto options = torch::TensorOptions()
                     .dtype(torch::kUInt8)
                     .device(torch::kCUDA, (int)device);
  return torch::from_blob(
      ptr, {nbytes}, [](void* p) { hipIpcCloseMemHandle(p); }, options);
}

void enable_peer_access(int64_t device, int64_t peer) {
  hipError_t err = hipSetDevice((int)device);
  TORCH_CHECK(err == hipSuccess);
  err = hipDeviceEnablePeerAccess((int)peer, 0);
  TORCH_CHECK(err == hipSuccess || err == hipErrorPeerAccessAlreadyEnabled,
              "hipDeviceEnablePeerAccess failed: ", hipGetErrorString(err));
}
