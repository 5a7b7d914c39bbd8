// ipc.hip — HIP-IPC one-sided transport.
//
// MI355X-native analog of the reference one-sided RMA engine
// (reference p2p/peer2pear.cpp:68-102: MPI_Win_create / MPI_Put / fence):
// a process exports a hipIpcMemHandle_t for its device buffer, the peer
// opens it and writes directly over xGMI with hipMemcpyAsync /
// hipMemcpyPeerAsync — a true one-sided put with no receiver involvement.
// Requires the dmabuf IPC mode (HSA_ENABLE_IPC_MODE_LEGACY=0).

#include "include/hpk.h"

#include <cstring>
#include <stdexcept>

namespace hpk {

std::vector<uint8_t> ipc_get_handle(void* dptr) {
  hipIpcMemHandle_t h;
  check_hip(hipIpcGetMemHandle(&h, dptr), "hipIpcGetMemHandle");
  std::vector<uint8_t> out(sizeof(h));
  std::memcpy(out.data(), &h, sizeof(h));
  return out;
}

void* ipc_open_handle(const std::vector<uint8_t>& handle) {
  if (handle.size() != sizeof(hipIpcMemHandle_t))
    throw std::runtime_error("bad IPC handle size");
  hipIpcMemHandle_t h;
  std::memcpy(&h, handle.data(), sizeof(h));
  void* ptr = nullptr;
  check_hip(hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess),
            "hipIpcOpenMemHandle");
  return ptr;
}

void ipc_close_handle(void* dptr) {
  check_hip(hipIpcCloseMemHandle(dptr), "hipIpcCloseMemHandle");
}

void enable_peer_access(int peer_device) {
  hipError_t e = hipDeviceEnablePeerAccess(peer_device, 0);
  if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled)
    check_hip(e, "hipDeviceEnablePeerAccess");
  (void)hipGetLastError(); // clear sticky already-enabled state
}

void memcpy_peer_async(void* dst, int dst_dev, const void* src, int src_dev,
                       size_t nbytes, hipStream_t stream) {
  check_hip(hipMemcpyPeerAsync(dst, dst_dev, src, src_dev, nbytes, stream),
            "hipMemcpyPeerAsync");
}

} // namespace hpk
