// sdma.hip — explicit SDMA-engine copies through ROCr.
//
// Why this exists (measured, profiles/copypath_r9 + rocprof trace): in a
// process where torch has initialized HIP, rocclr routes every pinned D2H
// hipMemcpyAsync through the __amd_rocclr_copyBuffer shader-blit kernel —
// it competes with compute kernels for CUs and H2D||D2H stops scaling
// (57 GB/s shared instead of ~2x57). The hardware has independent SDMA
// engines; ROCr exposes them directly via
// hsa_amd_memory_async_copy_on_engine. This module is the MI355X-native
// equivalent of the reference's copy-engine selection knobs
// (MPIR_CVAR_CH4_IPC_GPU_ENGINE_TYPE=copy_high_bandwidth, reference
// p2p/run.sh:7; SYCL_PI_LEVEL_ZERO_USE_COPY_ENGINE, run_sycl.sh:16):
// copies placed on a NAMED DMA engine, independent of runtime heuristics.

#include "include/hpk.h"

#include <hsa/hsa.h>
#include <hsa/hsa_ext_amd.h>

#include <mutex>
#include <stdexcept>
#include <vector>

namespace hpk {

namespace {

struct Agents {
  std::vector<hsa_agent_t> gpus;
  hsa_agent_t cpu{};
  bool have_cpu = false;
};

Agents& agents() {
  static Agents a;
  static std::once_flag once;
  std::call_once(once, [] {
    // HIP must be initialized first so ROCr is live (any hip* call does it).
    (void)hipFree(nullptr);
    hsa_iterate_agents(
        [](hsa_agent_t agent, void* data) -> hsa_status_t {
          auto* out = static_cast<Agents*>(data);
          hsa_device_type_t type;
          hsa_agent_get_info(agent, HSA_AGENT_INFO_DEVICE, &type);
          if (type == HSA_DEVICE_TYPE_GPU) {
            out->gpus.push_back(agent); // KFD order == HIP device order
          } else if (type == HSA_DEVICE_TYPE_CPU && !out->have_cpu) {
            out->cpu = agent;
            out->have_cpu = true;
          }
          return HSA_STATUS_SUCCESS;
        },
        &a);
  });
  return a;
}

hsa_agent_t agent_for_ptr(const void* ptr, hsa_agent_t gpu_agent,
                          bool& is_host) {
  hsa_amd_pointer_info_t info;
  info.size = sizeof(info);
  if (hsa_amd_pointer_info(const_cast<void*>(ptr), &info, nullptr, nullptr,
                           nullptr) == HSA_STATUS_SUCCESS) {
    if (info.type == HSA_EXT_POINTER_TYPE_LOCKED ||
        info.type == HSA_EXT_POINTER_TYPE_HSA) {
      // LOCKED = pinned host; HSA could be either — check the owner type
      if (info.type == HSA_EXT_POINTER_TYPE_LOCKED) {
        is_host = true;
        return agents().have_cpu ? agents().cpu : gpu_agent;
      }
      hsa_device_type_t t;
      hsa_agent_get_info(info.agentOwner, HSA_AGENT_INFO_DEVICE, &t);
      is_host = (t == HSA_DEVICE_TYPE_CPU);
      return info.agentOwner;
    }
  }
  is_host = false;
  return gpu_agent;
}

void check_hsa(hsa_status_t s, const char* what) {
  if (s != HSA_STATUS_SUCCESS) {
    const char* msg = nullptr;
    hsa_status_string(s, &msg);
    throw std::runtime_error(std::string("HSA error in ") + what + ": " +
                             (msg ? msg : "?"));
  }
}

} // namespace

namespace {
// Engine mask for copies src_agent -> dst_agent. The status API wants the
// REAL agent pair (gpu,gpu returns nothing useful on MI355X).
uint32_t engine_mask_for(hsa_agent_t dst_agent, hsa_agent_t src_agent) {
  uint32_t mask = 0;
  if (hsa_amd_memory_copy_engine_status(dst_agent, src_agent, &mask) !=
      HSA_STATUS_SUCCESS)
    return 0;
  return mask;
}

// engine_index-th set bit of mask (wrapping); 0 if mask empty.
uint32_t pick_engine_bit(uint32_t mask, int engine_index) {
  int n = __builtin_popcount(mask);
  if (n == 0) return 0;
  int want = engine_index % n;
  uint32_t m = mask;
  for (int i = 0; i < want; ++i) m &= m - 1; // drop lowest set bits
  return m & ~(m - 1);                       // lowest remaining set bit
}
} // namespace

int sdma_num_engines_pair(int dst_device, int src_device) {
  // engines usable for src_device -> dst_device copies (xGMI SDMA engines
  // for peer pairs: MI355X exposes 14 per GPU besides the 2 host engines)
  auto& a = agents();
  if (dst_device < 0 || dst_device >= (int)a.gpus.size() || src_device < 0 ||
      src_device >= (int)a.gpus.size())
    return 0;
  return __builtin_popcount(
      engine_mask_for(a.gpus[dst_device], a.gpus[src_device]));
}

int sdma_num_engines(int device) {
  auto& a = agents();
  if (device < 0 || device >= (int)a.gpus.size() || !a.have_cpu) return 0;
  // H2D engines (dst=gpu, src=cpu) and D2H engines (dst=cpu, src=gpu)
  uint32_t h2d = engine_mask_for(a.gpus[device], a.cpu);
  uint32_t d2h = engine_mask_for(a.cpu, a.gpus[device]);
  int nh = __builtin_popcount(h2d), nd = __builtin_popcount(d2h);
  return nh > nd ? nh : nd;
}

// Begin an explicit-engine async copy; returns an opaque handle to wait on.
// engine_index < 0 -> let ROCr pick (plain hsa_amd_memory_async_copy).
void* sdma_copy_begin(void* dst, const void* src, size_t nbytes, int device,
                      int engine_index) {
  auto& a = agents();
  if (device < 0 || device >= (int)a.gpus.size())
    throw std::runtime_error("sdma_copy_begin: bad device");
  hsa_agent_t gpu = a.gpus[device];

  bool dst_host = false, src_host = false;
  hsa_agent_t dst_agent = agent_for_ptr(dst, gpu, dst_host);
  hsa_agent_t src_agent = agent_for_ptr(src, gpu, src_host);

  auto* signal = new hsa_signal_t;
  check_hsa(hsa_signal_create(1, 0, nullptr, signal), "hsa_signal_create");

  hsa_status_t s = HSA_STATUS_ERROR;
  if (engine_index >= 0) {
    uint32_t bit =
        pick_engine_bit(engine_mask_for(dst_agent, src_agent), engine_index);
    if (bit != 0) {
      s = hsa_amd_memory_async_copy_on_engine(
          dst, dst_agent, src, src_agent, nbytes, 0, nullptr, *signal,
          (hsa_amd_sdma_engine_id_t)bit, /*force_copy_on_sdma=*/true);
    }
  }
  if (s != HSA_STATUS_SUCCESS) { // no engine requested/available: plain copy
    s = hsa_amd_memory_async_copy(dst, dst_agent, src, src_agent, nbytes, 0,
                                  nullptr, *signal);
  }
  if (s != HSA_STATUS_SUCCESS) {
    hsa_signal_destroy(*signal);
    delete signal;
    check_hsa(s, "hsa_amd_memory_async_copy(_on_engine)");
  }
  return signal;
}

void sdma_wait(void* handle) {
  auto* signal = static_cast<hsa_signal_t*>(handle);
  while (hsa_signal_wait_scacquire(*signal, HSA_SIGNAL_CONDITION_LT, 1,
                                   UINT64_MAX, HSA_WAIT_STATE_BLOCKED) >= 1) {
  }
  hsa_signal_destroy(*signal);
  delete signal;
}

} // namespace hpk
