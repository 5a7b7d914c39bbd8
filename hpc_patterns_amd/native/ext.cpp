// ext.cpp — pybind11 bindings for the MI355X-native core (_hpk module).
//
// Deliberately torch-free: Python passes raw device pointers
// (tensor.data_ptr()) and stream handles (torch.cuda.current_stream().
// cuda_stream, which IS a hipStream_t on ROCm), so the extension builds with
// plain hipcc, carries no torch ABI dependency, and the same library objects
// link into the standalone C++ binaries (cpp/*.cpp).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "include/hpk.h"

namespace py = pybind11;

namespace {

hipStream_t as_stream(uintptr_t s) { return reinterpret_cast<hipStream_t>(s); }

py::dict conc_bench_py(const std::string& mode,
                       const std::vector<std::string>& commands,
                       const std::map<std::string, size_t>& params,
                       bool enable_profiling, int n_queues, int n_repetitions,
                       bool verbose, int copy_engine) {
  hpk::ConcResult r;
  {
    py::gil_scoped_release release;
    r = hpk::conc_bench(mode, commands, params, enable_profiling, n_queues,
                        n_repetitions, verbose, copy_engine);
  }
  py::dict d;
  d["total_us"] = r.total_us;
  d["per_cmd_us"] = r.per_cmd_us;
  d["per_cmd_dev_ms"] = r.per_cmd_dev_ms;
  return d;
}

} // namespace

PYBIND11_MODULE(_hpk, m) {
  m.doc() = "MI355X-native HPC-patterns core (HIP/CDNA4 kernels, hipStream/"
            "hipGraph concurrency engine, xGMI topology, HIP-IPC)";

  // ---- kernels ----
  m.def("busy_wait",
        [](uintptr_t out, long tripcount, long globalsize, uintptr_t stream) {
          hpk::launch_busy_wait(reinterpret_cast<float*>(out), tripcount,
                                globalsize, as_stream(stream));
        },
        py::arg("out"), py::arg("tripcount"), py::arg("globalsize"),
        py::arg("stream") = 0);
  m.def("busy_wait_mfma",
        [](uintptr_t out, long tripcount, long n_waves, uintptr_t stream) {
          hpk::launch_busy_wait_mfma(reinterpret_cast<float*>(out), tripcount,
                                     n_waves, as_stream(stream));
        },
        py::arg("out"), py::arg("tripcount"), py::arg("n_waves"),
        py::arg("stream") = 0);
  m.def("gemm_bf16_nt",
        [](uintptr_t c, uintptr_t a, uintptr_t b, long m, long n, long k,
           uintptr_t stream, int xcd_swizzle) {
          hpk::launch_gemm_bf16_nt(reinterpret_cast<float*>(c),
                                   reinterpret_cast<const void*>(a),
                                   reinterpret_cast<const void*>(b), m, n, k,
                                   as_stream(stream), xcd_swizzle);
        },
        py::arg("c"), py::arg("a"), py::arg("b"), py::arg("m"), py::arg("n"),
        py::arg("k"), py::arg("stream") = 0, py::arg("xcd_swizzle") = 1);
  m.def("gemm_fp8_nt",
        [](uintptr_t c, uintptr_t a, uintptr_t b, long m, long n, long k,
           uintptr_t stream, int xcd_swizzle) {
          hpk::launch_gemm_fp8_nt(reinterpret_cast<float*>(c),
                                  reinterpret_cast<const void*>(a),
                                  reinterpret_cast<const void*>(b), m, n, k,
                                  as_stream(stream), xcd_swizzle);
        },
        py::arg("c"), py::arg("a"), py::arg("b"), py::arg("m"), py::arg("n"),
        py::arg("k"), py::arg("stream") = 0, py::arg("xcd_swizzle") = 0);
  m.def("gemm_i8_nt",
        [](uintptr_t c, uintptr_t a, uintptr_t b, long m, long n, long k,
           uintptr_t stream, int xcd_swizzle) {
          hpk::launch_gemm_i8_nt(reinterpret_cast<int*>(c),
                                 reinterpret_cast<const void*>(a),
                                 reinterpret_cast<const void*>(b), m, n, k,
                                 as_stream(stream), xcd_swizzle);
        },
        py::arg("c"), py::arg("a"), py::arg("b"), py::arg("m"), py::arg("n"),
        py::arg("k"), py::arg("stream") = 0, py::arg("xcd_swizzle") = 0);
  m.def("gemm_mxfp8_nt",
        [](uintptr_t c, uintptr_t a, uintptr_t b, uintptr_t as, uintptr_t bs,
           long m, long n, long k, uintptr_t stream, int xcd_swizzle) {
          hpk::launch_gemm_mxfp8_nt(reinterpret_cast<float*>(c),
                                    reinterpret_cast<const void*>(a),
                                    reinterpret_cast<const void*>(b),
                                    reinterpret_cast<const void*>(as),
                                    reinterpret_cast<const void*>(bs), m, n,
                                    k, as_stream(stream), xcd_swizzle);
        },
        py::arg("c"), py::arg("a"), py::arg("b"), py::arg("a_scale"),
        py::arg("b_scale"), py::arg("m"), py::arg("n"), py::arg("k"),
        py::arg("stream") = 0, py::arg("xcd_swizzle") = 0);
  m.def("gemm_mxfp4_nt",
        [](uintptr_t c, uintptr_t a, uintptr_t b, uintptr_t as, uintptr_t bs,
           long m, long n, long k, uintptr_t stream, int xcd_swizzle) {
          hpk::launch_gemm_mxfp4_nt(reinterpret_cast<float*>(c),
                                    reinterpret_cast<const void*>(a),
                                    reinterpret_cast<const void*>(b),
                                    reinterpret_cast<const void*>(as),
                                    reinterpret_cast<const void*>(bs), m, n,
                                    k, as_stream(stream), xcd_swizzle);
        },
        py::arg("c"), py::arg("a"), py::arg("b"), py::arg("a_scale"),
        py::arg("b_scale"), py::arg("m"), py::arg("n"), py::arg("k"),
        py::arg("stream") = 0, py::arg("xcd_swizzle") = 0);
  m.def("copy_kernel",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, uintptr_t stream) {
          hpk::launch_copy_kernel(reinterpret_cast<void*>(dst),
                                  reinterpret_cast<const void*>(src), nbytes,
                                  as_stream(stream));
        },
        py::arg("dst"), py::arg("src"), py::arg("nbytes"), py::arg("stream") = 0);
  m.def("copy_kernel_tuned",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, uintptr_t stream,
           int unroll, size_t grid_cap) {
          hpk::launch_copy_kernel_tuned(reinterpret_cast<void*>(dst),
                                        reinterpret_cast<const void*>(src),
                                        nbytes, as_stream(stream), unroll,
                                        grid_cap);
        },
        py::arg("dst"), py::arg("src"), py::arg("nbytes"), py::arg("stream") = 0,
        py::arg("unroll") = 1, py::arg("grid_cap") = 16384);
  m.def("fill_f32",
        [](uintptr_t dst, float value, size_t n, uintptr_t stream) {
          hpk::launch_fill_f32(reinterpret_cast<float*>(dst), value, n,
                               as_stream(stream));
        },
        py::arg("dst"), py::arg("value"), py::arg("n"), py::arg("stream") = 0);
  m.def("iota_f32",
        [](uintptr_t dst, size_t n, uintptr_t stream) {
          hpk::launch_iota_f32(reinterpret_cast<float*>(dst), n,
                               as_stream(stream));
        },
        py::arg("dst"), py::arg("n"), py::arg("stream") = 0);
  m.def("acc_f32",
        [](uintptr_t dst, uintptr_t src, size_t n, uintptr_t stream) {
          hpk::launch_acc_f32(reinterpret_cast<float*>(dst),
                              reinterpret_cast<const float*>(src), n,
                              as_stream(stream));
        },
        py::arg("dst"), py::arg("src"), py::arg("n"), py::arg("stream") = 0);
  m.def("acc_f32_nt",
        [](uintptr_t dst, uintptr_t src, size_t n, uintptr_t stream) {
          hpk::launch_acc_f32_nt(reinterpret_cast<float*>(dst),
                                 reinterpret_cast<const float*>(src), n,
                                 as_stream(stream));
        },
        py::arg("dst"), py::arg("src"), py::arg("n"), py::arg("stream") = 0);
  m.def("reduce_sum_f32",
        [](uintptr_t src, size_t n, uintptr_t stream) {
          py::gil_scoped_release release;
          return hpk::reduce_sum_f32(reinterpret_cast<const float*>(src), n,
                                     as_stream(stream));
        },
        py::arg("src"), py::arg("n"), py::arg("stream") = 0);

  // ---- concurrency engine ----
  m.attr("ALLOWED_MODES") = hpk::allowed_modes;
  m.def("mode_is_allowed", &hpk::mode_is_allowed);
  m.def("conc_bench", &conc_bench_py, py::arg("mode"), py::arg("commands"),
        py::arg("params"), py::arg("enable_profiling") = false,
        py::arg("n_queues") = -1, py::arg("n_repetitions") = 10,
        py::arg("verbose") = false, py::arg("copy_engine") = 0);

  // ---- topology ----
  m.def("device_count", [] {
    py::gil_scoped_release release;
    return hpk::device_count();
  });
  m.def("link_matrix", [] {
    std::vector<std::vector<hpk::LinkInfo>> mtx;
    {
      py::gil_scoped_release release;
      mtx = hpk::link_matrix();
    }
    py::list rows;
    for (auto& row : mtx) {
      py::list r;
      for (auto& li : row) {
        py::dict d;
        d["p2p"] = li.p2p_accessible;
        d["link_type"] = li.link_type;
        d["hops"] = li.hops;
        d["min_bw_mbps"] = li.min_bw_mbps;
        d["max_bw_mbps"] = li.max_bw_mbps;
        d["weight"] = li.weight;
        r.append(d);
      }
      rows.append(r);
    }
    return rows;
  });
  m.def("p2p_planes", [] {
    py::gil_scoped_release release;
    return hpk::p2p_planes();
  });
  m.def("partition_info", [] {
    std::vector<hpk::PartitionInfo> v;
    {
      py::gil_scoped_release release;
      v = hpk::partition_info();
    }
    py::list out;
    for (auto& pi : v) {
      py::dict d;
      d["compute"] = pi.compute;
      d["memory"] = pi.memory;
      out.append(d);
    }
    return out;
  });

  // ---- IPC / peer transport ----
  m.def("ipc_get_handle", [](uintptr_t dptr) {
    auto v = hpk::ipc_get_handle(reinterpret_cast<void*>(dptr));
    return py::bytes(reinterpret_cast<const char*>(v.data()), v.size());
  });
  m.def("ipc_open_handle", [](py::bytes handle) {
    std::string s = handle;
    std::vector<uint8_t> v(s.begin(), s.end());
    return reinterpret_cast<uintptr_t>(hpk::ipc_open_handle(v));
  });
  m.def("ipc_close_handle", [](uintptr_t dptr) {
    hpk::ipc_close_handle(reinterpret_cast<void*>(dptr));
  });
  m.def("enable_peer_access", &hpk::enable_peer_access);
  m.def("sdma_num_engines", &hpk::sdma_num_engines);
  m.def("sdma_num_engines_pair", &hpk::sdma_num_engines_pair);
  m.def("sdma_copy_begin",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, int device,
           int engine_index) {
          return reinterpret_cast<uintptr_t>(hpk::sdma_copy_begin(
              reinterpret_cast<void*>(dst), reinterpret_cast<const void*>(src),
              nbytes, device, engine_index));
        },
        py::arg("dst"), py::arg("src"), py::arg("nbytes"), py::arg("device") = 0,
        py::arg("engine_index") = -1);
  m.def("staged_copy",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, int device,
           int engine_index, bool h2d) {
          py::gil_scoped_release release;
          hpk::staged_copy(reinterpret_cast<void*>(dst),
                           reinterpret_cast<const void*>(src), nbytes, device,
                           engine_index, h2d);
        },
        py::arg("dst"), py::arg("src"), py::arg("nbytes"), py::arg("device") = 0,
        py::arg("engine_index") = 0, py::arg("h2d") = true);
  m.def("sdma_wait", [](uintptr_t handle) {
    py::gil_scoped_release release;
    hpk::sdma_wait(reinterpret_cast<void*>(handle));
  });
  m.def("trace_push", [](const std::string& n) { hpk::trace_push(n.c_str()); });
  m.def("trace_pop", &hpk::trace_pop);
  m.def("trace_mark", [](const std::string& n) { hpk::trace_mark(n.c_str()); });
  m.def("memcpy_peer_async",
        [](uintptr_t dst, int dst_dev, uintptr_t src, int src_dev,
           size_t nbytes, uintptr_t stream) {
          hpk::memcpy_peer_async(reinterpret_cast<void*>(dst), dst_dev,
                                 reinterpret_cast<const void*>(src), src_dev,
                                 nbytes, as_stream(stream));
        },
        py::arg("dst"), py::arg("dst_dev"), py::arg("src"), py::arg("src_dev"),
        py::arg("nbytes"), py::arg("stream") = 0);

  // ---- raw allocation helpers (for torch-free miniapp paths/tests) ----
  m.def("hip_malloc", [](size_t nbytes) {
    void* p = nullptr;
    hpk::check_hip(hipMalloc(&p, nbytes), "hipMalloc");
    return reinterpret_cast<uintptr_t>(p);
  });
  m.def("hip_free", [](uintptr_t p) {
    hpk::check_hip(hipFree(reinterpret_cast<void*>(p)), "hipFree");
  });
  m.def("host_malloc", [](size_t nbytes) {
    void* p = nullptr;
    hpk::check_hip(hipHostMalloc(&p, nbytes, hipHostMallocDefault),
                   "hipHostMalloc");
    return reinterpret_cast<uintptr_t>(p);
  });
  m.def("host_free", [](uintptr_t p) {
    hpk::check_hip(hipHostFree(reinterpret_cast<void*>(p)), "hipHostFree");
  });
  m.def("memcpy_async",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, uintptr_t stream) {
          hpk::check_hip(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                                        reinterpret_cast<const void*>(src),
                                        nbytes, hipMemcpyDefault,
                                        as_stream(stream)),
                         "hipMemcpyAsync");
        });
  // kind: 1=H2D 2=D2H 3=D2D 4=default (hipMemcpyKind values) — engine
  // selection differs between Default and explicit kinds on ROCm 7.2.
  m.def("memcpy_async_kind",
        [](uintptr_t dst, uintptr_t src, size_t nbytes, int kind,
           uintptr_t stream) {
          hpk::check_hip(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                                        reinterpret_cast<const void*>(src),
                                        nbytes, (hipMemcpyKind)kind,
                                        as_stream(stream)),
                         "hipMemcpyAsync(kind)");
        });
  m.def("device_synchronize", [] {
    py::gil_scoped_release release;
    hpk::check_hip(hipDeviceSynchronize(), "hipDeviceSynchronize");
  });
  m.def("stream_synchronize", [](uintptr_t stream) {
    py::gil_scoped_release release;
    hpk::check_hip(hipStreamSynchronize(as_stream(stream)),
                   "hipStreamSynchronize");
  });
  m.def("set_device", [](int dev) {
    hpk::check_hip(hipSetDevice(dev), "hipSetDevice");
  });
}
