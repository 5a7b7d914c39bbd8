// topology_main.cpp — hpk_topology: xGMI link topology tool.
//
// MI355X-native replacement of the reference Level-Zero Sysman fabric tool
// (reference p2p/topology.cpp:26-107). CLI contract preserved:
//   hpk_topology          -> print link matrix + connectivity planes
//   hpk_topology X        -> print the X-th GPU id in topology order, so a
//                            launcher can bind consecutive ranks to
//                            directly-connected GPUs (consumed by
//                            scripts/gpu_mapping.sh, the analog of the
//                            reference tile_mapping.sh compact_plan policy).
//
// On a single-node 8x MI355X the fabric is fully connected (7 xGMI links per
// GPU), so "planes" collapse to one; the actionable data is the per-pair
// link-type/hops/weight/bandwidth matrix.

#include "../hpc_patterns_amd/native/include/hpk.h"

#include <cstdio>
#include <cstdlib>
#include <string>

int main(int argc, char* argv[]) {
  int n = 0;
  try {
    n = hpk::device_count();
  } catch (const std::exception& e) {
    std::fprintf(stderr, "No HIP devices visible: %s\n", e.what());
    return 1;
  }
  if (n == 0) {
    std::fprintf(stderr, "No HIP devices visible\n");
    return 1;
  }

  auto planes = hpk::p2p_planes();

  if (argc > 1) {
    // Flatten planes and print the X-th GPU id (reference topology.cpp:91-106:
    // consecutive ranks land on directly-connected devices).
    long x = std::strtol(argv[1], nullptr, 10);
    std::vector<int> flat;
    for (const auto& p : planes) flat.insert(flat.end(), p.begin(), p.end());
    if (flat.empty()) return 1;
    std::printf("%d\n", flat[((x % (long)flat.size()) + flat.size()) % flat.size()]);
    return 0;
  }

  auto m = hpk::link_matrix();
  std::printf("# %d HIP device(s)\n", n);
  std::printf("# link matrix: type 2=xGMI (hipExtGetLinkTypeAndHopCount), "
              "weight/bw from rocm_smi\n");
  std::printf("%4s", "");
  for (int j = 0; j < n; ++j) std::printf("%14s%d", "gpu", j);
  std::printf("\n");
  for (int i = 0; i < n; ++i) {
    std::printf("gpu%d", i);
    for (int j = 0; j < n; ++j) {
      if (i == j) {
        std::printf("%15s", "-");
        continue;
      }
      const auto& li = m[i][j];
      char buf[64];
      std::snprintf(buf, sizeof(buf), "p2p=%d t=%d h=%d", li.p2p_accessible,
                    li.link_type, li.hops);
      std::printf("%15s", buf);
    }
    std::printf("\n");
  }
  std::printf("# per-pair bandwidth (rocm_smi min-max, MB/s):\n");
  for (int i = 0; i < n; ++i)
    for (int j = i + 1; j < n; ++j)
      if (m[i][j].max_bw_mbps > 0)
        std::printf("gpu%d <-> gpu%d : %ld - %ld MB/s (weight %ld)\n", i, j,
                    m[i][j].min_bw_mbps, m[i][j].max_bw_mbps, m[i][j].weight);

  auto parts = hpk::partition_info();
  if (!parts.empty()) {
    std::printf("# partition modes (compute/memory):\n");
    for (size_t i = 0; i < parts.size(); ++i)
      std::printf("gpu%zu: %s / %s\n", i,
                  parts[i].compute.empty() ? "?" : parts[i].compute.c_str(),
                  parts[i].memory.empty() ? "?" : parts[i].memory.c_str());
  }

  std::printf("# connectivity planes (direct-P2P reachability):\n");
  for (size_t p = 0; p < planes.size(); ++p) {
    std::printf("plane %zu:", p);
    for (int d : planes[p]) std::printf(" gpu%d", d);
    std::printf("\n");
  }
  return 0;
}
