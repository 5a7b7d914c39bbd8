// topo.hip — xGMI link topology discovery.
//
// MI355X-native replacement of the reference Level-Zero-Sysman fabric-port
// enumerator (reference p2p/topology.cpp:28-107). On an 8x MI355X node every
// GPU has 7 point-to-point xGMI links (~153 GB/s each, fully connected), so
// the reference's "connectivity planes" degenerate to one plane; the useful
// output here is the per-pair link matrix (type / hops / weight / min-max
// bandwidth) that drives rank->GPU placement (gpu_mapping.sh, the analog of
// reference p2p/tile_mapping.sh).
//
// Sources: HIP runtime (hipDeviceCanAccessPeer, hipExtGetLinkTypeAndHopCount)
// always; rocm_smi (link weight + min/max bandwidth) opportunistically.

#include "include/hpk.h"

#include <rocm_smi/rocm_smi.h>

#include <functional>

namespace hpk {

int device_count() {
  int n = 0;
  check_hip(hipGetDeviceCount(&n), "hipGetDeviceCount");
  return n;
}

std::vector<std::vector<LinkInfo>> link_matrix() {
  int n = device_count();
  std::vector<std::vector<LinkInfo>> m(n, std::vector<LinkInfo>(n));

  for (int i = 0; i < n; ++i) {
    for (int j = 0; j < n; ++j) {
      if (i == j) continue;
      LinkInfo& li = m[i][j];
      int can = 0;
      if (hipDeviceCanAccessPeer(&can, i, j) == hipSuccess)
        li.p2p_accessible = can;
      uint32_t type = 0, hops = 0;
      if (hipExtGetLinkTypeAndHopCount(i, j, &type, &hops) == hipSuccess) {
        li.link_type = (int)type;
        li.hops = (int)hops;
      }
    }
  }

  // rocm_smi enrichment — best-effort: device indices of rsmi match the HIP
  // enumeration order on this image (both KFD ordinal order); failures leave
  // the -1 defaults in place.
  if (rsmi_init(0) == RSMI_STATUS_SUCCESS) {
    uint32_t nsmi = 0;
    if (rsmi_num_monitor_devices(&nsmi) == RSMI_STATUS_SUCCESS) {
      int lim = (int)nsmi < n ? (int)nsmi : n;
      for (int i = 0; i < lim; ++i) {
        for (int j = 0; j < lim; ++j) {
          if (i == j) continue;
          uint64_t hops = 0, weight = 0;
          RSMI_IO_LINK_TYPE t = RSMI_IOLINK_TYPE_UNDEFINED;
          if (rsmi_topo_get_link_type(i, j, &hops, &t) == RSMI_STATUS_SUCCESS) {
            if (m[i][j].hops < 0) m[i][j].hops = (int)hops;
          }
          if (rsmi_topo_get_link_weight(i, j, &weight) == RSMI_STATUS_SUCCESS)
            m[i][j].weight = (long)weight;
          uint64_t bw_min = 0, bw_max = 0;
          if (rsmi_minmax_bandwidth_get(i, j, &bw_min, &bw_max) ==
              RSMI_STATUS_SUCCESS) {
            m[i][j].min_bw_mbps = (long)bw_min;
            m[i][j].max_bw_mbps = (long)bw_max;
          }
        }
      }
    }
    rsmi_shut_down();
  }
  return m;
}

std::vector<PartitionInfo> partition_info() {
  // XCD/memory partition modes (SPX/DPX/..., NPS1/NPS4) — the MI355X
  // replacement for the reference's tile-fission awareness
  // (devices.hpp:30-34): a CPX-partitioned GPU shows up as several HIP
  // devices, so placement policies need to know.
  std::vector<PartitionInfo> out;
  if (rsmi_init(0) != RSMI_STATUS_SUCCESS) return out;
  uint32_t n = 0;
  if (rsmi_num_monitor_devices(&n) == RSMI_STATUS_SUCCESS) {
    for (uint32_t i = 0; i < n; ++i) {
      PartitionInfo pi;
      char buf[64] = {0};
      if (rsmi_dev_compute_partition_get(i, buf, sizeof(buf)) ==
          RSMI_STATUS_SUCCESS)
        pi.compute = buf;
      char mbuf[64] = {0};
      if (rsmi_dev_memory_partition_get(i, mbuf, sizeof(mbuf)) ==
          RSMI_STATUS_SUCCESS)
        pi.memory = mbuf;
      out.push_back(pi);
    }
  }
  rsmi_shut_down();
  return out;
}

std::vector<std::vector<int>> p2p_planes() {
  int n = device_count();
  auto m = link_matrix();
  // Union-find over direct-P2P reachability.
  std::vector<int> parent(n);
  for (int i = 0; i < n; ++i) parent[i] = i;
  std::function<int(int)> find = [&](int x) {
    while (parent[x] != x) x = parent[x] = parent[parent[x]];
    return x;
  };
  for (int i = 0; i < n; ++i)
    for (int j = 0; j < n; ++j)
      if (i != j && m[i][j].p2p_accessible) {
        int a = find(i), b = find(j);
        if (a != b) parent[a] = b;
      }
  std::vector<std::vector<int>> planes;
  std::vector<int> root_to_plane(n, -1);
  for (int i = 0; i < n; ++i) {
    int r = find(i);
    if (root_to_plane[r] < 0) {
      root_to_plane[r] = (int)planes.size();
      planes.push_back({});
    }
    planes[root_to_plane[r]].push_back(i);
  }
  return planes;
}

} // namespace hpk
