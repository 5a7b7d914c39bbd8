// mpi_p2p_main.cpp — hpk_mpi_p2p: pairwise bandwidth over REAL MPI
// (MPICH), the direct twin of the reference P2P bench
// (reference p2p/peer2pear.cpp:104-156), with BOTH reference engines:
//   --engine isend   two-sided MPI_Isend/MPI_Irecv + Waitall
//                    (peer2pear.cpp:19-66, the default build)
//   --engine win     one-sided MPI_Win_create / Win_fence / MPI_Put
//                    (peer2pear.cpp:68-102, the -DUSE_WIN build)
//
// Protocol parity, line for line with the reference design (re-implemented):
// ranks pair even<->odd (:126-131), phase 1 unidirectional, phase 2
// bidirectional, 10 iterations, the GLOBAL interval is MPI_Reduce MIN of
// start / MAX of end epoch-nanosecond timestamps (:49-51), aggregate GB/s =
// bytes x pairs / min interval (:137-139,152-155), payload is a
// host-shuffled iota whose received sum must be exact (:8-17,56-63 — here
// an order-independent exact double sum instead of sort+sum).
//
// MPICH in this image (3.3.2 ch3:nemesis) is not GPU-aware, so the
// allocator flag chooses how close to the GPU the buffers live:
//   -M malloc (pure host), -H hipHostMalloc (pinned, device-visible
//   zero-copy — MPI consumes the same pointer the GPU kernels see),
//   -D hipMalloc staged through a pinned bounce inside the timed region
//   (the non-GPU-aware staging cost, measured honestly). The xGMI fast
//   paths live in hpk_p2p (peer/ipc/rccl engines).

#include "../hpc_patterns_amd/native/include/hpk.h"
#include "../hpc_patterns_amd/native/include/mpi_datatype.h"

#include <mpi.h>

#include <algorithm>
#include <chrono>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <numeric>
#include <random>
#include <string>
#include <vector>

namespace {

constexpr size_t kDefaultN = 47185920; // reference peer2pear.cpp:115
constexpr int kIters = 10;

void check_mpi(int rc, const char* what) {
  if (rc != MPI_SUCCESS) {
    std::fprintf(stderr, "MPI error in %s: %d\n", what, rc);
    MPI_Abort(MPI_COMM_WORLD, 1);
  }
}

unsigned long now_ns() {
  return (unsigned long)std::chrono::duration_cast<std::chrono::nanoseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

struct Buffers {
  char alloc = 'H';
  size_t n = 0;
  float* buf = nullptr;    // the payload buffer MPI sees (or stages into)
  float* dev = nullptr;    // device twin for -D
  double sum = 0.0;        // expected checksum of the send payload

  void allocate(size_t n_, char kind) {
    n = n_;
    alloc = kind;
    size_t bytes = n * sizeof(float);
    switch (kind) {
      case 'M': buf = (float*)std::malloc(bytes); break;
      case 'H':
        hpk::check_hip(hipHostMalloc((void**)&buf, bytes,
                                     hipHostMallocDefault), "pinned");
        break;
      case 'D':
        hpk::check_hip(hipMalloc((void**)&dev, bytes), "device");
        hpk::check_hip(hipHostMalloc((void**)&buf, bytes,
                                     hipHostMallocDefault), "stage");
        break;
      default: std::abort();
    }
    if (!buf) { std::fprintf(stderr, "alloc failed\n"); std::exit(1); }
  }

  // reference fill_randomly (peer2pear.cpp:8-17): shuffled iota + checksum
  void fill_payload(unsigned seed) {
    std::vector<float> v(n);
    std::iota(v.begin(), v.end(), 0.f);
    std::minstd_rand g(seed);
    std::shuffle(v.begin(), v.end(), g);
    std::memcpy(buf, v.data(), n * sizeof(float));
    if (alloc == 'D')
      hpk::check_hip(hipMemcpy(dev, buf, n * sizeof(float),
                               hipMemcpyHostToDevice), "payload h2d");
    sum = 0.0;
    for (size_t i = 0; i < n; ++i) sum += (double)v[i];
  }

  // staged mode: the timed region moves device->stage before the send and
  // stage->device after the recv (the honest cost of non-GPU-aware MPI)
  void pre_send() {
    if (alloc == 'D')
      hpk::check_hip(hipMemcpy(buf, dev, n * sizeof(float),
                               hipMemcpyDeviceToHost), "stage d2h");
  }
  void post_recv() {
    if (alloc == 'D')
      hpk::check_hip(hipMemcpy(dev, buf, n * sizeof(float),
                               hipMemcpyHostToDevice), "stage h2d");
  }

  double host_sum() const {
    double s = 0.0;
    for (size_t i = 0; i < n; ++i) s += (double)buf[i];
    return s;
  }
};

// One timed transfer phase (reference datatransfer/datatransfer_win,
// peer2pear.cpp:19-102): returns the min global interval in seconds.
double transfer_phase(const std::string& engine, Buffers& send, Buffers& recv,
                      int peer, bool sender, bool receiver, MPI_Win win) {
  double best = 1e30;
  for (int it = 0; it < kIters; ++it) {
    check_mpi(MPI_Barrier(MPI_COMM_WORLD), "barrier");
    unsigned long t0 = now_ns();
    if (engine == "win") {
      check_mpi(MPI_Win_fence(0, win), "fence open");
      if (sender && peer >= 0) {
        send.pre_send();
        check_mpi(MPI_Put(send.buf, (int)send.n, MPI_FLOAT, peer, 0,
                          (int)send.n, MPI_FLOAT, win), "put");
      }
      check_mpi(MPI_Win_fence(0, win), "fence close");
      if (receiver && peer >= 0) recv.post_recv();
    } else {
      MPI_Request reqs[2];
      int nreq = 0;
      if (peer >= 0) {
        if (sender) {
          send.pre_send();
          check_mpi(MPI_Isend(send.buf, (int)send.n, MPI_FLOAT, peer, 0,
                              MPI_COMM_WORLD, &reqs[nreq++]), "isend");
        }
        if (receiver)
          check_mpi(MPI_Irecv(recv.buf, (int)recv.n, MPI_FLOAT, peer, 0,
                              MPI_COMM_WORLD, &reqs[nreq++]), "irecv");
        check_mpi(MPI_Waitall(nreq, reqs, MPI_STATUSES_IGNORE), "waitall");
        if (receiver) recv.post_recv();
      }
    }
    unsigned long t1 = now_ns();
    // clock union (reference peer2pear.cpp:49-51): MIN(start), MAX(end)
    unsigned long g0 = 0, g1 = 0;
    check_mpi(MPI_Reduce(&t0, &g0, 1, MPI_UNSIGNED_LONG, MPI_MIN, 0,
                         MPI_COMM_WORLD), "min start");
    check_mpi(MPI_Reduce(&t1, &g1, 1, MPI_UNSIGNED_LONG, MPI_MAX, 0,
                         MPI_COMM_WORLD), "max end");
    double dt = (double)(g1 - g0) / 1e9;
    check_mpi(MPI_Bcast(&dt, 1, MPI_DOUBLE, 0, MPI_COMM_WORLD), "bcast");
    best = std::min(best, dt);
  }
  return best;
}

// Chunked staging pipeline for the -D (hipMalloc) path with the isend
// engine: the naive staged transfer serializes [full D2H] -> [full MPI] ->
// [full H2D]; here chunk c's MPI transfer overlaps chunk c+1's D2H and the
// H2D un-staging of already-landed chunks — the classic 3-stage pipeline a
// non-GPU-aware-MPI application uses to approach min(PCIe, MPI) instead of
// their serial sum. Structure: post all Irecvs; enqueue all D2H chunk
// copies with an event each; per chunk [event sync -> Isend]; drain recvs
// with Waitany, un-staging each landed chunk asynchronously.
double transfer_phase_pipelined(Buffers& send, Buffers& recv, int peer,
                                bool sender, bool receiver, int nchunks) {
  size_t n = send.n;
  size_t chunk = (n + nchunks - 1) / nchunks;
  hipStream_t s_out = nullptr, s_in = nullptr;
  hpk::check_hip(hipStreamCreateWithFlags(&s_out, hipStreamNonBlocking), "s_out");
  hpk::check_hip(hipStreamCreateWithFlags(&s_in, hipStreamNonBlocking), "s_in");
  std::vector<hipEvent_t> ev((size_t)nchunks);
  for (auto& e : ev)
    hpk::check_hip(hipEventCreateWithFlags(&e, hipEventDisableTiming), "ev");

  auto len_of = [&](int c) {
    size_t off = (size_t)c * chunk;
    return off >= n ? (size_t)0 : std::min(chunk, n - off);
  };

  double best = 1e30;
  for (int it = 0; it < kIters; ++it) {
    check_mpi(MPI_Barrier(MPI_COMM_WORLD), "barrier");
    unsigned long t0 = now_ns();
    std::vector<MPI_Request> rreq((size_t)nchunks, MPI_REQUEST_NULL);
    std::vector<MPI_Request> sreq((size_t)nchunks, MPI_REQUEST_NULL);
    if (peer >= 0 && receiver)
      for (int c = 0; c < nchunks; ++c) {
        size_t len = len_of(c);
        if (!len) continue;
        check_mpi(MPI_Irecv(recv.buf + (size_t)c * chunk, (int)len, MPI_FLOAT,
                            peer, 100 + c, MPI_COMM_WORLD, &rreq[c]),
                  "irecv chunk");
      }
    if (peer >= 0 && sender) {
      for (int c = 0; c < nchunks; ++c) {
        size_t len = len_of(c);
        if (!len) continue;
        hpk::check_hip(hipMemcpyAsync(send.buf + (size_t)c * chunk,
                                      send.dev + (size_t)c * chunk,
                                      len * sizeof(float),
                                      hipMemcpyDeviceToHost, s_out),
                       "pipe d2h");
        hpk::check_hip(hipEventRecord(ev[c], s_out), "pipe ev");
      }
      for (int c = 0; c < nchunks; ++c) {
        size_t len = len_of(c);
        if (!len) continue;
        hpk::check_hip(hipEventSynchronize(ev[c]), "pipe ev sync");
        check_mpi(MPI_Isend(send.buf + (size_t)c * chunk, (int)len, MPI_FLOAT,
                            peer, 100 + c, MPI_COMM_WORLD, &sreq[c]),
                  "isend chunk");
      }
    }
    if (peer >= 0 && receiver) {
      int remaining = 0;
      for (int c = 0; c < nchunks; ++c)
        if (rreq[c] != MPI_REQUEST_NULL) ++remaining;
      while (remaining-- > 0) {
        int idx = MPI_UNDEFINED;
        check_mpi(MPI_Waitany(nchunks, rreq.data(), &idx,
                              MPI_STATUS_IGNORE), "waitany");
        if (idx == MPI_UNDEFINED) break;
        size_t len = len_of(idx);
        hpk::check_hip(hipMemcpyAsync(recv.dev + (size_t)idx * chunk,
                                      recv.buf + (size_t)idx * chunk,
                                      len * sizeof(float),
                                      hipMemcpyHostToDevice, s_in),
                       "pipe h2d");
      }
      hpk::check_hip(hipStreamSynchronize(s_in), "pipe h2d sync");
    }
    if (peer >= 0 && sender)
      check_mpi(MPI_Waitall(nchunks, sreq.data(), MPI_STATUSES_IGNORE),
                "pipe waitall");
    unsigned long t1 = now_ns();
    unsigned long g0 = 0, g1 = 0;
    check_mpi(MPI_Reduce(&t0, &g0, 1, MPI_UNSIGNED_LONG, MPI_MIN, 0,
                         MPI_COMM_WORLD), "min start");
    check_mpi(MPI_Reduce(&t1, &g1, 1, MPI_UNSIGNED_LONG, MPI_MAX, 0,
                         MPI_COMM_WORLD), "max end");
    double dt = (double)(g1 - g0) / 1e9;
    check_mpi(MPI_Bcast(&dt, 1, MPI_DOUBLE, 0, MPI_COMM_WORLD), "bcast");
    best = std::min(best, dt);
  }
  for (auto& e : ev) (void)hipEventDestroy(e);
  (void)hipStreamDestroy(s_out);
  (void)hipStreamDestroy(s_in);
  return best;
}

void verify(const Buffers& recv, double expected, const char* what,
            int rank) {
  double got = recv.alloc == 'D'
                   ? hpk::reduce_sum_f32(recv.dev, recv.n, nullptr)
                   : recv.host_sum();
  if (got != expected) {
    std::fprintf(stderr, "CHECKSUM FAILURE (%s, rank %d): %.1f != %.1f\n",
                 what, rank, got, expected);
    MPI_Abort(MPI_COMM_WORLD, 2);
  }
}

} // namespace

int main(int argc, char* argv[]) {
  check_mpi(MPI_Init(&argc, &argv), "init");
  int rank = 0, size = 1;
  MPI_Comm_rank(MPI_COMM_WORLD, &rank);
  MPI_Comm_size(MPI_COMM_WORLD, &size);

  std::string engine = "isend";
  char alloc = 'M';
  size_t n = kDefaultN;
  int pipeline = 0;
  for (int i = 1; i < argc; ++i) {
    std::string s = argv[i];
    auto next = [&]() -> const char* {
      if (++i >= argc) { std::fprintf(stderr, "missing value\n"); std::exit(1); }
      return argv[i];
    };
    if (s == "--engine") engine = next();
    else if (s == "-n" || s == "--floats") n = std::strtoull(next(), nullptr, 10);
    else if (s == "-M") alloc = 'M';
    else if (s == "-H") alloc = 'H';
    else if (s == "-D") alloc = 'D';
    else if (s == "--pipeline") pipeline = std::atoi(next());
    else {
      if (rank == 0)
        std::printf("Usage: mpirun -np N %s [--engine isend|win] "
                    "[--floats N] [-M|-H|-D] [--pipeline K]\n", argv[0]);
      MPI_Finalize();
      return (s == "-h" || s == "--help") ? 0 : 1;
    }
  }
  if (pipeline > 0 && (engine != "isend" || alloc != 'D')) {
    if (rank == 0)
      std::fprintf(stderr, "--pipeline needs --engine isend -D (it chunks "
                   "the device<->pinned staging)\n");
    MPI_Finalize();
    return 1;
  }
  if (engine != "isend" && engine != "win") {
    if (rank == 0) std::fprintf(stderr, "unknown engine '%s'\n", engine.c_str());
    MPI_Finalize();
    return 1;
  }
  if (n > (size_t)INT32_MAX) {
    if (rank == 0)
      std::fprintf(stderr, "--floats must be <= 2^31-1 (MPI int count)\n");
    MPI_Finalize();
    return 1;
  }

  int ndev = 0;
  (void)hipGetDeviceCount(&ndev);
  if (alloc != 'M') {
    if (ndev == 0) {
      if (rank == 0)
        std::fprintf(stderr, "no HIP devices: -H/-D need a GPU (-M for "
                     "host buffers)\n");
      MPI_Finalize();
      return 1;
    }
    hpk::check_hip(hipSetDevice(rank % ndev), "hipSetDevice");
  }

  // pairing (reference peer2pear.cpp:126-131): even i <-> i+1
  int peer = (rank % 2 == 0) ? rank + 1 : rank - 1;
  if (peer >= size) peer = -1;

  Buffers send, recv;
  send.allocate(n, alloc);
  recv.allocate(n, alloc);
  send.fill_payload((unsigned)rank);

  // exchange expected checksums
  double peer_sum = 0.0;
  if (peer >= 0)
    check_mpi(MPI_Sendrecv(&send.sum, 1, MPI_DOUBLE, peer, 9, &peer_sum, 1,
                           MPI_DOUBLE, peer, 9, MPI_COMM_WORLD,
                           MPI_STATUS_IGNORE), "sum exchange");

  MPI_Win win = MPI_WIN_NULL;
  if (engine == "win")
    check_mpi(MPI_Win_create(recv.buf, n * sizeof(float), sizeof(float),
                             MPI_INFO_NULL, MPI_COMM_WORLD, &win),
              "win create"); // reference peer2pear.cpp:119-122

  size_t bytes = n * sizeof(float);
  int npairs = size / 2;
  std::string label = engine + (pipeline > 0 ? "-pipe" : "");
  for (int phase = 0; phase < 2; ++phase) {
    bool bidir = (phase == 1);
    bool sender = peer >= 0 && ((rank % 2 == 0) || bidir);
    bool receiver = peer >= 0 && ((rank % 2 == 1) || bidir);
    double best =
        pipeline > 0
            ? transfer_phase_pipelined(send, recv, peer, sender, receiver,
                                       pipeline)
            : transfer_phase(engine, send, recv, peer, sender, receiver, win);
    if (receiver) verify(recv, peer_sum, label.c_str(), rank);
    if (rank == 0) {
      double gb = (double)bytes * std::max(npairs, 1) * (bidir ? 2 : 1) / 1e9;
      std::printf("mpi-%s %s Bandwidth: %.2f GB/s (pairs=%d, %.1f MB each, "
                  "alloc=%c, min over %d iters)\n",
                  label.c_str(), bidir ? "Bidirectional" : "Unidirectional",
                  gb / best, npairs, bytes / 1e6, alloc, kIters);
    }
  }

  if (win != MPI_WIN_NULL) check_mpi(MPI_Win_free(&win), "win free");
  MPI_Finalize();
  return 0;
}
