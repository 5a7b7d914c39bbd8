"""GPU tests: topology, IPC across processes, flagship step, standalone
binaries (the CTest analog of the reference's mpirun -np 4 registrations)."""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(900)]

REPO = Path(__file__).resolve().parent.parent


# ---------------------------------------------------------------------------
# topology
# ---------------------------------------------------------------------------

def test_topology_native():
    from hpc_patterns_amd.parallel import topology

    m = topology.link_matrix()
    n = len(m)
    assert n >= 1
    planes = topology.planes_from_matrix(m)
    assert sum(len(p) for p in planes) == n
    order = topology.topology_order(n)
    assert sorted(order) == list(range(n))


# ---------------------------------------------------------------------------
# HIP-IPC one-sided put across processes (same or different GPUs)
# ---------------------------------------------------------------------------

def _ipc_child(handle_bytes, n, conn):
    import torch  # noqa

    from hpc_patterns_amd._native import native

    hpk = native()
    hpk.set_device(0)
    ptr = hpk.ipc_open_handle(handle_bytes)
    src = hpk.hip_malloc(n * 4)
    hpk.fill_f32(src, 42.0, n, 0)
    hpk.memcpy_async(ptr, src, n * 4, 0)
    hpk.stream_synchronize(0)
    hpk.ipc_close_handle(ptr)
    hpk.hip_free(src)
    conn.send("done")


def test_ipc_one_sided_put():
    import torch.multiprocessing as mp

    from hpc_patterns_amd import ops
    from hpc_patterns_amd._native import native

    hpk = native()
    torch.cuda.set_device(0)
    n = 1 << 20
    win = torch.zeros(n, dtype=torch.float32, device="cuda")
    torch.cuda.synchronize()
    handle = hpk.ipc_get_handle(win.data_ptr())

    ctx = mp.get_context("spawn")
    parent, child = ctx.Pipe()
    p = ctx.Process(target=_ipc_child, args=(handle, n, child))
    p.start()
    assert parent.recv() == "done"
    p.join(timeout=60)
    torch.cuda.synchronize()
    assert torch.all(win == 42.0), win[:5]


# ---------------------------------------------------------------------------
# flagship pattern step (1 GPU)
# ---------------------------------------------------------------------------

def test_flagship_smoke_step():
    from hpc_patterns_amd.models import SMOKE_CONFIG, FlagshipPatternStep

    step = FlagshipPatternStep(device=torch.device("cuda", 0), rank=0,
                               world_size=1, config=dict(SMOKE_CONFIG))
    for _ in range(3):
        step.step()
    torch.cuda.synchronize()
    ov = step.measure_overlap(reps=2)
    assert ov["speedup"] > 0 and ov["theoretical_speedup"] >= 1.0
    assert step.bytes_per_step_per_rank() > 0


def test_flagship_calibration():
    from hpc_patterns_amd.models import FlagshipPatternStep
    from hpc_patterns_amd.models.flagship import SMOKE_CONFIG

    cfg = dict(SMOKE_CONFIG)
    cfg["tripcount"] = -1  # force calibration
    step = FlagshipPatternStep(device=torch.device("cuda", 0), rank=0,
                               world_size=1, config=cfg)
    assert cfg["tripcount"] >= 1


# ---------------------------------------------------------------------------
# standalone binaries
# ---------------------------------------------------------------------------

def _run(cmd, **kw):
    return subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                          cwd=REPO, **kw)


def test_bin_topology():
    res = _run([str(REPO / "bin/hpk_topology")])
    assert res.returncode == 0, res.stderr
    assert "HIP device(s)" in res.stdout
    assert "plane" in res.stdout
    res1 = _run([str(REPO / "bin/hpk_topology"), "0"])
    assert res1.returncode == 0 and res1.stdout.strip().isdigit()


def test_bin_conc_balanced_pair():
    # 256 MB copies + a real compute grid: at the default 64 MB / 1-thread-C
    # scale the commands finish in tens of µs and overlap is launch-latency
    # noise (observed flaky) — the criterion needs ms-scale commands.
    # C || H2D on a named SDMA engine: kernel and DMA are independent
    # hardware units, so the reference criterion must pass on every box
    # (kernel||kernel co-scheduling is box-dependent — see
    # test_gpu_conc.test_overlap_compute_copy)
    res = _run([str(REPO / "bin/hpk_conc"), "in_order", "--repetitions", "10",
                "--globalsize_HD", str(1 << 26),
                "--globalsize_C", str(1 << 16), "--copy_engine", "sdma",
                "--commands", "C", "H2D"])
    assert "## in_order | C HD |" in res.stdout, res.stdout + res.stderr
    # the balanced C||H2D overlap must pass the reference criterion
    assert "SUCCESS" in res.stdout, res.stdout


def test_bin_conc_multi_lists_and_csv(tmp_path):
    csv = tmp_path / "out.csv"
    res = _run([str(REPO / "bin/hpk_conc"), "graph", "--repetitions", "3",
                "--globalsize_default_memory", str(1 << 24),
                "--csv", str(csv),
                "--commands", "D2D", "D2D", "--commands", "C", "H2D"])
    assert res.stdout.count("##") == 2, res.stdout + res.stderr
    lines = csv.read_text().strip().splitlines()
    assert lines[0].startswith("mode,commands")
    assert len(lines) == 3


def test_bin_conc_help():
    res = _run([str(REPO / "bin/hpk_conc")])
    assert res.returncode == 1 and "Usage" in res.stdout


def test_bin_allreduce_single_rank():
    res = _run([str(REPO / "bin/hpk_allreduce"), "-p", "20", "-i", "2",
                "-n", "1", "--algo", "rccl"])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "Passed rank 0" in res.stdout


def test_bin_allreduce_ring_single_rank():
    res = _run([str(REPO / "bin/hpk_allreduce"), "-p", "18", "-i", "1",
                "-n", "1", "--algo", "ring"])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "Passed rank 0" in res.stdout


def test_bin_p2p_ipc_engine():
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    res = _run([str(REPO / "bin/hpk_p2p"), "--engine", "ipc",
                "--floats", str(1 << 22)], env=env)
    assert res.returncode == 0, res.stdout + res.stderr
    # full RMA protocol (r2): both put phases through fence epochs, verified
    # in-binary (exit!=0 on checksum failure)
    assert "ipc Unidirectional Bandwidth" in res.stdout
    assert "ipc Bidirectional Bandwidth" in res.stdout
    assert "fence epochs" in res.stdout


def test_bin_p2p_ipc_engine_four_ranks():
    """All-pairs RMA oversubscribed to 4 ranks (2 concurrent fence-epoch
    pairs) — the reference ran all pairs concurrently (peer2pear.cpp:126-146);
    this exercises that shape even on a 1-GPU lease."""
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    res = _run([str(REPO / "bin/hpk_p2p"), "--engine", "ipc",
                "--floats", str(1 << 22), "--ranks", "4"], env=env)
    assert res.returncode == 0, res.stdout + res.stderr
    assert "pairs=2" in res.stdout


def test_bin_allreduce_ipc_transport_oversubscribed():
    """The native ring's (size-1)-step exchange executes with 2 and 4 ranks
    on one GPU: step count is printed by the binary."""
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    for n, steps in ((2, 1), (4, 3)):
        res = _run([str(REPO / "bin/hpk_allreduce"), "-p", "18", "-i", "2",
                    "--transport", "ipc", "-n", str(n)], env=env)
        assert res.returncode == 0, res.stdout + res.stderr
        for r in range(n):
            assert f"Passed rank {r}" in res.stdout
        assert f"steps={steps}" in res.stdout


def test_bin_p2p_peer_engine():
    # on a 1-GPU box this runs the same-device plumbing fallback
    res = _run([str(REPO / "bin/hpk_p2p"), "--engine", "peer",
                "--floats", str(1 << 22)])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "Unidirectional Bandwidth" in res.stdout
    assert "Bidirectional Bandwidth" in res.stdout


# ---------------------------------------------------------------------------
# bench.py smoke (the driver contract, 1 GPU)
# ---------------------------------------------------------------------------

def test_bench_smoke_json():
    import json

    res = _run([sys.executable, str(REPO / "bench.py"), "--steps", "3",
                "--warmup", "1", "--smoke"])
    assert res.returncode == 0, res.stdout + res.stderr
    line = [l for l in res.stdout.splitlines() if l.startswith("{")][-1]
    data = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in data, key
    assert data["n_gpus"] == 1 and data["value"] > 0


def test_bin_interop():
    res = _run([str(REPO / "bin/hpk_interop")])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "PASSED" in res.stdout


def test_bin_allreduce_int_dtype():
    res = _run([str(REPO / "bin/hpk_allreduce"), "-p", "18", "-i", "1",
                "-n", "1", "--algo", "rccl", "-t", "int"])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "Passed rank 0" in res.stdout and "dtype=int" in res.stdout


def test_bin_membench_quick():
    res = _run([str(REPO / "bin/hpk_membench"), "--quick"])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "TFLOP/s bf16" in res.stdout and "GB/s payload" in res.stdout


def test_bin_conc_min_bandwidth_floor():
    # floor respected end to end: an impossible floor must FAIL (exit 1),
    # a trivial floor must not trip the bandwidth check
    res = _run([str(REPO / "bin/hpk_conc"), "in_order", "--repetitions", "3",
                "--globalsize_default_memory", str(1 << 24),
                "--min_bandwidth", "999999", "--commands", "D2D", "D2D"])
    assert res.returncode == 1 and "Minimum Bandwidth not reached" in res.stdout
    res2 = _run([str(REPO / "bin/hpk_conc"), "in_order", "--repetitions", "3",
                 "--globalsize_default_memory", str(1 << 24),
                 "--min_bandwidth", "0.001", "--commands", "C", "D2D"])
    assert "Minimum Bandwidth not reached" not in res2.stdout


def test_bin_mpi_p2p_pipelined_staged():
    """Chunked staging pipeline for hipMalloc buffers over non-GPU-aware
    MPI: D2H || MPI || H2D overlap (checksummed in-binary; must beat or at
    least match launching — correctness asserted here, the bandwidth gain
    is recorded in profiles/)."""
    mpirun = "/opt/conda/bin/mpirun"
    if not os.path.exists(mpirun):
        import pytest

        pytest.skip("no MPICH in image")
    res = _run([mpirun, "-np", "2", str(REPO / "bin/hpk_mpi_p2p"),
                "--engine", "isend", "-D", "--pipeline", "8",
                "--floats", str(1 << 22)])
    assert res.returncode == 0, res.stdout + res.stderr
    assert "mpi-isend-pipe Unidirectional Bandwidth" in res.stdout
    assert "mpi-isend-pipe Bidirectional Bandwidth" in res.stdout


def test_pipelined_ring_stream_choreography_loopback(monkeypatch):
    """The two-stream pipelined ring (comm ops + per-chunk accumulate on a
    dedicated compute stream) is otherwise unreachable before a multi-GPU
    run: NCCL refuses two ranks on one device. This swaps the transport
    for an in-process loopback (recv := send on the current stream) while
    keeping every stream/wait/accumulate line of the real path; wrong
    cross-stream ordering would corrupt the deterministic result."""
    from types import SimpleNamespace

    from hpc_patterns_amd.parallel import ring

    class _FakeReq:
        def wait(self):
            pass

    def fake_batch(ops):
        sends = [op.tensor for op in ops if op.op == "isend"]
        recvs = [op.tensor for op in ops if op.op == "irecv"]
        for s, r in zip(sends, recvs):
            r.copy_(s)  # ordered on the calling stream, like the comm op
        return [_FakeReq()]

    fake_dist = SimpleNamespace(
        isend="isend", irecv="irecv",
        P2POp=lambda op, tensor, peer, group=None: SimpleNamespace(
            op=op, tensor=tensor),
        batch_isend_irecv=fake_batch,
    )
    # pretend world=2 with self as both neighbours; no real process group
    monkeypatch.setattr(ring, "_ring_neighbours", lambda group=None: (0, 2, 0, 0))
    monkeypatch.setattr(ring, "dist", fake_dist)

    t = torch.arange(1 << 20, dtype=torch.float32, device="cuda")
    expect = t * 2  # one loopback exchange step: chunks += their own send
    ring.ring_allreduce_pipelined(t.clone(), n_chunks=8)  # smoke alias path
    out = ring.ring_allreduce_pipelined(t, n_chunks=8)
    torch.cuda.synchronize()
    assert torch.equal(out, expect)
