#!/bin/bash
# ci_gpu.sh — the GPU CI battery: correctness + performance floors as
# pass/fail (the reference treats performance as a test criterion,
# SURVEY.md §4). Exit code != 0 on any failure.
set -u
cd "$(dirname "$0")/.."
rc=0

echo "== pytest -m gpu"
python -m pytest tests -m gpu -q || rc=1

echo "== overlap criterion + bandwidth floors (conservative: ~60% of measured)"
# C||H2D on a named SDMA engine must pass the overlap criterion (kernel
# and DMA are independent units; kernel||kernel is box-dependent)
./bin/hpk_conc in_order --repetitions 10 \
    --globalsize_HD $((1 << 26)) --globalsize_C $((1 << 16)) \
    --copy_engine sdma --commands C H2D || rc=1
# D2D shader-copy bandwidth floor: >=1.5 TB/s payload (measured 3.0)
./bin/hpk_conc in_order --repetitions 10 \
    --globalsize_default_memory $((1 << 26)) \
    --min_bandwidth 1500 --copy_engine shader --commands D2D || rc=1
# H2D||D2H duplex on explicit SDMA engines: >=60 GB/s aggregate (measured 97)
./bin/hpk_conc in_order --copy_engine sdma --repetitions 10 \
    --min_bandwidth 60 --commands H2D D2H || rc=1

echo "== miniapps"
./bin/hpk_allreduce -p 22 -i 2 --algo ring || rc=1
./bin/hpk_allreduce -p 22 -i 2 --algo pipeline || rc=1
./bin/hpk_allreduce -p 22 -i 2 --algo rccl -t int || rc=1
# oversubscribed multi-rank exchange paths (run the REAL ring / all-pairs
# RMA protocol even on a 1-GPU lease)
./bin/hpk_allreduce -p 20 -i 2 --transport ipc -n 4 || rc=1
./bin/hpk_p2p --engine ipc --floats $((1 << 22)) || rc=1
./bin/hpk_p2p --engine ipc --floats $((1 << 22)) --ranks 4 || rc=1
./bin/hpk_interop || rc=1
./bin/hpk_membench --quick || rc=1
# graph_explicit mode + per-command device times in graph mode (r2).
# Command shape is kernel||DMA (C H2D): kernel||kernel graph branches hit
# the runtime scheduler serialization (findings.md #9) and would fail the
# 30% criterion through no fault of the engine; kernel||copy overlaps on
# every box (findings.md #8; measured 2.34x for C H2D D2H in r2_call1).
./bin/hpk_conc graph_explicit --repetitions 10 --enable_profiling \
    --globalsize_HD $((1 << 26)) --globalsize_C $((1 << 16)) \
    --commands C H2D || rc=1

echo "== real-MPI miniapps (MPICH, pinned-direct + staged-device modes)"
if [ -x /opt/conda/bin/mpirun ] && [ -x bin/hpk_mpi_allreduce ]; then
  /opt/conda/bin/mpirun -np 4 ./bin/hpk_mpi_allreduce -H -p 20 -i 2 || rc=1
  /opt/conda/bin/mpirun -np 4 ./bin/hpk_mpi_allreduce -D -a -p 20 -i 2 || rc=1
  /opt/conda/bin/mpirun -np 4 ./bin/hpk_mpi_p2p --engine isend -H \
      --floats $((1 << 22)) || rc=1
  /opt/conda/bin/mpirun -np 4 ./bin/hpk_mpi_p2p --engine win -H \
      --floats $((1 << 22)) || rc=1
else
  echo "# skipped (no mpirun or binary)"
fi

echo "== cmake + ctest harness (the reference's build contract)"
rm -rf build-cmake && mkdir -p build-cmake
(cd build-cmake && CXX=/opt/rocm/bin/hipcc cmake .. > cmake.log 2>&1 \
   && make -j16 > make.log 2>&1 && ctest --output-on-failure) || rc=1

echo "== bench smoke"
python bench.py --smoke --steps 3 --warmup 1 || rc=1

echo "== ci_gpu: $([ $rc -eq 0 ] && echo PASS || echo FAIL)"
exit $rc
