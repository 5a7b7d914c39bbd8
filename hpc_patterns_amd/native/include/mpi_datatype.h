// mpi_datatype.h — C++ type -> MPI_Datatype trait.
//
// The MPI twin of rccl_datatype.h, covering the reference's trait header
// directly (reference aurora.mpich.miniapps/src/include/mpi_datatype.hpp:
// 18-53, mpi::get_datatype<T>() specializations). Same design decision as
// the RCCL trait: unmapped types fail to COMPILE — the reference's
// MPI_BYTE default under a SUM reduction would be a silent wrong answer.
//
// MPI_Datatype values are runtime handles in MPICH, so the accessor is a
// plain inline function rather than a constexpr member.
#pragma once

#include <mpi.h>

#include <cstdint>

namespace hpk {

template <typename T>
struct mpi_datatype; // unmapped type -> compile error

template <> struct mpi_datatype<float>    { static MPI_Datatype value() { return MPI_FLOAT; } };
template <> struct mpi_datatype<double>   { static MPI_Datatype value() { return MPI_DOUBLE; } };
template <> struct mpi_datatype<int8_t>   { static MPI_Datatype value() { return MPI_INT8_T; } };
template <> struct mpi_datatype<uint8_t>  { static MPI_Datatype value() { return MPI_UINT8_T; } };
template <> struct mpi_datatype<int16_t>  { static MPI_Datatype value() { return MPI_INT16_T; } };
template <> struct mpi_datatype<uint16_t> { static MPI_Datatype value() { return MPI_UINT16_T; } };
template <> struct mpi_datatype<int32_t>  { static MPI_Datatype value() { return MPI_INT32_T; } };
template <> struct mpi_datatype<uint32_t> { static MPI_Datatype value() { return MPI_UINT32_T; } };
template <> struct mpi_datatype<int64_t>  { static MPI_Datatype value() { return MPI_INT64_T; } };
template <> struct mpi_datatype<uint64_t> { static MPI_Datatype value() { return MPI_UINT64_T; } };
template <> struct mpi_datatype<long double> { static MPI_Datatype value() { return MPI_LONG_DOUBLE; } };

template <typename T>
inline MPI_Datatype get_mpi_datatype() {
  return mpi_datatype<T>::value();
}

} // namespace hpk
