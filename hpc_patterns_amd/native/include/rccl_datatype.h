// rccl_datatype.h — C++ type -> ncclDataType_t trait.
//
// MI355X-native equivalent of the reference's MPI datatype trait header
// (reference aurora.mpich.miniapps/src/include/mpi_datatype.hpp:18-53:
// mpi::get_datatype<T>() specializations): the same compile-time mapping,
// retargeted at RCCL. Unmapped types fail to compile on purpose — the
// reference's MPI_BYTE fallback under a SUM reduction would be a silent
// wrong answer, not a fallback.
#pragma once

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <rccl/rccl.h>

#include <cstdint>

namespace hpk {

template <typename T>
struct rccl_datatype;

template <> struct rccl_datatype<float>         { static constexpr ncclDataType_t value = ncclFloat32; };
template <> struct rccl_datatype<double>        { static constexpr ncclDataType_t value = ncclFloat64; };
template <> struct rccl_datatype<int8_t>        { static constexpr ncclDataType_t value = ncclInt8; };
template <> struct rccl_datatype<uint8_t>       { static constexpr ncclDataType_t value = ncclUint8; };
template <> struct rccl_datatype<int32_t>       { static constexpr ncclDataType_t value = ncclInt32; };
template <> struct rccl_datatype<uint32_t>      { static constexpr ncclDataType_t value = ncclUint32; };
template <> struct rccl_datatype<int64_t>       { static constexpr ncclDataType_t value = ncclInt64; };
template <> struct rccl_datatype<uint64_t>      { static constexpr ncclDataType_t value = ncclUint64; };
template <> struct rccl_datatype<__half>        { static constexpr ncclDataType_t value = ncclFloat16; };
template <> struct rccl_datatype<__hip_bfloat16>{ static constexpr ncclDataType_t value = ncclBfloat16; };

template <typename T>
constexpr ncclDataType_t get_rccl_datatype() {
  return rccl_datatype<T>::value;
}

} // namespace hpk
