// Vectorized load/store helpers: bf16 as 16-byte lanes (G13 in the CDNA guide:
// hipcc does not auto-vectorize bf16 scalar loads; 8-16B/lane is the sweet spot).
#pragma once
#include "common.h"

namespace dla {

template <typename T, int V>
struct alignas(sizeof(T) * V) Vec {
  T v[V];
};

// preferred elements-per-lane for 16B accesses
template <typename T> struct VecWidth { static constexpr int value = 16 / sizeof(T); };

template <typename T, int V>
__device__ __forceinline__ Vec<T, V> vload(const T* p) {
  return *reinterpret_cast<const Vec<T, V>*>(p);
}

template <typename T, int V>
__device__ __forceinline__ void vstore(T* p, const Vec<T, V>& x) {
  *reinterpret_cast<Vec<T, V>*>(p) = x;
}

}  // namespace dla
