// grapehip — device-side graph view (POD) for CDNA4 kernels.
//
// MI355X-first design (vs reference grape/cuda/fragment/device_fragment.h):
// kernels address vertices by GLOBAL 32-bit vid everywhere. Per-vertex state
// arrays (depth/dist/rank/label) are replicated at global size in each
// rank's HBM3E — at datagen-9_0-fb scale a u32 array over all 405M vertices
// is 1.6 GB of 288 GB, which buys us: no device hashmap for outer vertices
// (reference needs cuda_hashmap), owner lookup is a <=8-entry segment scan,
// halo dedup falls out of the atomicMin on the state array itself, and RCCL
// slices (reduce-scatter / allgather over owned ranges) are contiguous.
// The CSR covers only OWNED vertices (row r = global vid v_begin + r).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace grapehip {

struct DevGraphView {
  uint32_t nv_global;
  uint32_t v_begin, v_end;  // owned global range
  uint32_t slice;           // uniform segment width: owner(v) = v / slice
  int rank, world;
  // out-CSR over owned rows
  const uint64_t* __restrict__ oe_off;  // [owned+1]
  const uint32_t* __restrict__ oe_dst;  // global vids
  const float* __restrict__ oe_w;       // nullptr if unweighted
  // optional in-CSR over owned rows (directed graphs)
  const uint64_t* __restrict__ ie_off;
  const uint32_t* __restrict__ ie_dst;
  const float* __restrict__ ie_w;
  // ownership segments, [world+1]
  const uint32_t* __restrict__ seg;

  __device__ __forceinline__ uint32_t owned() const { return v_end - v_begin; }
  __device__ __forceinline__ bool is_owned(uint32_t v) const {
    return v >= v_begin && v < v_end;
  }
  __device__ __forceinline__ uint32_t row(uint32_t v) const {
    return v - v_begin;
  }
  __device__ __forceinline__ int owner(uint32_t v) const {
    int f = v / slice;
    return f < world ? f : world - 1;
  }
};

}  // namespace grapehip
