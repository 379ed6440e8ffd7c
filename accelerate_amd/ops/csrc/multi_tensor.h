// Multi-tensor-apply plumbing for CDNA4 (gfx950).
//
// All optimizer-side ops (fused AdamW, global L2 norm, unscale+nonfinite
// check, scale) are memory-bound elementwise sweeps over hundreds of
// parameter tensors. Instead of one launch per tensor we pack device
// pointers + a chunk prefix-sum into small device buffers and launch ONE
// grid over all chunks: each workgroup binary-searches its tensor, then
// sweeps its chunk with float4 (16 B/lane) accesses — the HBM3E coalescing
// sweet spot per the CDNA4 guide (Guideline 13).
//
// Replaces the reference's delegation to torch foreach/fused CUDA kernels
// (reference: SURVEY.md §2.9 N5/N7/N8).
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

// elements per chunk: 256 threads × float4 × 16 iterations
constexpr int64_t kChunkSize = 16384;
constexpr int kBlockThreads = 256;

struct TensorListMeta {
  // device pointers (int64-encoded), laid out [list][tensor]
  const int64_t* addrs;   // [n_lists * n_tensors]
  const int64_t* numels;  // [n_tensors]
  const int32_t* chunk_prefix;  // [n_tensors + 1], cumulative chunk counts
  int32_t n_tensors;
  int32_t n_lists;
};

// find which tensor owns chunk `cid` via binary search on the prefix sums
__device__ __forceinline__ int find_tensor(const int32_t* prefix, int n, int cid) {
  int lo = 0, hi = n;  // invariant: prefix[lo] <= cid < prefix[hi]
  while (hi - lo > 1) {
    int mid = (lo + hi) >> 1;
    if (prefix[mid] <= cid) lo = mid; else hi = mid;
  }
  return lo;
}
