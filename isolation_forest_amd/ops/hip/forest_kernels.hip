// MI355X (gfx950, CDNA4) Isolation Forest kernels.
//
// Hand-written HIP implementations of the reference's JVM hot loops
// (design rationale and measured limits in
// docs/DESIGN.md):
//  * build_forest_kernel          — K1 (min/max scan + constant-feature
//    retry), K2 (stable row partition), K11 (per-node feature Fisher-Yates):
//    one 64-lane wavefront per tree, row-index permutation in LDS, ballot/
//    popcount stable partition, shfl-xor min/max reductions; emits per-node
//    depths for the scoring pack.
//  * build_extended_forest_kernel — K3-K5 (EIF Gaussian hyperplane draw,
//    L2 normalize, per-coordinate min/max + offset, dot-product partition).
//  * score_forest_v4              — K6/K8 standard scoring: integer-KEY
//    compares (rows staged in LDS as order-preserving u16/u32 keys,
//    per-node integer thresholds), depth folded into f32 leaf values,
//    self-looping leaves => FIXED-trip walks with zero bookkeeping,
//    2 rows x 4 staged trees = 8 chains/thread with batch-phased LDS reads
//    (staggered lgkmcnt waits). Bitwise == cpu_engine.path_lengths.
//  * score_extended_dense_v2      — K7 fast path (hyperplanes densified to
//    D in {8,16,32,64} columns; f32 rows): rows resident in f32 REGISTERS
//    across the whole tree loop, one tree's nodes+values+weights staged in
//    LDS (conflict-breaking D/4+1 float4 row stride), exec-masked finished
//    lanes, offset=-inf self-looping leaves. Tolerance contract (4-partial
//    fma dot), routing in ops/gpu_engine.score_extended_forest.
//  * score_extended_dense_v3      — K7 fast path for bf16 rows, D up to
//    128: weights staged as RNE-rounded PACKED bf16 (half the LDS bytes
//    of v2 — its measured bound), rows as packed-bf16 register pairs,
//    dot on v_dot2c_f32_bf16. Contract: PARITY.md "Numeric contracts".
//  * score_extended_sparse_v2     — K7 small-nnz path (templated NNZ<=5):
//    same fixed-trip structure, strict oracle j-order dot (bitwise).
//    extensionLevel-0 forests do NOT use it: they are packed onto
//    score_forest_v4 via exact key thresholds + mirrored subtrees
//    (ops/gpu_engine._eif0_packed_v4 — bitwise, zero per-visit overhead).
//  * score_extended_forest_kernel — K7 general strict-order fallback
//    (wide d / ragged hyperplane widths).
//  * score_forest_wide /
//    score_extended_wide          — full-width int4 node records for
//    forests past the packed 12-bit-feature/15-bit-node caps (foreign or
//    CPU-built deep models); global-memory walks, correctness path.
//  * bag_gather_kernel            — K9/K10 device side: gather sampled rows
//    into per-tree bags (indices drawn host-side by the same Philox).
//
// PRECISION CONTRACT: see core/cpu_engine.py. All randomness: philox.h;
// EIF gaussians: det_math.h. Wavefront size is 64 (CDNA4) and is hard-coded.
// This TU is torch-free: tools/native/ifa_score.cpp links it directly.

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "det_math.h"
#include "philox.h"

#define WAVE 64
#define LANE_MASK_LT(lane) ((1ull << (lane)) - 1ull)

namespace ifa {

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------

__device__ __forceinline__ float wave_reduce_min(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fminf(v, __shfl_xor(v, off, WAVE));
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// split point: float64 draw rounded to float32, bumped off the left edge
// (cpu_engine._split_value32)
__device__ __forceinline__ float split_value32(float fmin, float fmax, double u) {
  double s64 = (double)fmin + u * ((double)fmax - (double)fmin);
  float s32 = (float)s64;
  if (!(fmin < s32)) s32 = nextafterf(fmin, __builtin_inff());
  return s32;
}

struct SegEntry {
  uint16_t start, end;
  int16_t height;
  int16_t patch;  // node whose `right` points at the node this entry emits
};

// ---------------------------------------------------------------------------
// bag gather: bags[T][n][d] (f32) <- X[N][d] (f32 or bf16) at idx[T][n]
// ---------------------------------------------------------------------------

template <typename XT>
__device__ __forceinline__ float load_feat(const XT* X, int64_t off);

template <>
__device__ __forceinline__ float load_feat<float>(const float* X, int64_t off) {
  return X[off];
}
template <>
__device__ __forceinline__ float load_feat<uint16_t>(const uint16_t* X, int64_t off) {
  union { uint32_t u; float f; } cv;
  cv.u = ((uint32_t)X[off]) << 16;  // bf16 -> f32 (exact)
  return cv.f;
}

template <typename XT>
__global__ void bag_gather_kernel(const XT* __restrict__ X,
                                  const int64_t* __restrict__ bag_idx,
                                  float* __restrict__ bags, int64_t T, int64_t n,
                                  int64_t d) {
  int64_t total = T * n * d;
  for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; g < total;
       g += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = g % d;
    int64_t ts = g / d;  // t*n + s
    int64_t row = bag_idx[ts];
    bags[g] = load_feat<XT>(X, row * d + j);
  }
}

// ---------------------------------------------------------------------------
// standard build: one wavefront per tree
// ---------------------------------------------------------------------------

extern __shared__ __attribute__((aligned(16))) char smem[];

__global__ void __launch_bounds__(WAVE) build_forest_kernel(
    const float* __restrict__ bags,        // [T][n][d]
    const int32_t* __restrict__ feat_sub,  // [T][k] sorted global feature ids
    int32_t* __restrict__ out_feat,        // [T][max_nodes]
    float* __restrict__ out_value,         // [T][max_nodes]
    int32_t* __restrict__ out_right,       // [T][max_nodes]
    int32_t* __restrict__ out_count,       // [T][max_nodes] (leaf counts; -1)
    int32_t* __restrict__ out_ncount,      // [T]
    int32_t* __restrict__ out_depth,       // [T][max_nodes]
    const float* __restrict__ leaf_lut,    // [n+1]: c(m) float32
    uint64_t seed, int32_t tree_id_offset, int32_t n, int32_t d, int32_t k,
    int32_t max_nodes, int32_t height_limit) {
  const int t = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const uint32_t gtree = (uint32_t)(t + tree_id_offset);

  uint16_t* idx = (uint16_t*)smem;               // [n]
  uint16_t* tmp = idx + n;                       // [n]
  uint16_t* avail = tmp + n;                     // [k]
  SegEntry* stack = (SegEntry*)(((uintptr_t)(avail + k) + 15) & ~15ull);

  const float* bag = bags + (int64_t)t * n * d;
  int32_t* feat = out_feat + (int64_t)t * max_nodes;
  float* value = out_value + (int64_t)t * max_nodes;
  int32_t* right = out_right + (int64_t)t * max_nodes;
  int32_t* count = out_count + (int64_t)t * max_nodes;
  int32_t* depth = out_depth + (int64_t)t * max_nodes;
  const int32_t* features = feat_sub + (int64_t)t * k;

  for (int i = lane; i < n; i += WAVE) idx[i] = (uint16_t)i;
  if (lane == 0) stack[0] = SegEntry{0, (uint16_t)n, 0, -1};
  __syncthreads();

  int sp = 1;
  int next_id = 0;
  while (sp > 0) {
    SegEntry e = stack[--sp];
    const int node = next_id++;
    if (lane == 0) {
      if (e.patch >= 0) right[e.patch] = node;
      depth[node] = e.height;
    }
    const int start = e.start, end = e.end, m = end - start;

    if (m <= 1 || e.height >= height_limit) {
      if (lane == 0) {
        feat[node] = -1;
        value[node] = leaf_lut[m];
        count[node] = m;
      }
      continue;
    }

    // ---- feature selection with constant-feature retry (K1 + K11) ----
    for (int j = lane; j < k; j += WAVE) avail[j] = (uint16_t)features[j];
    __syncthreads();
    int found = -1;
    float fmin = 0.f, fmax = 0.f;
    for (int j = 0; j < k; ++j) {
      uint32_t r = rng_below(seed, P_FEATSEL, gtree, (uint32_t)node,
                             (uint32_t)(k - j), (uint32_t)j);
      int tsel = j + (int)r;
      if (lane == 0) {
        uint16_t a = avail[j];
        avail[j] = avail[tsel];
        avail[tsel] = a;
      }
      __syncthreads();
      const int f = avail[j];
      float lo = __builtin_inff(), hi = -__builtin_inff();
      for (int i = start + lane; i < end; i += WAVE) {
        float v = bag[(int64_t)idx[i] * d + f];
        lo = fminf(lo, v);
        hi = fmaxf(hi, v);
      }
      lo = wave_reduce_min(lo);
      hi = wave_reduce_max(hi);
      if (lo < hi) {
        found = f;
        fmin = lo;
        fmax = hi;
        break;
      }
    }
    if (found < 0) {
      if (lane == 0) {
        feat[node] = -1;
        value[node] = leaf_lut[m];
        count[node] = m;
      }
      continue;
    }

    const double u = rng_uniform(seed, P_SPLIT, gtree, (uint32_t)node);
    const float s32 = split_value32(fmin, fmax, u);

    // ---- stable partition via ballot/popcount (K2) ----
    int cntL = 0;
    for (int base = start; base < end; base += WAVE) {
      const int i = base + lane;
      const bool valid = i < end;
      float v = valid ? bag[(int64_t)idx[i] * d + found] : 0.f;
      const bool p = valid && (v < s32);
      cntL += __popcll(__ballot(p));
    }
    int runL = 0, runR = 0;
    for (int base = start; base < end; base += WAVE) {
      const int i = base + lane;
      const bool valid = i < end;
      uint16_t my = valid ? idx[i] : (uint16_t)0;
      float v = valid ? bag[(int64_t)my * d + found] : 0.f;
      const bool p = valid && (v < s32);
      const uint64_t maskL = __ballot(p);
      const uint64_t maskR = __ballot(valid && !p);
      if (p)
        tmp[start + runL + __popcll(maskL & LANE_MASK_LT(lane))] = my;
      else if (valid)
        tmp[start + cntL + runR + __popcll(maskR & LANE_MASK_LT(lane))] = my;
      runL += __popcll(maskL);
      runR += __popcll(maskR);
    }
    __syncthreads();
    for (int i = start + lane; i < end; i += WAVE) idx[i] = tmp[i];
    __syncthreads();

    if (lane == 0) {
      feat[node] = found;
      value[node] = s32;
      count[node] = -1;
      // pre-order: push right, then left (left pops first = node+1)
      stack[sp] = SegEntry{(uint16_t)(start + cntL), (uint16_t)end,
                           (int16_t)(e.height + 1), (int16_t)node};
      stack[sp + 1] = SegEntry{(uint16_t)start, (uint16_t)(start + cntL),
                               (int16_t)(e.height + 1), (int16_t)-1};
    }
    __syncthreads();
    sp += 2;
  }
  if (lane == 0) out_ncount[t] = next_id;
}

// ---------------------------------------------------------------------------
// extended build: one wavefront per tree
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(WAVE) build_extended_forest_kernel(
    const float* __restrict__ bags, const int32_t* __restrict__ feat_sub,
    int32_t* __restrict__ out_feat, float* __restrict__ out_value,
    int32_t* __restrict__ out_right, int32_t* __restrict__ out_count,
    int32_t* __restrict__ out_ncount,
    int32_t* __restrict__ out_hidx,    // [T][max_nodes][nnz]
    float* __restrict__ out_hw,        // [T][max_nodes][nnz]
    double* __restrict__ out_off64,    // [T][max_nodes]
    int32_t* __restrict__ out_depth,   // [T][max_nodes]
    const float* __restrict__ leaf_lut, uint64_t seed, int32_t tree_id_offset,
    int32_t n, int32_t d, int32_t k, int32_t nnz, int32_t max_nodes,
    int32_t height_limit) {
  const int t = blockIdx.x;
  const int lane = threadIdx.x & (WAVE - 1);
  const uint32_t gtree = (uint32_t)(t + tree_id_offset);

  uint16_t* idx = (uint16_t*)smem;   // [n]
  uint16_t* tmp = idx + n;           // [n]
  uint16_t* avail = tmp + n;         // [k]
  int32_t* coords =                  // [nnz]
      (int32_t*)(((uintptr_t)(avail + k) + 15) & ~15ull);
  float* wf = (float*)(coords + nnz);  // [nnz]
  SegEntry* stack = (SegEntry*)(((uintptr_t)(wf + nnz) + 15) & ~15ull);

  const float* bag = bags + (int64_t)t * n * d;
  int32_t* feat = out_feat + (int64_t)t * max_nodes;
  float* value = out_value + (int64_t)t * max_nodes;
  int32_t* right = out_right + (int64_t)t * max_nodes;
  int32_t* count = out_count + (int64_t)t * max_nodes;
  int32_t* hidx = out_hidx + (int64_t)t * max_nodes * nnz;
  float* hw = out_hw + (int64_t)t * max_nodes * nnz;
  double* off64p = out_off64 + (int64_t)t * max_nodes;
  int32_t* depth = out_depth + (int64_t)t * max_nodes;
  const int32_t* features = feat_sub + (int64_t)t * k;

  for (int i = lane; i < n; i += WAVE) idx[i] = (uint16_t)i;
  if (lane == 0) stack[0] = SegEntry{0, (uint16_t)n, 0, -1};
  __syncthreads();

  int sp = 1;
  int next_id = 0;
  while (sp > 0) {
    SegEntry e = stack[--sp];
    const int node = next_id++;
    if (lane == 0) {
      if (e.patch >= 0) right[e.patch] = node;
      depth[node] = e.height;
    }
    const int start = e.start, end = e.end, m = end - start;

    if (m <= 1 || e.height >= height_limit) {
      if (lane == 0) {
        feat[node] = -1;
        value[node] = leaf_lut[m];
        count[node] = m;
      }
      continue;
    }

    // ---- coordinate subset: Fisher-Yates over the tree's subspace ----
    for (int j = lane; j < k; j += WAVE) avail[j] = (uint16_t)features[j];
    __syncthreads();
    for (int j = 0; j < nnz; ++j) {
      uint32_t r = rng_below(seed, P_EIF_COORD, gtree, (uint32_t)node,
                             (uint32_t)(k - j), (uint32_t)j);
      if (lane == 0) {
        int tsel = j + (int)r;
        uint16_t a = avail[j];
        avail[j] = avail[tsel];
        avail[tsel] = a;
      }
      __syncthreads();
    }
    // sort the first nnz ascending (lane 0, insertion sort; nnz <= ~128)
    if (lane == 0) {
      for (int a = 1; a < nnz; ++a) {
        uint16_t key = avail[a];
        int b = a - 1;
        while (b >= 0 && avail[b] > key) {
          avail[b + 1] = avail[b];
          --b;
        }
        avail[b + 1] = key;
      }
      for (int j = 0; j < nnz; ++j) coords[j] = (int32_t)avail[j];
    }
    __syncthreads();

    // ---- Gaussian weights per sorted slot (det Box-Muller), parallel ----
    const uint32_t base = (uint32_t)node * 4096u;
    for (int j = lane; j < nnz; j += WAVE) {
      double u1, u2;
      rng_uniform2(seed, P_EIF_NORMAL, gtree, base + (uint32_t)j, &u1, &u2);
      wf[j] = (float)det_gaussian(u1, u2);
    }
    __syncthreads();
    // sequential float32 norm (order == CPU oracle)
    float acc = 0.f;
    for (int j = 0; j < nnz; ++j) acc = __fadd_rn(acc, __fmul_rn(wf[j], wf[j]));
    if (acc <= 0.f) {  // theoretical zero-norm leaf
      if (lane == 0) {
        feat[node] = -1;
        value[node] = leaf_lut[m];
        count[node] = m;
      }
      continue;
    }
    // sqrt/divide through double: __fsqrt_rn/__fdiv_rn are NOT correctly
    // rounded on this toolchain (1 ulp), numpy's f32 ops are; f64-then-round
    // is provably identical to correctly-rounded f32 (53 >= 2*24+2).
    const float norm = (float)sqrt((double)acc);
    for (int j = lane; j < nnz; j += WAVE)
      wf[j] = (float)((double)wf[j] / (double)norm);
    __syncthreads();

    // ---- intercepts per sorted coordinate + offset (float64) ----
    double off64 = 0.0;
    for (int j = 0; j < nnz; ++j) {
      const int c = coords[j];
      float lo = __builtin_inff(), hi = -__builtin_inff();
      for (int i = start + lane; i < end; i += WAVE) {
        float v = bag[(int64_t)idx[i] * d + c];
        lo = fminf(lo, v);
        hi = fmaxf(hi, v);
      }
      lo = wave_reduce_min(lo);
      hi = wave_reduce_max(hi);
      const double u =
          rng_uniform(seed, P_EIF_INTERCEPT, gtree, base + (uint32_t)j);
      const double intercept = (double)lo + u * ((double)hi - (double)lo);
      off64 += (double)wf[j] * intercept;
    }
    const float off32 = (float)off64;

    // ---- partition by hyperplane dot (no retry; empty sides allowed) ----
    int cntL = 0;
    for (int basei = start; basei < end; basei += WAVE) {
      const int i = basei + lane;
      const bool valid = i < end;
      float dot = 0.f;
      if (valid) {
        const float* rowp = bag + (int64_t)idx[i] * d;
        for (int j = 0; j < nnz; ++j)
          dot = __fadd_rn(dot, __fmul_rn(wf[j], rowp[coords[j]]));
      }
      cntL += __popcll(__ballot(valid && (dot < off32)));
    }
    int runL = 0, runR = 0;
    for (int basei = start; basei < end; basei += WAVE) {
      const int i = basei + lane;
      const bool valid = i < end;
      uint16_t my = valid ? idx[i] : (uint16_t)0;
      float dot = 0.f;
      if (valid) {
        const float* rowp = bag + (int64_t)my * d;
        for (int j = 0; j < nnz; ++j)
          dot = __fadd_rn(dot, __fmul_rn(wf[j], rowp[coords[j]]));
      }
      const bool p = valid && (dot < off32);
      const uint64_t maskL = __ballot(p);
      const uint64_t maskR = __ballot(valid && !p);
      if (p)
        tmp[start + runL + __popcll(maskL & LANE_MASK_LT(lane))] = my;
      else if (valid)
        tmp[start + cntL + runR + __popcll(maskR & LANE_MASK_LT(lane))] = my;
      runL += __popcll(maskL);
      runR += __popcll(maskR);
    }
    __syncthreads();
    for (int i = start + lane; i < end; i += WAVE) idx[i] = tmp[i];
    __syncthreads();

    if (lane == 0) {
      feat[node] = nnz;
      value[node] = off32;
      off64p[node] = off64;
      count[node] = -1;
      for (int j = 0; j < nnz; ++j) {
        hidx[(int64_t)node * nnz + j] = coords[j];
        hw[(int64_t)node * nnz + j] = wf[j];
      }
      stack[sp] = SegEntry{(uint16_t)(start + cntL), (uint16_t)end,
                           (int16_t)(e.height + 1), (int16_t)node};
      stack[sp + 1] = SegEntry{(uint16_t)start, (uint16_t)(start + cntL),
                               (int16_t)(e.height + 1), (int16_t)-1};
    }
    __syncthreads();
    sp += 2;
  }
  if (lane == 0) out_ncount[t] = next_id;
}

// ---------------------------------------------------------------------------
// standard scoring kernel (K6/K8) — v4
//
// Node record is 8 bytes {uint32 w0, uint32 w1}:
//   w0 = feature(12 bits) | right_child(15 bits) << 12
//   internal: w1 = integer KEY threshold (order-preserving transform of the
//             f32 split value; bf16 keys live in the HIGH 16 bits)
//   leaf:     feature = d (a sentinel column staged as all-ones), right =
//             OWN id (self-loop), w1 = f32 bits of (depth + c(numInstances))
//
// Rows are staged in LDS as order-preserving integer keys (u16 for bf16
// input, u32 for f32), so each visit is: ds_read_b64 node, ds_read LDS key,
// one unsigned compare, one select — no float convert, no leaf branch, no
// depth counter. Leaves self-loop (sentinel key 0xFF..F compares false
// against every threshold), so the walk is a FIXED height_limit-trip loop
// with zero per-chain bookkeeping; the leaf value (depth folded in, as the
// reference's ONNX converter does: isolation_forest_converter.py:343-373)
// is read once after the loop. Each thread walks RPT rows x 4 trees = 4*RPT
// independent dependent-load chains. Bitwise-identical to
// cpu_engine.path_lengths (tree-order f32 accumulation).
// ---------------------------------------------------------------------------

__device__ __forceinline__ float cvt_feat(float v) { return v; }
__device__ __forceinline__ float cvt_feat(uint16_t v) {
  union { uint32_t u; float f; } cv;
  cv.u = ((uint32_t)v) << 16;
  return cv.f;
}

__device__ __forceinline__ int pn_feat(int meta) { return meta & 0xFFF; }
__device__ __forceinline__ int pn_right(int meta) { return (meta >> 12) & 0x7FFF; }

// order-preserving key transforms (match gpu_engine._key16/_key32):
// +-0 collapse to the +0 key, NaN maps to the max key (so `x < t` is false:
// NaN routes right, same as the f32 compare in the oracle).
__device__ __forceinline__ uint32_t key16(uint32_t b) {
  const uint32_t m = b & 0x7FFFu;
  uint32_t k = (b & 0x8000u) ? (0x7FFFu - m) : (0x8000u + m);
  if (m == 0) k = 0x8000u;
  if (m > 0x7F80u) k = 0xFFFFu;
  return k;
}
__device__ __forceinline__ uint32_t key32(uint32_t b) {
  const uint32_t m = b & 0x7FFFFFFFu;
  uint32_t k = (b & 0x80000000u) ? (0x7FFFFFFFu - m) : (0x80000000u + m);
  if (m == 0) k = 0x80000000u;
  if (m > 0x7F800000u) k = 0xFFFFFFFFu;
  return k;
}

// KT = uint16_t (bf16 input) or uint32_t (f32 input)
template <typename KT>
__device__ __forceinline__ KT key_of_bits(KT b);
template <>
__device__ __forceinline__ uint16_t key_of_bits<uint16_t>(uint16_t b) {
  return (uint16_t)key16((uint32_t)b);
}
template <>
__device__ __forceinline__ uint32_t key_of_bits<uint32_t>(uint32_t b) {
  return key32(b);
}

// widen a stored key for the compare: bf16 keys sit in the high 16 bits with
// low bits saturated so the sentinel (0xFFFF) widens to 0xFFFFFFFF and
// compares false against every w1 (including leaf f32 payloads).
__device__ __forceinline__ uint32_t widen_key(uint16_t k) {
  return ((uint32_t)k << 16) | 0xFFFFu;
}
__device__ __forceinline__ uint32_t widen_key(uint32_t k) { return k; }

// row stride in elements for the extended kernels: keep rows 16-B aligned
template <typename XT>
__device__ __host__ __forceinline__ int row_stride(int d) {
  const int pad = 16 / (int)sizeof(XT);  // 8 for bf16, 4 for f32
  return d + pad;
}


template <typename KT, int RPT, bool ROWS_LDS, int V4_ILP, bool NODES_LDS>
__global__ void __launch_bounds__(256) score_forest_v4(
    const KT* __restrict__ X,          // raw bf16/f32 bits [N][d]
    const int2* __restrict__ nodes,    // [Tpad][max_nodes] packed v4
    const int32_t* __restrict__ ncnt,  // [Tpad]
    float* __restrict__ out,           // [N] (path sum or score)
    int64_t N, int32_t d, int32_t dpad, int32_t Tpad, int32_t max_nodes,
    int32_t height_limit, float fT, float c_norm, int32_t finalize) {
  const int tid = threadIdx.x;
  const int rows_per_iter = RPT * 256;

  int2* tlds = (int2*)smem;  // [V4_ILP * max_nodes] when NODES_LDS
  KT* rows = (KT*)(tlds + (NODES_LDS ? V4_ILP * max_nodes : 0));

  for (int64_t block_row0 = (int64_t)blockIdx.x * rows_per_iter; block_row0 < N;
       block_row0 += (int64_t)gridDim.x * rows_per_iter) {
    const int rows_here = (int)min((int64_t)rows_per_iter, N - block_row0);

    if (ROWS_LDS) {
      __syncthreads();  // previous iteration's readers done
      const int64_t total = (int64_t)rows_here * d;
      for (int64_t g = tid; g < total; g += 256) {
        const int r = (int)(g / (uint32_t)d), c = (int)(g % (uint32_t)d);
        rows[r * dpad + c] = key_of_bits<KT>(X[(block_row0 + r) * d + c]);
      }
      for (int r = tid; r < rows_per_iter; r += 256)
        rows[r * dpad + d] = (KT)~(KT)0;  // leaf sentinel column
      __syncthreads();
    }

    const KT* rb[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) rb[r] = rows + (tid + r * 256) * dpad;

    float psum[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) psum[r] = 0.f;

    for (int t = 0; t < Tpad; t += V4_ILP) {
      if (NODES_LDS) {
        __syncthreads();
#pragma unroll
        for (int s = 0; s < V4_ILP; ++s) {
          const int ncs = ncnt[t + s];
          const int2* ss = nodes + (int64_t)(t + s) * max_nodes;
          for (int i = tid; i < ncs; i += 256) tlds[s * max_nodes + i] = ss[i];
        }
        __syncthreads();
      }
      const int2* nbase[V4_ILP];
#pragma unroll
      for (int s = 0; s < V4_ILP; ++s)
        nbase[s] = NODES_LDS ? nullptr
                             : nodes + (int64_t)(t + s) * max_nodes;

      int cur[RPT][V4_ILP];
#pragma unroll
      for (int r = 0; r < RPT; ++r)
#pragma unroll
        for (int s = 0; s < V4_ILP; ++s) cur[r][s] = 0;

#pragma unroll 2
      for (int it = 0; it < height_limit; ++it) {
        // three explicit phases so the compiler BATCHES the LDS reads
        // (8 node reads -> one wait -> 8 key reads -> one wait -> VALU)
        // instead of serializing 8 dependent read+wait round-trips.
        int2 nd[RPT][V4_ILP];
#pragma unroll
        for (int r = 0; r < RPT; ++r)
#pragma unroll
          for (int s = 0; s < V4_ILP; ++s)
            nd[r][s] = NODES_LDS ? tlds[s * max_nodes + cur[r][s]]
                                 : nbase[s][cur[r][s]];
        KT kraw[RPT][V4_ILP];
#pragma unroll
        for (int r = 0; r < RPT; ++r) {
#pragma unroll
          for (int s = 0; s < V4_ILP; ++s) {
            const int f = pn_feat(nd[r][s].x);
            if (ROWS_LDS) {
              kraw[r][s] = rb[r][f];
            } else {
              // global fallback: the sentinel column does not exist in X,
              // so guard the leaf feature index.
              const int64_t my_row = block_row0 + tid + r * 256;
              kraw[r][s] = (f < d && my_row < N) ? X[my_row * d + f] : (KT)0;
            }
          }
        }
#pragma unroll
        for (int r = 0; r < RPT; ++r) {
#pragma unroll
          for (int s = 0; s < V4_ILP; ++s) {
            uint32_t x;
            if (ROWS_LDS) {
              x = widen_key(kraw[r][s]);
            } else {
              const int f = pn_feat(nd[r][s].x);
              x = (f < d) ? widen_key(key_of_bits<KT>(kraw[r][s]))
                          : 0xFFFFFFFFu;
            }
            cur[r][s] = (x < (uint32_t)nd[r][s].y) ? cur[r][s] + 1
                                                   : pn_right(nd[r][s].x);
          }
        }
      }
      // every chain rests at its leaf (self-loop); read value = depth + c(m)
#pragma unroll
      for (int r = 0; r < RPT; ++r)
#pragma unroll
        for (int s = 0; s < V4_ILP; ++s)
          psum[r] = __fadd_rn(
              psum[r],
              __int_as_float((NODES_LDS ? tlds[s * max_nodes + cur[r][s]]
                                        : nbase[s][cur[r][s]]).y));
    }

#pragma unroll
    for (int r = 0; r < RPT; ++r) {
      const int64_t my_row = block_row0 + tid + r * 256;
      if (my_row < N) {
        if (finalize) {
          const float mean32 = (float)((double)psum[r] / (double)fT);
          const double ratio = (double)mean32 / (double)c_norm;
          out[my_row] = (float)exp2(-ratio);
        } else {
          out[my_row] = psum[r];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// extended scoring, dense fast path (nnz == d): per visit a dense dot
// row[0..d) . w[0..d). WLDS stages the live tree's weight matrix in LDS
// (L2 could not feed the ~17 TB/s the global-weight variant demanded);
// rows are read as 8-byte vectors (4 bf16 or 2 f32 at a time).
// Accumulation: 4 partial sums, pairwise combine (deterministic, not the
// oracle's j-order — scoring parity tests use tolerance).
// ---------------------------------------------------------------------------

__device__ __forceinline__ void row4(const uint16_t* p, int j4, float& x0,
                                     float& x1, float& x2, float& x3) {
  const uint2 v = *(const uint2*)(p + 4 * j4);  // 8 B: 4 bf16
  union { uint32_t u; float f; } a, b, c, e;
  a.u = (v.x & 0xFFFFu) << 16;
  b.u = (v.x & 0xFFFF0000u);
  c.u = (v.y & 0xFFFFu) << 16;
  e.u = (v.y & 0xFFFF0000u);
  x0 = a.f; x1 = b.f; x2 = c.f; x3 = e.f;
}

__device__ __forceinline__ void row4(const float* p, int j4, float& x0,
                                     float& x1, float& x2, float& x3) {
  const float4 v = *(const float4*)(p + 4 * j4);
  x0 = v.x; x1 = v.y; x2 = v.z; x3 = v.w;
}

template <typename XT, bool ROWS_LDS, bool WLDS>
__global__ void __launch_bounds__(256) score_extended_dense_kernel(
    const XT* __restrict__ X, const int2* __restrict__ nodes,
    const float* __restrict__ hw_g,  // [T][max_nodes][d] dense weights
    const int32_t* __restrict__ ncnt, float* __restrict__ out, int64_t N,
    int32_t d, int32_t dpad, int32_t T, int32_t max_nodes, float fT,
    float c_norm, int32_t finalize) {
  const int tid = threadIdx.x;
  const int d4 = d >> 2;

  int2* tlds = (int2*)smem;                  // [max_nodes]
  float* wlds = (float*)(tlds + max_nodes);  // [max_nodes * d] if WLDS
  XT* rows = (XT*)(wlds + (WLDS ? max_nodes * d : 0));

  for (int64_t block_row0 = (int64_t)blockIdx.x * 256; block_row0 < N;
       block_row0 += (int64_t)gridDim.x * 256) {
    const int64_t my_row = block_row0 + tid;
    const int rows_here = (int)min((int64_t)256, N - block_row0);

    if (ROWS_LDS) {
      __syncthreads();
      const int64_t total = (int64_t)rows_here * d;
      for (int64_t g = tid; g < total; g += 256) {
        const int r = (int)(g / d), c = (int)(g % d);
        rows[r * dpad + c] = X[(block_row0 + r) * d + c];
      }
      __syncthreads();
    }
    const XT* my_lrow = rows + tid * dpad;

    float path_sum = 0.f;
    for (int t = 0; t < T; ++t) {
      __syncthreads();
      const int nc = ncnt[t];
      const int2* src = nodes + (int64_t)t * max_nodes;
      for (int i = tid; i < nc; i += 256) tlds[i] = src[i];
      if (WLDS) {
        const float4* wsrc = (const float4*)(hw_g + (int64_t)t * max_nodes * d);
        float4* wdst = (float4*)wlds;
        const int n4 = nc * d4;
        for (int i = tid; i < n4; i += 256) wdst[i] = wsrc[i];
      }
      __syncthreads();

      if (my_row < N) {
        const float* wbase =
            WLDS ? wlds : hw_g + (int64_t)t * max_nodes * d;
        int cur = 0, dep = 0;
        float leaf = 0.f;
        while (true) {
          const int2 nd = tlds[cur];
          if (nd.x < 0) {
            leaf = __int_as_float(nd.y);
            break;
          }
          const float4* wp = (const float4*)(wbase + (int64_t)cur * d);
          float ax = 0.f, ay = 0.f, az = 0.f, aw = 0.f;
          for (int j = 0; j < d4; ++j) {
            const float4 w = wp[j];
            float x0, x1, x2, x3;
            if (ROWS_LDS) {
              row4(my_lrow, j, x0, x1, x2, x3);
            } else {
              x0 = load_feat<XT>(X, my_row * d + 4 * j);
              x1 = load_feat<XT>(X, my_row * d + 4 * j + 1);
              x2 = load_feat<XT>(X, my_row * d + 4 * j + 2);
              x3 = load_feat<XT>(X, my_row * d + 4 * j + 3);
            }
            ax = __fadd_rn(ax, __fmul_rn(w.x, x0));
            ay = __fadd_rn(ay, __fmul_rn(w.y, x1));
            az = __fadd_rn(az, __fmul_rn(w.z, x2));
            aw = __fadd_rn(aw, __fmul_rn(w.w, x3));
          }
          const float dot =
              __fadd_rn(__fadd_rn(ax, ay), __fadd_rn(az, aw));
          cur = (dot < __int_as_float(nd.y)) ? cur + 1 : pn_right(nd.x);
          ++dep;
        }
        path_sum = __fadd_rn(path_sum, __fadd_rn((float)dep, leaf));
      }
    }
    if (my_row < N) {
      if (finalize) {
        const float mean32 = (float)((double)path_sum / (double)fT);
        const double ratio = (double)mean32 / (double)c_norm;
        out[my_row] = (float)exp2(-ratio);
      } else {
        out[my_row] = path_sum;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// extended scoring, dense fast path v2 (nnz == d, d <= 32).
//
// Rows live in REGISTERS (f32, templated D) across the whole tree loop —
// loaded once per 512-row block iteration, reused for all T trees. Per tree
// the node records {right|self-loop, offset32 / -inf}, the depth-folded leaf
// values and the dense weight matrix (row stride D+4 words to spread the
// b128 reads across banks) are staged in LDS. Each thread walks RPT rows
// concurrently through a FIXED height-trip loop (leaves self-loop: zero
// weight row + offset -inf routes right). The dot uses 4 partial fma
// accumulators (same reassociation contract as v1 — tolerance, not
// bitwise; build stays bitwise).
// ---------------------------------------------------------------------------

template <typename KT>  // u16 = bf16 rows, u32 = f32 rows
__device__ __forceinline__ float load_row_f32(const KT* X, int64_t off);
template <>
__device__ __forceinline__ float load_row_f32<uint16_t>(const uint16_t* X,
                                                        int64_t off) {
  return cvt_feat(X[off]);
}
template <>
__device__ __forceinline__ float load_row_f32<uint32_t>(const uint32_t* X,
                                                        int64_t off) {
  union { uint32_t u; float f; } cv;
  cv.u = X[off];
  return cv.f;
}

// Row storage for the dense-EIF walk: rows are converted to f32 ONCE at
// load (bf16->f32 is an exact shift) and held in registers across the
// whole tree loop, so the dot needs zero per-visit unpack VALU. Register
// budget: D=32 runs RPT=1 (32 row VGPRs), D<=16 runs RPT=2.
template <typename KT, int D>
struct RowReg {
  float v[D];
  __device__ __forceinline__ void load(const KT* X, int64_t base, int d,
                                       bool ok) {
#pragma unroll
    for (int j = 0; j < D; ++j)
      v[j] = (ok && j < d) ? load_row_f32<KT>(X, base + j) : 0.f;
  }
  __device__ __forceinline__ float get(int j) const { return v[j]; }
};

#define EIFD_THREADS 512

template <typename KT, int D, int RPT>
__global__ void __launch_bounds__(EIFD_THREADS,
                                  (D == 64) ? 2 : 4) score_extended_dense_v2(
    const KT* __restrict__ X,           // raw bits [N][d]
    const int2* __restrict__ nodes,     // [T][max_nodes] {w0, offset/-inf}
    const float* __restrict__ values,   // [T][max_nodes] leaf value+depth
    const float* __restrict__ hw,       // [T][max_nodes][D] PADDED dense w
    const int32_t* __restrict__ ncnt,   // [T]
    float* __restrict__ out, int64_t N, int32_t d, int32_t T,
    int32_t max_nodes, int32_t height_limit, float fT, float c_norm,
    int32_t finalize) {
  const int tid = threadIdx.x;
  const int rows_per_iter = RPT * EIFD_THREADS;
  const int DW4 = D / 4 + 1;  // LDS weight-row stride in float4s

  int2* tlds = (int2*)smem;                  // [max_nodes]
  float* vlds = (float*)(tlds + max_nodes);  // [max_nodes]
  float4* wlds = (float4*)(((uintptr_t)(vlds + max_nodes) + 15) & ~15ull);

  for (int64_t block_row0 = (int64_t)blockIdx.x * rows_per_iter; block_row0 < N;
       block_row0 += (int64_t)gridDim.x * rows_per_iter) {
    RowReg<KT, D> row[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) {
      const int64_t my_row = block_row0 + tid + r * EIFD_THREADS;
      row[r].load(X, my_row * d, d, my_row < N);
    }

    float psum[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) psum[r] = 0.f;

    for (int t = 0; t < T; ++t) {
      __syncthreads();
      const int nc = ncnt[t];
      {
        const int2* ss = nodes + (int64_t)t * max_nodes;
        const float* vs = values + (int64_t)t * max_nodes;
        for (int i = tid; i < nc; i += EIFD_THREADS) {
          tlds[i] = ss[i];
          vlds[i] = vs[i];
        }
        // weights are host-padded to D columns: pure float4 copy with
        // shift indexing (D/4 is a power of two for D in {8,16,32})
        const float4* ws = (const float4*)(hw + (int64_t)t * max_nodes * D);
        const int total4 = nc * (D / 4);
        for (int g = tid; g < total4; g += EIFD_THREADS) {
          const int i = g / (D / 4), j4 = g % (D / 4);
          wlds[i * DW4 + j4] = ws[g];
        }
      }
      __syncthreads();

      int cur[RPT];
#pragma unroll
      for (int r = 0; r < RPT; ++r) cur[r] = 0;

      for (int it = 0; it < height_limit; ++it) {
        int2 nd[RPT];
#pragma unroll
        for (int r = 0; r < RPT; ++r) nd[r] = tlds[cur[r]];
        // weight reads in chunks to bound live registers (quarter chunks at
        // D=32 keep the kernel under 128 VGPRs). Lanes whose walk already
        // ended (leaf self-loop: right == own id) are exec-masked out of
        // the dot — masked lanes issue no LDS accesses, cutting the bank-
        // conflict cycles that bound this kernel.
        constexpr int CHUNKS = (D >= 64) ? 8 : ((D >= 32) ? 4 : 2);
        constexpr int CW4 = D / (4 * CHUNKS);  // float4s per chunk
#pragma unroll
        for (int r = 0; r < RPT; ++r) {
          const int right = pn_right(nd[r].x);
          if (right != cur[r]) {
            float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
#pragma unroll
            for (int ch = 0; ch < CHUNKS; ++ch) {
              float4 w[CW4];
              const float4* wp = wlds + cur[r] * DW4 + ch * CW4;
#pragma unroll
              for (int j4 = 0; j4 < CW4; ++j4) w[j4] = wp[j4];
#pragma unroll
              for (int j4 = 0; j4 < CW4; ++j4) {
                const int j = ch * (D / CHUNKS) + 4 * j4;
                a0 = __builtin_fmaf(w[j4].x, row[r].get(j + 0), a0);
                a1 = __builtin_fmaf(w[j4].y, row[r].get(j + 1), a1);
                a2 = __builtin_fmaf(w[j4].z, row[r].get(j + 2), a2);
                a3 = __builtin_fmaf(w[j4].w, row[r].get(j + 3), a3);
              }
            }
            const float dot = __fadd_rn(__fadd_rn(a0, a1),
                                        __fadd_rn(a2, a3));
            cur[r] = (dot < __int_as_float(nd[r].y)) ? cur[r] + 1 : right;
          }
        }
      }
#pragma unroll
      for (int r = 0; r < RPT; ++r)
        psum[r] = __fadd_rn(psum[r], vlds[cur[r]]);
    }

#pragma unroll
    for (int r = 0; r < RPT; ++r) {
      const int64_t my_row = block_row0 + tid + r * EIFD_THREADS;
      if (my_row < N) {
        if (finalize) {
          const float mean32 = (float)((double)psum[r] / (double)fT);
          const double ratio = (double)mean32 / (double)c_norm;
          out[my_row] = (float)exp2(-ratio);
        } else {
          out[my_row] = psum[r];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// extended scoring, dense fast path v3 — bf16 INPUT rows only.
//
// The v2 kernel is bound by its per-visit 128-B f32 LDS weight-row read
// (PMC: 27% LDS conflict cycles, WAIT_ANY 44% — profiles/
// r01_bench_and_kernels.md). v3 halves that traffic: weights are staged in
// LDS as PACKED bf16 (round-to-nearest-even from the trained f32 weights,
// done once on device), rows are kept as packed-bf16 register pairs (the
// input is already bf16, so the pack is exact), and the dot runs on
// v_dot2c_f32_bf16 (2 MACs/instr, f32 accumulate) — 4 partial accumulator
// chains, same reassociation family as v2. Per wave-visit: 4x ds_read_b128
// (16 LDS-array cycles before conflicts) instead of v2's 8 (32 cycles),
// and ~half the dot VALU. Numerics: weights are bf16-rounded => knife-edge
// walk decisions can differ from the f32-weight oracle; the contract is
// documented in PARITY.md and tested against a bf16-weight oracle with the
// same isolated-flip allowance as the v2 dense route. f32 input rows keep
// the v2 kernel (unchanged f32-weight numerics).
// ---------------------------------------------------------------------------

typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));

__device__ __forceinline__ float dot2_bf16(uint32_t w, uint32_t x, float acc) {
  union { uint32_t u; bf16x2_t v; } cw, cx;
  cw.u = w;
  cx.u = x;
  return __builtin_amdgcn_fdot2_f32_bf16(cw.v, cx.v, acc, false);
}

template <int D, int RPT>
__global__ void __launch_bounds__(EIFD_THREADS,
                                  (D == 128) ? 2
                                             : ((D == 64
                                                 || (D == 32 && RPT == 2))
                                                    ? 4
                                                    : 6))
score_extended_dense_v3(
    const uint16_t* __restrict__ X,     // raw bf16 bits [N][d]
    const int2* __restrict__ nodes,     // [T][max_nodes] {right<<12, offset/-inf}
    const float* __restrict__ values,   // [T][max_nodes] leaf value+depth
    const uint32_t* __restrict__ hwp,   // [T][max_nodes][D/2] packed bf16 pairs
    const int32_t* __restrict__ ncnt,   // [T]
    float* __restrict__ out, int64_t N, int32_t d, int32_t T,
    int32_t max_nodes, int32_t height_limit, float fT, float c_norm,
    int32_t finalize) {
  const int tid = threadIdx.x;
  const int rows_per_iter = RPT * EIFD_THREADS;
  constexpr int PW4 = D / 8 + 1;  // LDS weight-row stride in uint4s

  int2* tlds = (int2*)smem;                  // [max_nodes]
  float* vlds = (float*)(tlds + max_nodes);  // [max_nodes]
  uint4* wlds = (uint4*)(((uintptr_t)(vlds + max_nodes) + 15) & ~15ull);

  for (int64_t block_row0 = (int64_t)blockIdx.x * rows_per_iter; block_row0 < N;
       block_row0 += (int64_t)gridDim.x * rows_per_iter) {
    uint32_t xp[RPT][D / 2];  // packed bf16 feature pairs, exact
#pragma unroll
    for (int r = 0; r < RPT; ++r) {
      const int64_t my_row = block_row0 + tid + r * EIFD_THREADS;
      const bool ok = my_row < N;
      const int64_t base = my_row * d;
#pragma unroll
      for (int j2 = 0; j2 < D / 2; ++j2) {
        const uint32_t lo = (ok && 2 * j2 < d) ? X[base + 2 * j2] : 0u;
        const uint32_t hi = (ok && 2 * j2 + 1 < d) ? X[base + 2 * j2 + 1] : 0u;
        xp[r][j2] = lo | (hi << 16);
      }
    }

    float psum[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) psum[r] = 0.f;

    for (int t = 0; t < T; ++t) {
      __syncthreads();
      const int nc = ncnt[t];
      {
        const int2* ss = nodes + (int64_t)t * max_nodes;
        const float* vs = values + (int64_t)t * max_nodes;
        for (int i = tid; i < nc; i += EIFD_THREADS) {
          tlds[i] = ss[i];
          vlds[i] = vs[i];
        }
        // packed rows are D/8 uint4s wide in global, PW4 in LDS
        const uint4* ws = (const uint4*)(hwp + (int64_t)t * max_nodes * (D / 2));
        const int total4 = nc * (D / 8);
        for (int g = tid; g < total4; g += EIFD_THREADS) {
          const int i = g / (D / 8), j4 = g % (D / 8);
          wlds[i * PW4 + j4] = ws[g];
        }
      }
      __syncthreads();

      int cur[RPT];
#pragma unroll
      for (int r = 0; r < RPT; ++r) cur[r] = 0;

      for (int it = 0; it < height_limit; ++it) {
        int2 nd[RPT];
#pragma unroll
        for (int r = 0; r < RPT; ++r) nd[r] = tlds[cur[r]];
#pragma unroll
        for (int r = 0; r < RPT; ++r) {
          const int right = pn_right(nd[r].x);
          if (right != cur[r]) {  // exec-mask finished walks out of the dot
            // chunked reads bound live w registers (D=32 stays under the
            // 80-VGPR budget that keeps 6 waves/SIMD resident)
            constexpr int CHUNKS = (D >= 32) ? 2 : 1;
            constexpr int CW4 = D / (8 * CHUNKS);  // uint4s per chunk
            const uint4* wp = wlds + cur[r] * PW4;
            float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
#pragma unroll
            for (int ch = 0; ch < CHUNKS; ++ch) {
              uint4 w[CW4];
#pragma unroll
              for (int j4 = 0; j4 < CW4; ++j4) w[j4] = wp[ch * CW4 + j4];
#pragma unroll
              for (int j4 = 0; j4 < CW4; ++j4) {
                const int p = (ch * CW4 + j4) * 4;
                a0 = dot2_bf16(w[j4].x, xp[r][p + 0], a0);
                a1 = dot2_bf16(w[j4].y, xp[r][p + 1], a1);
                a2 = dot2_bf16(w[j4].z, xp[r][p + 2], a2);
                a3 = dot2_bf16(w[j4].w, xp[r][p + 3], a3);
              }
            }
            const float dot = __fadd_rn(__fadd_rn(a0, a1),
                                        __fadd_rn(a2, a3));
            cur[r] = (dot < __int_as_float(nd[r].y)) ? cur[r] + 1 : right;
          }
        }
      }
#pragma unroll
      for (int r = 0; r < RPT; ++r)
        psum[r] = __fadd_rn(psum[r], vlds[cur[r]]);
    }

#pragma unroll
    for (int r = 0; r < RPT; ++r) {
      const int64_t my_row = block_row0 + tid + r * EIFD_THREADS;
      if (my_row < N) {
        if (finalize) {
          const float mean32 = (float)((double)psum[r] / (double)fT);
          const double ratio = (double)mean32 / (double)c_norm;
          out[my_row] = (float)exp2(-ratio);
        } else {
          out[my_row] = psum[r];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// extended scoring, sparse fast path v2 (nnz <= 5, templated):
// same node/value packing as the dense v2 kernel (self-looping leaves with
// offset -inf and zero weights => fixed-trip walks), rows staged native in
// LDS, hyperplanes (idx+w) staged in LDS, RPT=2 chains per thread with
// batch-phased reads. The dot keeps the ORACLE's strict j-order f32
// accumulation (bitwise vs cpu_engine.path_lengths_extended).
// ---------------------------------------------------------------------------

template <typename KT, int NNZ, int RPT>
__global__ void __launch_bounds__(256) score_extended_sparse_v2(
    const KT* __restrict__ X,           // raw bits [N][d]
    const int2* __restrict__ nodes,     // [T][max_nodes] {w0, offset/-inf}
    const float* __restrict__ values,   // [T][max_nodes] leaf value+depth
    const int32_t* __restrict__ hidx_g, // [T][max_nodes][NNZ]
    const float* __restrict__ hw_g,     // [T][max_nodes][NNZ]
    const int32_t* __restrict__ ncnt, float* __restrict__ out, int64_t N,
    int32_t d, int32_t dpad, int32_t T, int32_t max_nodes,
    int32_t height_limit, float fT, float c_norm, int32_t finalize) {
  const int tid = threadIdx.x;
  const int rows_per_iter = RPT * 256;

  int2* tlds = (int2*)smem;                   // [max_nodes]
  float* vlds = (float*)(tlds + max_nodes);   // [max_nodes]
  int32_t* ilds = (int32_t*)(vlds + max_nodes);  // [max_nodes][NNZ]
  float* wlds = (float*)(ilds + max_nodes * NNZ);  // [max_nodes][NNZ]
  KT* rows = (KT*)(wlds + max_nodes * NNZ);   // [rows_per_iter][dpad]

  for (int64_t block_row0 = (int64_t)blockIdx.x * rows_per_iter; block_row0 < N;
       block_row0 += (int64_t)gridDim.x * rows_per_iter) {
    const int rows_here = (int)min((int64_t)rows_per_iter, N - block_row0);
    __syncthreads();  // previous iteration's readers done
    {
      const int64_t total = (int64_t)rows_here * d;
      for (int64_t g = tid; g < total; g += 256) {
        const int r = (int)(g / (uint32_t)d), c = (int)(g % (uint32_t)d);
        rows[r * dpad + c] = X[(block_row0 + r) * d + c];
      }
    }
    __syncthreads();
    const KT* rb[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) rb[r] = rows + (tid + r * 256) * dpad;

    float psum[RPT];
#pragma unroll
    for (int r = 0; r < RPT; ++r) psum[r] = 0.f;

    for (int t = 0; t < T; ++t) {
      __syncthreads();
      const int nc = ncnt[t];
      {
        const int2* ss = nodes + (int64_t)t * max_nodes;
        const float* vs = values + (int64_t)t * max_nodes;
        for (int i = tid; i < nc; i += 256) {
          tlds[i] = ss[i];
          vlds[i] = vs[i];
        }
        const int32_t* is = hidx_g + (int64_t)t * max_nodes * NNZ;
        const float* ws = hw_g + (int64_t)t * max_nodes * NNZ;
        for (int g = tid; g < nc * NNZ; g += 256) {
          ilds[g] = is[g];
          wlds[g] = ws[g];
        }
      }
      __syncthreads();

      int cur[RPT];
#pragma unroll
      for (int r = 0; r < RPT; ++r) cur[r] = 0;

      for (int it = 0; it < height_limit; ++it) {
        int2 nd[RPT];
#pragma unroll
        for (int r = 0; r < RPT; ++r) nd[r] = tlds[cur[r]];
        int32_t ci[RPT][NNZ];
#pragma unroll
        for (int r = 0; r < RPT; ++r)
#pragma unroll
          for (int j = 0; j < NNZ; ++j) ci[r][j] = ilds[cur[r] * NNZ + j];
        float cw[RPT][NNZ];
#pragma unroll
        for (int r = 0; r < RPT; ++r)
#pragma unroll
          for (int j = 0; j < NNZ; ++j) cw[r][j] = wlds[cur[r] * NNZ + j];
        float xv[RPT][NNZ];
#pragma unroll
        for (int r = 0; r < RPT; ++r)
#pragma unroll
          for (int j = 0; j < NNZ; ++j)
            xv[r][j] = load_row_f32<KT>(rb[r], ci[r][j]);
#pragma unroll
        for (int r = 0; r < RPT; ++r) {
          float dot = 0.f;
#pragma unroll
          for (int j = 0; j < NNZ; ++j)  // oracle j-order
            dot = __fadd_rn(dot, __fmul_rn(cw[r][j], xv[r][j]));
          const int right = pn_right(nd[r].x);
          cur[r] = (dot < __int_as_float(nd[r].y)) ? cur[r] + 1 : right;
        }
      }
#pragma unroll
      for (int r = 0; r < RPT; ++r)
        psum[r] = __fadd_rn(psum[r], vlds[cur[r]]);
    }

#pragma unroll
    for (int r = 0; r < RPT; ++r) {
      const int64_t my_row = block_row0 + tid + r * 256;
      if (my_row < N) {
        if (finalize) {
          const float mean32 = (float)((double)psum[r] / (double)fT);
          const double ratio = (double)mean32 / (double)c_norm;
          out[my_row] = (float)exp2(-ratio);
        } else {
          out[my_row] = psum[r];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// extended scoring, sparse general path (nnz < d): exact oracle order.
// ---------------------------------------------------------------------------

template <typename XT, bool ROWS_LDS, bool HYPER_LDS, bool NODES_LDS>
__global__ void __launch_bounds__(256) score_extended_forest_kernel(
    const XT* __restrict__ X, const int2* __restrict__ nodes,
    const int32_t* __restrict__ hidx_g,  // [T][max_nodes][nnz]
    const float* __restrict__ hw_g,      // [T][max_nodes][nnz]
    const int32_t* __restrict__ ncnt, float* __restrict__ out, int64_t N,
    int32_t d, int32_t T, int32_t max_nodes, int32_t nnz, float fT,
    float c_norm, int32_t finalize) {
  const int tid = threadIdx.x;
  const int dpad = row_stride<XT>(d);

  int2* tlds = (int2*)smem;  // [max_nodes] when NODES_LDS
  int32_t* hidx_lds = (int32_t*)(tlds + (NODES_LDS ? max_nodes : 0));
  float* hw_lds = (float*)(hidx_lds + (HYPER_LDS ? max_nodes * nnz : 0));
  XT* rows = (XT*)(hw_lds + (HYPER_LDS ? max_nodes * nnz : 0));

  for (int64_t block_row0 = (int64_t)blockIdx.x * 256; block_row0 < N;
       block_row0 += (int64_t)gridDim.x * 256) {
    const int64_t my_row = block_row0 + tid;
    const int rows_here = (int)min((int64_t)256, N - block_row0);

    if (ROWS_LDS) {
      __syncthreads();
      const int64_t total = (int64_t)rows_here * d;
      for (int64_t g = tid; g < total; g += 256) {
        const int r = (int)(g / d), c = (int)(g % d);
        rows[r * dpad + c] = X[(block_row0 + r) * d + c];
      }
      __syncthreads();
    }
    const XT* my_lrow = rows + tid * dpad;

    float path_sum = 0.f;
    for (int t = 0; t < T; ++t) {
      __syncthreads();
      const int nc = ncnt[t];
      const int2* src = nodes + (int64_t)t * max_nodes;
      if (NODES_LDS)
        for (int i = tid; i < nc; i += 256) tlds[i] = src[i];
      if (HYPER_LDS) {
        const int64_t hbase = (int64_t)t * max_nodes * nnz;
        for (int i = tid; i < nc * nnz; i += 256) {
          hidx_lds[i] = hidx_g[hbase + i];
          hw_lds[i] = hw_g[hbase + i];
        }
      }
      __syncthreads();

      if (my_row < N) {
        const int64_t hbase = (int64_t)t * max_nodes * nnz;
        int cur = 0, dep = 0;
        float leaf = 0.f;
        while (true) {
          const int2 nd = NODES_LDS ? tlds[cur] : src[cur];
          if (nd.x < 0) {
            leaf = __int_as_float(nd.y);
            break;
          }
          float dot = 0.f;
          const int32_t* ci = HYPER_LDS ? hidx_lds + cur * nnz
                                        : hidx_g + hbase + (int64_t)cur * nnz;
          const float* cw = HYPER_LDS ? hw_lds + cur * nnz
                                      : hw_g + hbase + (int64_t)cur * nnz;
          for (int j = 0; j < nnz; ++j) {
            const float xv = ROWS_LDS
                                 ? cvt_feat(my_lrow[ci[j]])
                                 : load_feat<XT>(X, my_row * d + ci[j]);
            dot = __fadd_rn(dot, __fmul_rn(cw[j], xv));
          }
          cur = (dot < __int_as_float(nd.y)) ? cur + 1 : pn_right(nd.x);
          ++dep;
        }
        path_sum = __fadd_rn(path_sum, __fadd_rn((float)dep, leaf));
      }
    }
    if (my_row < N) {
      if (finalize) {
        const float mean32 = (float)((double)path_sum / (double)fT);
        const double ratio = (double)mean32 / (double)c_norm;
        out[my_row] = (float)exp2(-ratio);
      } else {
        out[my_row] = path_sum;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// WIDE-format fallback scoring (no packed-field width limits).
//
// The packed v4/EIF node formats cap feature ids at 12 bits (4094) and
// node ids at 15 bits (32767 nodes/tree = maxSamples 16384). Forests
// loaded from foreign files (or CPU-built past the GPU build cap) can
// exceed both; these kernels store each node as a full int4
// {feat, right, key/value, 0} read from GLOBAL memory (L2-served: every
// workgroup walks trees in the same order) with rows read from global —
// a correctness path, not a throughput path (the reference has no such
// size limits: IsolationTree.scala). Same integer-key compare math as v4
// => bitwise vs cpu_engine.path_lengths.
// ---------------------------------------------------------------------------

template <typename KT>
__global__ void __launch_bounds__(256) score_forest_wide(
    const KT* __restrict__ X,          // raw bf16/f32 bits [N][d]
    const int4* __restrict__ nodes,    // [T][max_nodes] {feat, right, w, 0}
    float* __restrict__ out, int64_t N, int32_t d, int32_t T,
    int64_t max_nodes, int32_t height_limit, float fT, float c_norm,
    int32_t finalize) {
  for (int64_t my_row = (int64_t)blockIdx.x * 256 + threadIdx.x; my_row < N;
       my_row += (int64_t)gridDim.x * 256) {
    float psum = 0.f;
    for (int t = 0; t < T; ++t) {
      const int4* base = nodes + (int64_t)t * max_nodes;
      int cur = 0;
      for (int it = 0; it < height_limit; ++it) {
        const int4 nd = base[cur];
        if (nd.y != cur) {  // leaves self-loop (right == own id)
          const uint32_t x =
              widen_key(key_of_bits<KT>(X[my_row * d + nd.x]));
          cur = (x < (uint32_t)nd.z) ? cur + 1 : nd.y;
        }
      }
      psum = __fadd_rn(psum, __int_as_float(base[cur].z));
    }
    if (finalize) {
      const float mean32 = (float)((double)psum / (double)fT);
      const double ratio = (double)mean32 / (double)c_norm;
      out[my_row] = (float)exp2(-ratio);
    } else {
      out[my_row] = psum;
    }
  }
}

// EIF wide fallback: full-width node records, strict oracle j-order dot
// from global hyperplanes (bitwise vs cpu_engine.path_lengths_extended).
template <typename XT>
__global__ void __launch_bounds__(256) score_extended_wide(
    const XT* __restrict__ X, const int4* __restrict__ nodes,
    const int32_t* __restrict__ hidx_g,  // [T][max_nodes][nnz]
    const float* __restrict__ hw_g,      // [T][max_nodes][nnz]
    float* __restrict__ out, int64_t N, int32_t d, int32_t T,
    int64_t max_nodes, int32_t nnz, float fT, float c_norm,
    int32_t finalize) {
  for (int64_t my_row = (int64_t)blockIdx.x * 256 + threadIdx.x; my_row < N;
       my_row += (int64_t)gridDim.x * 256) {
    float path_sum = 0.f;
    for (int t = 0; t < T; ++t) {
      const int4* base = nodes + (int64_t)t * max_nodes;
      const int64_t hbase = (int64_t)t * max_nodes * nnz;
      int cur = 0, dep = 0;
      float leaf = 0.f;
      while (true) {
        const int4 nd = base[cur];
        if (nd.x < 0) {
          leaf = __int_as_float(nd.z);
          break;
        }
        float dot = 0.f;
        const int32_t* ci = hidx_g + hbase + (int64_t)cur * nnz;
        const float* cw = hw_g + hbase + (int64_t)cur * nnz;
        for (int j = 0; j < nnz; ++j) {
          const float xv = load_feat<XT>(X, my_row * d + ci[j]);
          dot = __fadd_rn(dot, __fmul_rn(cw[j], xv));
        }
        cur = (dot < __int_as_float(nd.z)) ? cur + 1 : nd.y;
        ++dep;
      }
      path_sum = __fadd_rn(path_sum, __fadd_rn((float)dep, leaf));
    }
    if (finalize) {
      const float mean32 = (float)((double)path_sum / (double)fT);
      const double ratio = (double)mean32 / (double)c_norm;
      out[my_row] = (float)exp2(-ratio);
    } else {
      out[my_row] = path_sum;
    }
  }
}

}  // namespace ifa

// ---------------------------------------------------------------------------
// host-side launchers (bindings.cpp is compiled by g++, so every kernel
// launch lives in this hipcc TU)
// ---------------------------------------------------------------------------

namespace ifa {

static inline void raise_lds(const void* func, size_t bytes) {
  if (bytes > 64 * 1024)
    (void)hipFuncSetAttribute(func, hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)bytes);
}

void launch_bag_gather(bool bf16, const void* X, const int64_t* bag_idx,
                       float* bags, int64_t T, int64_t n, int64_t d,
                       hipStream_t stream) {
  int64_t total = T * n * d;
  int threads = 256;
  int blocks = (int)((total + threads - 1) / threads);
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  if (bf16)
    hipLaunchKernelGGL(bag_gather_kernel<uint16_t>, dim3(blocks), dim3(threads),
                       0, stream, (const uint16_t*)X, bag_idx, bags, T, n, d);
  else
    hipLaunchKernelGGL(bag_gather_kernel<float>, dim3(blocks), dim3(threads), 0,
                       stream, (const float*)X, bag_idx, bags, T, n, d);
}

void launch_build_forest(const float* bags, const int32_t* feat_sub,
                         int32_t* feat, float* value, int32_t* right,
                         int32_t* count, int32_t* ncount, int32_t* depth,
                         const float* leaf_lut, uint64_t seed,
                         int32_t tree_id_offset, int32_t T, int32_t n,
                         int32_t d, int32_t k, int32_t max_nodes,
                         int32_t height_limit, size_t lds,
                         hipStream_t stream) {
  raise_lds((const void*)build_forest_kernel, lds);
  hipLaunchKernelGGL(build_forest_kernel, dim3(T), dim3(WAVE), lds, stream,
                     bags, feat_sub, feat, value, right, count, ncount,
                     depth, leaf_lut, seed, tree_id_offset, n, d, k,
                     max_nodes, height_limit);
}

void launch_build_extended_forest(const float* bags, const int32_t* feat_sub,
                                  int32_t* feat, float* value, int32_t* right,
                                  int32_t* count, int32_t* ncount,
                                  int32_t* hidx, float* hw, double* off64,
                                  int32_t* depth,
                                  const float* leaf_lut, uint64_t seed,
                                  int32_t tree_id_offset, int32_t T, int32_t n,
                                  int32_t d, int32_t k, int32_t nnz,
                                  int32_t max_nodes, int32_t height_limit,
                                  size_t lds, hipStream_t stream) {
  raise_lds((const void*)build_extended_forest_kernel, lds);
  hipLaunchKernelGGL(build_extended_forest_kernel, dim3(T), dim3(WAVE), lds,
                     stream, bags, feat_sub, feat, value, right, count, ncount,
                     hidx, hw, off64, depth, leaf_lut, seed, tree_id_offset,
                     n, d, k, nnz, max_nodes, height_limit);
}

void launch_score_forest(bool bf16, int rpt, bool rows_lds, bool nodes_lds,
                         int ilp, const void* X, const void* nodes,
                         const int32_t* ncount, float* out, int64_t N,
                         int32_t d, int32_t dpad, int32_t Tpad,
                         int32_t max_nodes, int32_t height_limit, float fT,
                         float c_norm, int finalize, size_t lds, int blocks,
                         hipStream_t stream) {
#define LS(KT, RPT, RL, TI)                                                   \
  do {                                                                        \
    if (nodes_lds) {                                                          \
      raise_lds((const void*)score_forest_v4<KT, RPT, RL, TI, true>, lds);    \
      hipLaunchKernelGGL((score_forest_v4<KT, RPT, RL, TI, true>),            \
                         dim3(blocks), dim3(256), lds, stream, (const KT*)X,  \
                         (const int2*)nodes, ncount, out, N, d, dpad, Tpad,   \
                         max_nodes, height_limit, fT, c_norm, finalize);      \
    } else {                                                                  \
      raise_lds((const void*)score_forest_v4<KT, RPT, RL, TI, false>, lds);   \
      hipLaunchKernelGGL((score_forest_v4<KT, RPT, RL, TI, false>),           \
                         dim3(blocks), dim3(256), lds, stream, (const KT*)X,  \
                         (const int2*)nodes, ncount, out, N, d, dpad, Tpad,   \
                         max_nodes, height_limit, fT, c_norm, finalize);      \
    }                                                                         \
  } while (0)
#define LS_ILP(KT, RPT, RL)                                                   \
  do {                                                                        \
    if (ilp == 8) LS(KT, RPT, RL, 8);                                         \
    else if (ilp == 2) LS(KT, RPT, RL, 2);                                    \
    else LS(KT, RPT, RL, 4);                                                  \
  } while (0)
  if (bf16) {
    if (rows_lds && rpt == 2) LS_ILP(uint16_t, 2, true);
    else if (rows_lds) LS_ILP(uint16_t, 1, true);
    else LS_ILP(uint16_t, 1, false);
  } else {
    if (rows_lds && rpt == 2) LS_ILP(uint32_t, 2, true);
    else if (rows_lds) LS_ILP(uint32_t, 1, true);
    else LS_ILP(uint32_t, 1, false);
  }
#undef LS_ILP
#undef LS
}

void launch_score_extended_dense(bool bf16, bool rows_lds, bool wlds,
                                 const void* X, const void* nodes,
                                 const float* hw, const int32_t* ncount,
                                 float* out, int64_t N, int32_t d,
                                 int32_t dpad, int32_t T, int32_t max_nodes,
                                 float fT, float c_norm, int finalize,
                                 size_t lds, int blocks, hipStream_t stream) {
#define LSD(XT, RL, WL)                                                       \
  do {                                                                        \
    raise_lds((const void*)score_extended_dense_kernel<XT, RL, WL>, lds);     \
    hipLaunchKernelGGL((score_extended_dense_kernel<XT, RL, WL>),             \
                       dim3(blocks), dim3(256), lds, stream, (const XT*)X,    \
                       (const int2*)nodes, hw, ncount, out, N, d, dpad, T,    \
                       max_nodes, fT, c_norm, finalize);                      \
  } while (0)
  if (bf16) {
    if (rows_lds && wlds) LSD(uint16_t, true, true);
    else if (rows_lds) LSD(uint16_t, true, false);
    else if (wlds) LSD(uint16_t, false, true);
    else LSD(uint16_t, false, false);
  } else {
    if (rows_lds && wlds) LSD(float, true, true);
    else if (rows_lds) LSD(float, true, false);
    else if (wlds) LSD(float, false, true);
    else LSD(float, false, false);
  }
#undef LSD
}

void launch_score_extended_dense_v2(bool bf16, int D, const void* X,
                                    const void* nodes, const float* values,
                                    const float* hw, const int32_t* ncount,
                                    float* out, int64_t N, int32_t d,
                                    int32_t T, int32_t max_nodes,
                                    int32_t height_limit, float fT,
                                    float c_norm, int finalize, size_t lds,
                                    int blocks, hipStream_t stream) {
#define LSD2(KT, DD, RR)                                                      \
  do {                                                                        \
    raise_lds((const void*)score_extended_dense_v2<KT, DD, RR>, lds);         \
    hipLaunchKernelGGL((score_extended_dense_v2<KT, DD, RR>), dim3(blocks),   \
                       dim3(EIFD_THREADS), lds, stream, (const KT*)X,         \
                       (const int2*)nodes, values, hw, ncount, out, N, d, T,  \
                       max_nodes, height_limit, fT, c_norm, finalize);        \
  } while (0)
  // D=32 runs RPT=1: 2 rows/thread would spill past the 128-VGPR budget
  // that keeps 4 waves/SIMD resident (2 blocks/CU at ~80 KB LDS each)
  if (bf16) {
    if (D == 8) LSD2(uint16_t, 8, 2);
    else if (D == 16) LSD2(uint16_t, 16, 2);
    else if (D == 64) LSD2(uint16_t, 64, 1);
    else LSD2(uint16_t, 32, 1);
  } else {
    if (D == 8) LSD2(uint32_t, 8, 2);
    else if (D == 16) LSD2(uint32_t, 16, 2);
    else if (D == 64) LSD2(uint32_t, 64, 1);
    else LSD2(uint32_t, 32, 1);
  }
#undef LSD2
}

void launch_score_forest_wide(bool bf16, const void* X, const void* nodes,
                              float* out, int64_t N, int32_t d, int32_t T,
                              int64_t max_nodes, int32_t height_limit,
                              float fT, float c_norm, int finalize,
                              int blocks, hipStream_t stream) {
  if (bf16)
    hipLaunchKernelGGL((score_forest_wide<uint16_t>), dim3(blocks), dim3(256),
                       0, stream, (const uint16_t*)X, (const int4*)nodes, out,
                       N, d, T, max_nodes, height_limit, fT, c_norm, finalize);
  else
    hipLaunchKernelGGL((score_forest_wide<uint32_t>), dim3(blocks), dim3(256),
                       0, stream, (const uint32_t*)X, (const int4*)nodes, out,
                       N, d, T, max_nodes, height_limit, fT, c_norm, finalize);
}

void launch_score_extended_wide(bool bf16, const void* X, const void* nodes,
                                const int32_t* hidx, const float* hw,
                                float* out, int64_t N, int32_t d, int32_t T,
                                int64_t max_nodes, int32_t nnz, float fT,
                                float c_norm, int finalize, int blocks,
                                hipStream_t stream) {
  if (bf16)
    hipLaunchKernelGGL((score_extended_wide<uint16_t>), dim3(blocks),
                       dim3(256), 0, stream, (const uint16_t*)X,
                       (const int4*)nodes, hidx, hw, out, N, d, T, max_nodes,
                       nnz, fT, c_norm, finalize);
  else
    hipLaunchKernelGGL((score_extended_wide<float>), dim3(blocks), dim3(256),
                       0, stream, (const float*)X, (const int4*)nodes, hidx,
                       hw, out, N, d, T, max_nodes, nnz, fT, c_norm,
                       finalize);
}

void launch_score_extended_dense_v3(int D, bool rpt2, const void* X,
                                    const void* nodes, const float* values,
                                    const uint32_t* hwp,
                                    const int32_t* ncount, float* out,
                                    int64_t N, int32_t d, int32_t T,
                                    int32_t max_nodes, int32_t height_limit,
                                    float fT, float c_norm, int finalize,
                                    size_t lds, int blocks,
                                    hipStream_t stream) {
#define LSD3(DD, RR)                                                          \
  do {                                                                        \
    raise_lds((const void*)score_extended_dense_v3<DD, RR>, lds);             \
    hipLaunchKernelGGL((score_extended_dense_v3<DD, RR>), dim3(blocks),       \
                       dim3(EIFD_THREADS), lds, stream, (const uint16_t*)X,   \
                       (const int2*)nodes, values, hwp, ncount, out, N, d, T, \
                       max_nodes, height_limit, fT, c_norm, finalize);        \
  } while (0)
  if (D == 8) LSD3(8, 2);
  else if (D == 16) LSD3(16, 2);
  else if (D == 32 && rpt2) LSD3(32, 2);  // A/B: halved staging amortization
  else if (D == 32) LSD3(32, 1);
  else if (D == 64) LSD3(64, 1);
  else LSD3(128, 1);
#undef LSD3
}

void launch_score_extended_sparse_v2(bool bf16, int nnz, const void* X,
                                     const void* nodes, const float* values,
                                     const int32_t* hidx, const float* hw,
                                     const int32_t* ncount, float* out,
                                     int64_t N, int32_t d, int32_t dpad,
                                     int32_t T, int32_t max_nodes,
                                     int32_t height_limit, float fT,
                                     float c_norm, int finalize, size_t lds,
                                     int blocks, hipStream_t stream) {
#define LSS2(KT, NZ)                                                          \
  do {                                                                        \
    raise_lds((const void*)score_extended_sparse_v2<KT, NZ, 2>, lds);         \
    hipLaunchKernelGGL((score_extended_sparse_v2<KT, NZ, 2>), dim3(blocks),   \
                       dim3(256), lds, stream, (const KT*)X,                  \
                       (const int2*)nodes, values, hidx, hw, ncount, out, N,  \
                       d, dpad, T, max_nodes, height_limit, fT, c_norm,       \
                       finalize);                                             \
  } while (0)
#define LSS2_ALL(KT)                                                          \
  do {                                                                        \
    switch (nnz) {                                                            \
      case 1: LSS2(KT, 1); break;                                             \
      case 2: LSS2(KT, 2); break;                                             \
      case 3: LSS2(KT, 3); break;                                             \
      case 4: LSS2(KT, 4); break;                                             \
      default: LSS2(KT, 5); break;                                            \
    }                                                                         \
  } while (0)
  if (bf16) LSS2_ALL(uint16_t); else LSS2_ALL(uint32_t);
#undef LSS2_ALL
#undef LSS2
}

void launch_score_extended_forest(bool bf16, bool rows_lds, bool hyper_lds,
                                  bool nodes_lds, const void* X,
                                  const void* nodes, const int32_t* hidx,
                                  const float* hw, const int32_t* ncount,
                                  float* out, int64_t N, int32_t d, int32_t T,
                                  int32_t max_nodes, int32_t nnz, float fT,
                                  float c_norm, int finalize, size_t lds,
                                  int blocks, hipStream_t stream) {
#define LSE(XT, RL, HL)                                                       \
  do {                                                                        \
    if (nodes_lds) {                                                          \
      raise_lds((const void*)score_extended_forest_kernel<XT, RL, HL, true>,  \
                lds);                                                         \
      hipLaunchKernelGGL((score_extended_forest_kernel<XT, RL, HL, true>),    \
                         dim3(blocks), dim3(256), lds, stream, (const XT*)X,  \
                         (const int2*)nodes, hidx, hw, ncount, out, N, d, T,  \
                         max_nodes, nnz, fT, c_norm, finalize);               \
    } else {                                                                  \
      raise_lds((const void*)score_extended_forest_kernel<XT, RL, HL, false>, \
                lds);                                                         \
      hipLaunchKernelGGL((score_extended_forest_kernel<XT, RL, HL, false>),   \
                         dim3(blocks), dim3(256), lds, stream, (const XT*)X,  \
                         (const int2*)nodes, hidx, hw, ncount, out, N, d, T,  \
                         max_nodes, nnz, fT, c_norm, finalize);               \
    }                                                                         \
  } while (0)
  if (bf16) {
    if (rows_lds && hyper_lds) LSE(uint16_t, true, true);
    else if (rows_lds) LSE(uint16_t, true, false);
    else if (hyper_lds) LSE(uint16_t, false, true);
    else LSE(uint16_t, false, false);
  } else {
    if (rows_lds && hyper_lds) LSE(float, true, true);
    else if (rows_lds) LSE(float, true, false);
    else if (hyper_lds) LSE(float, false, true);
    else LSE(float, false, false);
  }
#undef LSE
}

}  // namespace ifa
