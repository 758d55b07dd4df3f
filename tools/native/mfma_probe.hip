// MFMA experiment for the EIF dense scoring dot.
//
// Question: can v_mfma_f32_32x32x16_bf16 beat the per-visit LDS dot of
// score_extended_dense_v3 by precomputing ALL node dots per (row-tile,
// tree) — S = X_tile @ W^T — and turning the walk into compare-only?
//
// Static analysis (profiles/r02_eif_v3.md) says no: S must round-trip
// through LDS (64 KB per 64-row tile per tree => ~1 KB of ds_write per
// row-tree, vs v3's ~550 B of conflicted ds_read per row-tree), and a
// 64-row tile re-stages the 16 KB weight matrix 16x more often than
// v3's 1024-row blocks. This probe measures the floor of the MFMA
// variant's per-(row, tree) cost against a v3-style per-visit loop on
// identical synthetic data, so the decision is grounded in a measurement
// on the target silicon, not an argument.
//
// Build: hipcc --offload-arch=gfx950 -O3 -o mfma_probe mfma_probe.hip
// Run:   ./mfma_probe [rows] [trees]   (prints one JSON line)

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define HIP_CHECK(x)                                                     \
  do {                                                                   \
    hipError_t e = (x);                                                  \
    if (e != hipSuccess) {                                               \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e),   \
              __FILE__, __LINE__);                                       \
      exit(1);                                                           \
    }                                                                    \
  } while (0)

typedef __bf16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;   // A/B frag
typedef __attribute__((ext_vector_type(16))) float f32x16;   // C/D frag
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

constexpr int D = 32;       // feature dim (k of the GEMM)
constexpr int NODES = 256;  // internal nodes per tree
constexpr int TILE = 64;    // rows per MFMA tile
constexpr int VISITS = 8;   // walk length

__device__ __forceinline__ float dot2_bf16(uint32_t w, uint32_t x,
                                           float acc) {
  union { uint32_t u; bf16x2 v; } cw, cx;
  cw.u = w;
  cx.u = x;
  return __builtin_amdgcn_fdot2_f32_bf16(cw.v, cx.v, acc, false);
}

// ---------------------------------------------------------------------------
// layout self-check: one 32x32x16 MFMA, D written back with the documented
// C/D mapping col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5); A/B lane
// mapping row(col)=lane&31, k=8*(lane>>5)+e.
// ---------------------------------------------------------------------------
__global__ void layout_check_kernel(const bf16* A, const bf16* B, float* Dm) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
  for (int e = 0; e < 8; ++e) {
    const int k = 8 * (lane >> 5) + e;
    a[e] = A[(lane & 31) * 16 + k];  // A[row][k], row-major 32x16
    b[e] = B[(lane & 31) * 16 + k];  // B[col][k] (B^T row-major 32x16)
  }
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    const int col = lane & 31;
    Dm[row * 32 + col] = c[r];
  }
}

// ---------------------------------------------------------------------------
// Probe A: MFMA precompute. Block = 256 threads (4 waves) owning TILE=64
// rows. Per tree: stage W (NODES x D bf16) from global, each wave MFMAs a
// 64-col slice of S = X_tile @ W^T into LDS (64x256 f32), then wave 0
// walks all 64 rows compare-only (VISITS dependent S reads each).
// ---------------------------------------------------------------------------
extern __shared__ __attribute__((aligned(16))) char smem[];

__global__ void __launch_bounds__(256) probe_mfma(
    const bf16* __restrict__ X,   // [rows][D]
    const bf16* __restrict__ W,   // [NODES][D] (one tree, re-read per tree)
    float* __restrict__ out, int64_t rows, int trees) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;

  bf16* xl = (bf16*)smem;                       // [TILE][D]
  bf16* wl = xl + TILE * D;                     // [NODES][D]
  float* S = (float*)(wl + NODES * D);          // [TILE][NODES]

  for (int64_t row0 = (int64_t)blockIdx.x * TILE; row0 < rows;
       row0 += (int64_t)gridDim.x * TILE) {
    __syncthreads();
    for (int i = tid; i < TILE * D; i += 256)
      xl[i] = X[row0 * D + i];
    __syncthreads();

    float acc = 0.f;
    for (int t = 0; t < trees; ++t) {
      __syncthreads();
      for (int i = tid; i < NODES * D; i += 256) wl[i] = W[i];
      __syncthreads();

      // wave w computes S[:, 64w : 64w+64): 2 row-tiles x 2 col-tiles,
      // k split into 2 MFMA steps of 16
      for (int rt = 0; rt < 2; ++rt) {
        for (int ct = 0; ct < 2; ++ct) {
          f32x16 c = {};
          for (int kk = 0; kk < 2; ++kk) {
            bf16x8 a, b;
            const int arow = rt * 32 + (lane & 31);
            const int bcol = wave * 64 + ct * 32 + (lane & 31);
            for (int e = 0; e < 8; ++e) {
              const int k = kk * 16 + 8 * (lane >> 5) + e;
              a[e] = xl[arow * D + k];
              b[e] = wl[bcol * D + k];
            }
            c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
          }
          for (int r = 0; r < 16; ++r) {
            const int row = rt * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
            const int col = wave * 64 + ct * 32 + (lane & 31);
            S[row * NODES + col] = c[r];
          }
        }
      }
      __syncthreads();

      // compare-only walk: wave 0 walks the 64 rows (others idle — the
      // cost model charges the whole block anyway)
      if (wave == 0) {
        int cur = 0;
        for (int v = 0; v < VISITS; ++v) {
          const float s = S[lane * NODES + cur];
          // synthetic branch: next node from the dot bits (data-dependent)
          cur = (((__float_as_uint(s) >> 9) ^ cur * 2654435761u)
                 % (NODES - 1)) + 1;
          acc += s;
        }
      }
      __syncthreads();
    }
    if (wave == 0) out[row0 + lane] = acc;
  }
}

// ---------------------------------------------------------------------------
// Probe B: v3-style per-visit dot. Block = 512 threads, each thread owns a
// row (packed-bf16 x in registers), per visit 4 uint4 LDS weight reads +
// 16 v_dot2c. Same W staging per tree, 1024... here 512 rows per block to
// keep geometry comparable to the shipped v3 (RPT=1).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(512, 6) probe_visit(
    const bf16* __restrict__ X, const bf16* __restrict__ W,
    float* __restrict__ out, int64_t rows, int trees) {
  const int tid = threadIdx.x;
  constexpr int PW4 = D / 8 + 1;  // padded row stride in uint4
  uint4* wl = (uint4*)smem;       // [NODES][PW4]

  for (int64_t row0 = (int64_t)blockIdx.x * 512; row0 < rows;
       row0 += (int64_t)gridDim.x * 512) {
    uint32_t xp[D / 2];
    const int64_t base = (row0 + tid) * D;
    for (int j2 = 0; j2 < D / 2; ++j2) {
      union { bf16 b[2]; uint32_t u; } cv;
      cv.b[0] = X[base + 2 * j2];
      cv.b[1] = X[base + 2 * j2 + 1];
      xp[j2] = cv.u;
    }
    float acc = 0.f;
    for (int t = 0; t < trees; ++t) {
      __syncthreads();
      const uint4* ws = (const uint4*)W;
      for (int g = tid; g < NODES * (D / 8); g += 512) {
        const int i = g / (D / 8), j4 = g % (D / 8);
        wl[i * PW4 + j4] = ws[g];
      }
      __syncthreads();
      int cur = 0;
      for (int v = 0; v < VISITS; ++v) {
        const uint4* wp = wl + cur * PW4;
        float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
        for (int ch = 0; ch < 2; ++ch) {
          uint4 w[2];
          for (int j4 = 0; j4 < 2; ++j4) w[j4] = wp[ch * 2 + j4];
          for (int j4 = 0; j4 < 2; ++j4) {
            const int p = (ch * 2 + j4) * 4;
            a0 = dot2_bf16(w[j4].x, xp[p + 0], a0);
            a1 = dot2_bf16(w[j4].y, xp[p + 1], a1);
            a2 = dot2_bf16(w[j4].z, xp[p + 2], a2);
            a3 = dot2_bf16(w[j4].w, xp[p + 3], a3);
          }
        }
        const float s = (a0 + a1) + (a2 + a3);
        cur = (((__float_as_uint(s) >> 9) ^ cur * 2654435761u)
               % (NODES - 1)) + 1;
        acc += s;
      }
    }
    out[row0 + tid] = acc;
  }
}

static double run_ms(void (*kernel)(const bf16*, const bf16*, float*,
                                    int64_t, int),
                     int block, size_t lds, const bf16* X, const bf16* W,
                     float* out, int64_t rows, int trees, int reps) {
  int rows_per_block = (kernel == probe_mfma) ? TILE : 512;
  int blocks = (int)std::min<int64_t>((rows + rows_per_block - 1)
                                      / rows_per_block, 8192);
  if (lds > 65536)
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)kernel, hipFuncAttributeMaxDynamicSharedMemorySize,
        (int)lds));
  hipEvent_t a, b;
  HIP_CHECK(hipEventCreate(&a));
  HIP_CHECK(hipEventCreate(&b));
  // warmup
  hipLaunchKernelGGL(kernel, dim3(blocks), dim3(block), lds, 0, X, W, out,
                     rows, trees);
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(a));
  for (int r = 0; r < reps; ++r)
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(block), lds, 0, X, W, out,
                       rows, trees);
  HIP_CHECK(hipEventRecord(b));
  HIP_CHECK(hipEventSynchronize(b));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, a, b));
  return ms / reps;
}

int main(int argc, char** argv) {
  int64_t rows = argc > 1 ? atoll(argv[1]) : 2'000'000;
  int trees = argc > 2 ? atoi(argv[2]) : 200;

  // layout self-check
  std::vector<bf16> A(32 * 16), B(32 * 16);
  std::vector<float> Dref(32 * 32), Dgot(32 * 32);
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 16; ++k) {
      A[i * 16 + k] = (bf16)((i * 7 + k * 3) % 13 - 6);
      B[i * 16 + k] = (bf16)((i * 5 + k * 11) % 17 - 8);  // asymmetric
    }
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j) {
      float s = 0;
      for (int k = 0; k < 16; ++k)
        s += (float)A[i * 16 + k] * (float)B[j * 16 + k];
      Dref[i * 32 + j] = s;
    }
  bf16 *dA, *dB;
  float* dD;
  HIP_CHECK(hipMalloc(&dA, A.size() * 2));
  HIP_CHECK(hipMalloc(&dB, B.size() * 2));
  HIP_CHECK(hipMalloc(&dD, Dref.size() * 4));
  HIP_CHECK(hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice));
  hipLaunchKernelGGL(layout_check_kernel, dim3(1), dim3(64), 0, 0, dA, dB,
                     dD);
  HIP_CHECK(hipMemcpy(Dgot.data(), dD, Dref.size() * 4,
                      hipMemcpyDeviceToHost));
  int bad = 0;
  for (int i = 0; i < 32 * 32; ++i)
    if (Dref[i] != Dgot[i]) ++bad;

  // probe data
  bf16 *dX, *dW;
  float* dout;
  HIP_CHECK(hipMalloc(&dX, (size_t)rows * D * 2));
  HIP_CHECK(hipMalloc(&dW, (size_t)NODES * D * 2));
  HIP_CHECK(hipMalloc(&dout, (size_t)rows * 4));
  {
    std::vector<bf16> h((size_t)1 << 20);
    for (size_t i = 0; i < h.size(); ++i)
      h[i] = (bf16)(((int)(i * 2654435761u >> 16) % 997) / 997.0f - 0.5f);
    for (size_t off = 0; off < (size_t)rows * D; off += h.size()) {
      size_t n = std::min(h.size(), (size_t)rows * D - off);
      HIP_CHECK(hipMemcpy(dX + off, h.data(), n * 2,
                          hipMemcpyHostToDevice));
    }
    HIP_CHECK(hipMemcpy(dW, h.data(), (size_t)NODES * D * 2,
                        hipMemcpyHostToDevice));
  }

  size_t lds_mfma = (size_t)TILE * D * 2 + (size_t)NODES * D * 2
                    + (size_t)TILE * NODES * 4;
  size_t lds_visit = (size_t)NODES * (D / 8 + 1) * 16;
  double ms_mfma = run_ms(probe_mfma, 256, lds_mfma, dX, dW, dout, rows,
                          trees, 3);
  double ms_visit = run_ms(probe_visit, 512, lds_visit, dX, dW, dout, rows,
                           trees, 3);

  printf("{\"probe\": \"eif_mfma_vs_visit\", \"layout_check_bad\": %d, "
         "\"rows\": %lld, \"trees\": %d, \"d\": %d, \"nodes\": %d, "
         "\"visits\": %d, \"ms_mfma\": %.3f, \"ms_visit\": %.3f, "
         "\"mfma_over_visit\": %.3f}\n",
         bad, (long long)rows, trees, D, NODES, VISITS, ms_mfma, ms_visit,
         ms_mfma / ms_visit);
  return bad == 0 ? 0 : 2;
}
