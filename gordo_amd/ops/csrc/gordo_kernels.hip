// gordo_amd HIP kernels for MI355X (gfx950, CDNA4).
//
// The compute core of the framework: grouped (batched-over-models) MFMA
// GEMMs and the fused pointwise training ops. Every kernel operates on
// packed [G, ...] tensors where G = models trained in lockstep
// (SURVEY.md §2.3 kernels K1-K9).
//
// GEMM design (cdna_hip_programming.md §5):
//  * mfma_f32_16x16x32_bf16, fp32 accumulation in AGPRs.
//  * 64x64 output tile per 256-thread workgroup; 4 waves in a 2x2
//    layout, each wave owns a 32x32 sub-tile = 2x2 fragments of 16x16.
//  * K-step 32. A and B tiles staged through LDS with rows padded to
//    40 bf16 (80 B row stride = 20 dword banks; 16 consecutive rows map
//    to 16 distinct bank groups, so ds_read_b128 fragment reads are
//    conflict-free; 80 % 16 == 0 keeps the b128 alignment rule of §6
//    Guideline 17).
//  * The B operand is staged TRANSPOSED into LDS (Bt[n][k]) so both
//    fragment reads are row-contiguous 16-byte ds_read_b128.
//  * Grid is (g, n-tile, m-tile) flattened with consecutive blocks
//    sharing the same weight panel, remapped XCD-aware (bijective,
//    §5.5 T1) so panel reuse hits the same XCD's L2.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

#define DEV_INLINE __device__ __forceinline__

using bf16 = __hip_bfloat16;

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

// activation codes (keep in sync with gordo_amd/ops/reference.py)
constexpr int ACT_LINEAR = 0, ACT_TANH = 1, ACT_RELU = 2, ACT_SIGMOID = 3;

DEV_INLINE float fast_tanhf(float x);

DEV_INLINE float act_apply(float z, int act) {
  switch (act) {
    case ACT_TANH: return fast_tanhf(z);
    case ACT_RELU: return z > 0.f ? z : 0.f;
    case ACT_SIGMOID: return 1.f / (1.f + __expf(-z));
    default: return z;
  }
}

DEV_INLINE float act_grad_from_output(float y, int act) {
  switch (act) {
    case ACT_TANH: return 1.f - y * y;
    case ACT_RELU: return y > 0.f ? 1.f : 0.f;
    case ACT_SIGMOID: return y * (1.f - y);
    default: return 1.f;
  }
}

DEV_INLINE float fast_tanhf(float x) {
  // tanh via one __expf: the ocml tanh is an order of magnitude slower
  // and the LSTM gate math is transcendental-bound.
  float e = __expf(2.f * x);
  return 1.f - 2.f / (e + 1.f);  // saturates to 1 when e overflows to inf
}

DEV_INLINE float bf2f(bf16 v) { return __bfloat162float(v); }
DEV_INLINE bf16 f2bf(float v) { return __float2bfloat16(v); }

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

// gather 8 bf16 into registers, store as ONE 16-byte ds_write_b128:
// element-wise 2-byte staging writes were the dominant LDS bank
// conflict source (profiles/lstm_pmc_r01.txt).
DEV_INLINE void lds_store8(bf16* dst, const bf16 (&v)[8]) {
  *reinterpret_cast<u32x4*>(dst) = *reinterpret_cast<const u32x4*>(v);
}

// ---------------------------------------------------------------------------
// Grouped GEMM: C[g, M, N] = op( A[g, M, K] @ B + bias ), B per mode:
//   BMODE_KN: B[g, K, N] row-major (forward: X @ W)
//   BMODE_NK: B[g, N, K] row-major (dgrad: dZ @ W^T with W=[K_out? see py])
// ACC: C += result instead of overwrite (LSTM recurrent gates).
// ---------------------------------------------------------------------------
constexpr int BM = 64, BN = 64, BK = 32;
constexpr int LDT = 40;  // padded LDS row length (bf16 elements)

struct GemmCoord {
  int g, m0, n0;
};

DEV_INLINE GemmCoord decode_block(int nblocks, int mt, int nt) {
  // bijective XCD-aware remap (cdna_hip_programming.md §5 T1):
  // blocks sharing a weight panel (same g, n-tile) become contiguous per
  // XCD so the panel stays in one XCD's L2.
  int bid = blockIdx.x;
  constexpr int NXCD = 8;
  if (nblocks >= NXCD) {
    int q = nblocks / NXCD, r = nblocks % NXCD;
    int xcd = bid % NXCD, idx = bid / NXCD;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  GemmCoord c;
  int per_g = mt * nt;
  c.g = bid / per_g;
  int rem = bid % per_g;
  c.n0 = (rem / mt) * BN;
  c.m0 = (rem % mt) * BM;
  return c;
}

template <int BMODE, bool ACC, bool FUSE_BIAS_ACT>
__global__ __launch_bounds__(256) void grouped_gemm_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ bias, bf16* __restrict__ C,
    int M, int N, int K, int act, int mt, int nt, int nblocks) {
  __shared__ bf16 sm[2 * BM * LDT];
  bf16* As = sm;               // [BM][LDT]
  bf16* Bs = sm + BM * LDT;    // [BN][LDT]  (Bt: [n][k])

  GemmCoord blk = decode_block(nblocks, mt, nt);
  const int g = blk.g, m0 = blk.m0, n0 = blk.n0;
  const bf16* Ag = A + (size_t)g * M * K;
  const bf16* Bg = B + (size_t)g * ((BMODE == 0) ? (size_t)K * N : (size_t)N * K);
  bf16* Cg = C + (size_t)g * M * N;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // 4 waves
  const int wm = (wid & 1) * 32;     // wave row offset in tile
  const int wn = (wid >> 1) * 32;    // wave col offset in tile

  f32x4 acc[2][2] = {};

  const int l15 = lane & 15;
  const int kslot = lane >> 4;  // 0..3 -> k offset kslot*8

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A[m0..m0+64][k0..k0+32] -> As[row][k] ----
    // 256 threads x 8 elements; thread t: row=t/4, k8=(t%4)*8
    {
      int row = tid >> 2;
      int kk = (tid & 3) * 8;
      int gm = m0 + row;
      bf16 v[8];
      if ((K & 7) == 0 && gm < M && k0 + kk + 8 <= K) {
        // K 8-aligned => every row's 8-element k-slice is one 16-byte
        // naturally-aligned global load (guide G13; the scalar loop
        // below is the unaligned-K fallback)
        *reinterpret_cast<bf16x8*>(v) = *reinterpret_cast<const bf16x8*>(
            &Ag[(size_t)gm * K + k0 + kk]);
      } else {
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
          int gk = k0 + kk + e;
          v[e] = (gm < M && gk < K) ? Ag[(size_t)gm * K + gk] : f2bf(0.f);
        }
      }
      lds_store8(&As[row * LDT + kk], v);
    }
    // ---- stage B -> Bs[n][k] (transposed for BMODE_KN) ----
    if (BMODE == 0) {
      // B[K][N]: thread t reads col n=t/4 over k; scatter into Bs[n][k]
      int n = tid >> 2;
      int kk = (tid & 3) * 8;
      int gn = n0 + n;
      bf16 v[8];
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        int gk = k0 + kk + e;
        v[e] = (gn < N && gk < K) ? Bg[(size_t)gk * N + gn] : f2bf(0.f);
      }
      lds_store8(&Bs[n * LDT + kk], v);
    } else {
      // B[N][K]: rows are n — direct copy
      int n = tid >> 2;
      int kk = (tid & 3) * 8;
      int gn = n0 + n;
      bf16 v[8];
      if ((K & 7) == 0 && gn < N && k0 + kk + 8 <= K) {
        *reinterpret_cast<bf16x8*>(v) = *reinterpret_cast<const bf16x8*>(
            &Bg[(size_t)gn * K + k0 + kk]);
      } else {
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
          int gk = k0 + kk + e;
          v[e] = (gn < N && gk < K) ? Bg[(size_t)gn * K + gk] : f2bf(0.f);
        }
      }
      lds_store8(&Bs[n * LDT + kk], v);
    }
    __syncthreads();

    // ---- MFMA over the two 16-deep k sub-steps of BK=32 ----
    #pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      #pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        // A fragment: rows wm+fm*16+l15, k = ks + kslot*8 .. +8
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &As[(wm + fm * 16 + l15) * LDT + ks + kslot * 8]);
        #pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          bf16x8 b = *reinterpret_cast<const bf16x8*>(
              &Bs[(wn + fn * 16 + l15) * LDT + ks + kslot * 8]);
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[fm][fn], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: bias + activation, guarded store ----
  #pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
    #pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      int col = n0 + wn + fn * 16 + l15;
      if (col >= N) continue;
      float bv = FUSE_BIAS_ACT ? bias[(size_t)g * N + col] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm + fm * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[fm][fn][r] + bv;
        if (FUSE_BIAS_ACT) v = act_apply(v, act);
        size_t off = (size_t)row * N + col;
        if (ACC) v += bf2f(Cg[off]);
        Cg[off] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Grouped wgrad: dW[g, K, N] (fp32) = A[g, M, K]^T @ dZ[g, M, N].
// Output tile 64x64 over (K, N); reduction dim is M. Both operands are
// staged transposed/direct so fragment reads stay row-contiguous:
//   At[k][m] (transpose of A tile), Zt[n][m] (transpose of dZ tile).
// ---------------------------------------------------------------------------
// tshiftT == 0: A rows are read as-is. tshiftT == T > 0: row gm maps
// to (b, t) = (gm/T, gm%T) and reads A at row gm-1 (i.e. h_{t-1}),
// with t == 0 rows ZERO — the recurrent-wgrad addressing that
// removes the h_prev_all concat from the BPTT path (a full [G,B,T,H]
// copy per layer per batch; VERDICT round-1 weak #7).
// A2/K1: optional SECOND A operand for the recurrent layer's combined
// weight-grad — K columns [0, K1) read A (stride K1, the layer input
// sequence) and [K1, K) read A2 (stride K-K1, the h sequence with the
// t-1 shift), producing [dWx ; dWh] in ONE pass so dZ is staged once
// instead of twice (dZ staging is half of each call's global traffic).
__global__ __launch_bounds__(256) void grouped_wgrad_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ dZ,
    float* __restrict__ dW, float* __restrict__ db, int M, int N, int K,
    int kt, int nt, int nblocks, int mchunk, int tshiftT,
    const bf16* __restrict__ A2, int K1) {
  __shared__ bf16 sm[2 * BM * LDT];
  bf16* As = sm;             // At: [k 64][m 32+pad]
  bf16* Zs = sm + BM * LDT;  // Zt: [n 64][m 32+pad]

  // split-M: blockIdx.y selects an M-chunk; partials atomicAdd into the
  // fp32 dW (the reduction dim M is B*T ~ 37k for LSTM wgrads — without
  // the split only G*kt*nt workgroups exist and the chip idles).
  const int m_begin = blockIdx.y * mchunk;
  const int m_end = min(M, m_begin + mchunk);

  GemmCoord blk = decode_block(nblocks, kt, nt);
  const int g = blk.g, k0 = blk.m0, n0 = blk.n0;  // m-slot carries k-tile
  const int K2 = K - K1;
  const bf16* Ag = A + (size_t)g * M * K1;
  const bf16* A2g = A2 ? A2 + (size_t)g * M * K2 : nullptr;
  const bf16* Zg = dZ + (size_t)g * M * N;
  float* Wg = dW + (size_t)g * K * N;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wk = (wid & 1) * 32;
  const int wn = (wid >> 1) * 32;
  const int l15 = lane & 15;
  const int mslot = lane >> 4;

  f32x4 acc[2][2] = {};
  float db_part = 0.f;

  // Vectorized transpose staging (row-contiguous 16-byte loads +
  // scalar LDS-transposed writes) MEASURED SLOWER than the legacy
  // column gather at the fleet shape (0.288 vs 0.254 ms at
  // M=36864 K=56 N=192 G=31 — gpurun r2_call10): the column gather's
  // lane pattern already coalesces into 4x32B segments per
  // instruction, while 8 ds_write_b16 per thread cost more LDS issue
  // than the loads saved. Kept compiled-out for future re-tuning.
  constexpr bool vecA = false;
  constexpr bool vecZ = false;
  (void)0;
  for (int m0 = m_begin; m0 < m_end; m0 += BK) {
    // stage A[m0..+32][k0..+64] transposed into As[k][m]
    if (vecA) {
      int m = tid >> 3;           // 0..31
      int kk8 = (tid & 7) * 8;    // 0..56
      int gm = m0 + m;
      int gk0 = k0 + kk8;
      bool ok = gm < M;
      size_t row = (size_t)gm;
      if (tshiftT > 0) {
        ok = ok && (gm % tshiftT) > 0;  // t == 0 -> h_prev is zero
        row = (size_t)gm - 1;
      }
      bf16 v[8];
      if (ok && gk0 + 8 <= K) {
        *reinterpret_cast<bf16x8*>(v) =
            *reinterpret_cast<const bf16x8*>(&Ag[row * K + gk0]);
      } else {
        #pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = (ok && gk0 + e < K) ? Ag[row * K + gk0 + e] : f2bf(0.f);
      }
      #pragma unroll
      for (int e = 0; e < 8; ++e) As[(kk8 + e) * LDT + m] = v[e];
    } else {
      int k = tid >> 2;           // 0..63
      int mm = (tid & 3) * 8;     // 0..24
      int gk = k0 + k;
      const bool part2 = A2g != nullptr && gk >= K1;
      bf16 v[8];
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        int gm = m0 + mm + e;
        bool ok = (gk < K && gm < M);
        size_t row = (size_t)gm;
        // the t-1 shift applies to the whole A when single-operand
        // (the hprev entry), or to the A2 half in combined mode
        if (tshiftT > 0 && (part2 || A2g == nullptr)) {
          ok = ok && (gm % tshiftT) > 0;  // t == 0 -> h_prev is zero
          row = (size_t)gm - 1;
        }
        v[e] = !ok ? f2bf(0.f)
               : part2 ? A2g[row * K2 + (gk - K1)]
                       : Ag[row * K1 + gk];
      }
      lds_store8(&As[k * LDT + mm], v);
    }
    // stage dZ[m0..+32][n0..+64] transposed into Zs[n][m]; k-tile-0
    // blocks fold the bias-grad column sum into the same pass (what a
    // separate colsum kernel did with a latency-bound column walk).
    if (vecZ) {
      int m = tid >> 3;
      int nn8 = (tid & 7) * 8;
      int gm = m0 + m;
      int gn0 = n0 + nn8;
      bf16 v[8];
      if (gm < M && gn0 + 8 <= N) {
        *reinterpret_cast<bf16x8*>(v) =
            *reinterpret_cast<const bf16x8*>(&Zg[(size_t)gm * N + gn0]);
      } else {
        #pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = (gm < M && gn0 + e < N) ? Zg[(size_t)gm * N + gn0 + e]
                                         : f2bf(0.f);
      }
      #pragma unroll
      for (int e = 0; e < 8; ++e) Zs[(nn8 + e) * LDT + m] = v[e];
      if (k0 == 0) {
        // bias-grad partials from LDS after the transpose (the legacy
        // in-flight accumulation had per-n thread ownership)
        __syncthreads();
        int n = tid >> 2;
        int mm = (tid & 3) * 8;
        #pragma unroll
        for (int e = 0; e < 8; ++e)
          db_part += bf2f(Zs[n * LDT + mm + e]);
      }
    } else {
      int n = tid >> 2;
      int mm = (tid & 3) * 8;
      int gn = n0 + n;
      bf16 v[8];
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        int gm = m0 + mm + e;
        v[e] = (gn < N && gm < M) ? Zg[(size_t)gm * N + gn] : f2bf(0.f);
        if (k0 == 0) db_part += bf2f(v[e]);
      }
      lds_store8(&Zs[n * LDT + mm], v);
    }
    __syncthreads();

    #pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &As[(wk + fm * 16 + l15) * LDT + mslot * 8]);
      #pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &Zs[(wn + fn * 16 + l15) * LDT + mslot * 8]);
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, acc[fm][fn], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  if (k0 == 0) {
    int gn = n0 + (tid >> 2);
    if (gn < N) atomicAdd(&db[(size_t)g * N + gn], db_part);
  }
  const bool single_chunk = gridDim.y == 1;
  #pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
    #pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      int col = n0 + wn + fn * 16 + l15;
      if (col >= N) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = k0 + wk + fm * 16 + (lane >> 4) * 4 + r;
        if (row >= K) continue;
        if (single_chunk)
          Wg[(size_t)row * N + col] = acc[fm][fn][r];
        else
          atomicAdd(&Wg[(size_t)row * N + col], acc[fm][fn][r]);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Fused pointwise kernels
// ---------------------------------------------------------------------------

// dZ = (dA + l1*sign(Y)) * act'(Y)
// ---------------------------------------------------------------------------
// K7 — sliding-window featurizer: gather lookback
// windows [G, B, T, F] straight out of the resident series [G, N, F].
// One workgroup per (g, window); lanes stream the T*F window elements
// with 16-byte loads when the feature row is 8-aligned (bf16 x 8).
// The device analog of create_keras_timeseriesgenerator
// (reference gordo/machine/model/models.py:713-793).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void window_gather_kernel(
    const bf16* __restrict__ X, const int* __restrict__ idx,
    bf16* __restrict__ out, int N, int F, int B, int T) {
  const int g = blockIdx.x / B;
  const int b = blockIdx.x % B;
  const int start = idx[(size_t)g * B + b];
  const bf16* src = X + ((size_t)g * N + start) * F;
  bf16* dst = out + (((size_t)g * B + b) * T) * F;
  const size_t n = (size_t)T * F;
  const int tid = threadIdx.x;
  if ((F & 7) == 0) {
    const bf16x8* s8 = reinterpret_cast<const bf16x8*>(src);
    bf16x8* d8 = reinterpret_cast<bf16x8*>(dst);
    for (size_t i = tid; i < n / 8; i += 256) d8[i] = s8[i];
  } else {
    for (size_t i = tid; i < n; i += 256) dst[i] = src[i];
  }
}

__global__ void act_l1_bwd_kernel(const bf16* __restrict__ dA,
                                  const bf16* __restrict__ Y,
                                  bf16* __restrict__ dZ, size_t n, int act,
                                  float l1) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float y = bf2f(Y[i]);
    float gin = bf2f(dA[i]);
    if (l1 != 0.f) gin += l1 * (y > 0.f ? 1.f : (y < 0.f ? -1.f : 0.f));
    dZ[i] = f2bf(gin * act_grad_from_output(y, act));
  }
}

// loss[g] = mean((Y-T)^2) over (B*F); dY = 2(Y-T)/(B*F)
// real_n: divisor for the mean — the count of REAL elements when the
// feature dim is zero-padded to 8 (pad diffs are exactly 0, so only
// the normalization changes; see engine/pack.py padding notes).
__global__ void mse_bwd_kernel(const bf16* __restrict__ Y,
                               const bf16* __restrict__ T,
                               bf16* __restrict__ dY,
                               float* __restrict__ loss, int per_g,
                               int real_n) {
  int g = blockIdx.y;
  const bf16* Yg = Y + (size_t)g * per_g;
  const bf16* Tg = T + (size_t)g * per_g;
  bf16* dYg = dY + (size_t)g * per_g;
  float inv_n = 1.f / (float)real_n;
  float local = 0.f;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < (size_t)per_g;
       i += (size_t)gridDim.x * blockDim.x) {
    float d = bf2f(Yg[i]) - bf2f(Tg[i]);
    local += d * d;
    dYg[i] = f2bf(2.f * d * inv_n);
  }
  // wave -> block -> global reduction
  #pragma unroll
  for (int off = 32; off; off >>= 1) local += __shfl_down(local, off, 64);
  __shared__ float warp_part[4];
  int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) warp_part[wid] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) s += warp_part[w];
    atomicAdd(&loss[g], s * inv_n);
  }
}

// fused Adam over the flat fp32 master buffer + bf16 mirror refresh.
// The step counter lives ON DEVICE (step_buf) so the kernel sequence is
// hipGraph-capturable: bias correction is recomputed from memory at
// every replay instead of being frozen into the captured kernel args.
__global__ void adam_bump_kernel(int* step_buf) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *step_buf += 1;
}

__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            bf16* __restrict__ plp, size_t n, float lr,
                            float b1, float b2, float eps,
                            const int* __restrict__ step_buf) {
  float fstep = (float)*step_buf;
  float bc1 = 1.f - __powf(b1, fstep);
  float bc2 = 1.f - __powf(b2, fstep);
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float gi = g[i];
    float mi = b1 * m[i] + (1.f - b1) * gi;
    float vi = b2 * v[i] + (1.f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    float pi = p[i] - lr / bc1 * mi / (sqrtf(vi / bc2) + eps);
    p[i] = pi;
    if (plp) plp[i] = f2bf(pi);
  }
}

// LSTM pointwise fwd: gates[g,b,4H] pre-act (+c_prev f32) ->
// h bf16, c f32, gact bf16 (keras gate order i,f,g,o).
__global__ void lstm_pw_fwd_kernel(const bf16* __restrict__ gates,
                                   const float* __restrict__ c_prev,
                                   bf16* __restrict__ h, float* __restrict__ c,
                                   bf16* __restrict__ gact, size_t rows,
                                   int H) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t n = rows * H;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; idx < n; idx += stride) {
    size_t row = idx / H;
    int hh = idx % H;
    const bf16* grow = gates + row * 4 * H;
    float i_g = 1.f / (1.f + __expf(-bf2f(grow[hh])));
    float f_g = 1.f / (1.f + __expf(-bf2f(grow[H + hh])));
    float g_g = fast_tanhf(bf2f(grow[2 * H + hh]));
    float o_g = 1.f / (1.f + __expf(-bf2f(grow[3 * H + hh])));
    float cc = f_g * c_prev[idx] + i_g * g_g;
    c[idx] = cc;
    h[idx] = f2bf(o_g * fast_tanhf(cc));
    bf16* ga = gact + row * 4 * H;
    ga[hh] = f2bf(i_g);
    ga[H + hh] = f2bf(f_g);
    ga[2 * H + hh] = f2bf(g_g);
    ga[3 * H + hh] = f2bf(o_g);
  }
}

// LSTM pointwise bwd -> dgates (pre-act) bf16, dc_prev f32
__global__ void lstm_pw_bwd_kernel(
    const bf16* __restrict__ dh, const float* __restrict__ dc_next,
    const bf16* __restrict__ gact, const float* __restrict__ c,
    const float* __restrict__ c_prev, bf16* __restrict__ dgates,
    float* __restrict__ dc_prev, size_t rows, int H) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t n = rows * H;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; idx < n; idx += stride) {
    size_t row = idx / H;
    int hh = idx % H;
    const bf16* ga = gact + row * 4 * H;
    float i_g = bf2f(ga[hh]);
    float f_g = bf2f(ga[H + hh]);
    float g_g = bf2f(ga[2 * H + hh]);
    float o_g = bf2f(ga[3 * H + hh]);
    float tc = fast_tanhf(c[idx]);
    float dhv = bf2f(dh[idx]);
    float dc = dc_next[idx] + dhv * o_g * (1.f - tc * tc);
    float di = dc * g_g;
    float df = dc * c_prev[idx];
    float dg = dc * i_g;
    float do_ = dhv * tc;
    dc_prev[idx] = dc * f_g;
    bf16* dgrow = dgates + row * 4 * H;
    dgrow[hh] = f2bf(di * i_g * (1.f - i_g));
    dgrow[H + hh] = f2bf(df * f_g * (1.f - f_g));
    dgrow[2 * H + hh] = f2bf(dg * (1.f - g_g * g_g));
    dgrow[3 * H + hh] = f2bf(do_ * o_g * (1.f - o_g));
  }
}

// ---------------------------------------------------------------------------
// Host-side launchers (torch extension API)
// ---------------------------------------------------------------------------
namespace {

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")

inline int ceil_div(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

torch::Tensor to_bf16c(const torch::Tensor& t) {
  return t.to(torch::kBFloat16).contiguous();
}

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

torch::Tensor grouped_linear_fwd(torch::Tensor X, torch::Tensor W,
                                 torch::Tensor b, int64_t act) {
  CHECK_GPU(X);
  auto Xc = to_bf16c(X);
  auto Wc = to_bf16c(W);
  auto bc = b.to(torch::kFloat32).contiguous();
  int G = Xc.size(0), M = Xc.size(1), K = Xc.size(2), N = Wc.size(2);
  TORCH_CHECK(Wc.size(1) == K, "W K mismatch");
  auto Y = torch::empty({G, M, N}, Xc.options());
  int mt = ceil_div(M, BM), nt = ceil_div(N, BN);
  int nblocks = G * mt * nt;
  hipLaunchKernelGGL((grouped_gemm_kernel<0, false, true>), dim3(nblocks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)Xc.data_ptr(), (const bf16*)Wc.data_ptr(),
                     bc.data_ptr<float>(), (bf16*)Y.data_ptr(), M, N, K,
                     (int)act, mt, nt, nblocks);
  return Y;
}

torch::Tensor grouped_linear_bwd_data(torch::Tensor dZ, torch::Tensor W) {
  CHECK_GPU(dZ);
  auto Zc = to_bf16c(dZ);
  auto Wc = to_bf16c(W);
  int G = Zc.size(0), M = Zc.size(1), N = Zc.size(2);
  int Kout = Wc.size(1);
  TORCH_CHECK(Wc.size(2) == N, "W N mismatch");
  auto dX = torch::empty({G, M, Kout}, Zc.options());
  // dX = dZ @ W^T: output cols = Kout, inner dim = N; W[Kout][N] = BMODE_NK
  int mt = ceil_div(M, BM), nt = ceil_div(Kout, BN);
  int nblocks = G * mt * nt;
  hipLaunchKernelGGL((grouped_gemm_kernel<1, false, false>), dim3(nblocks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)Zc.data_ptr(), (const bf16*)Wc.data_ptr(),
                     nullptr, (bf16*)dX.data_ptr(), M, Kout, N, ACT_LINEAR,
                     mt, nt, nblocks);
  return dX;
}

torch::Tensor grouped_gemm_acc(torch::Tensor A, torch::Tensor B,
                               torch::Tensor C) {
  CHECK_GPU(A);
  auto Ac = to_bf16c(A);
  auto Bc = to_bf16c(B);
  TORCH_CHECK(C.is_contiguous() && C.scalar_type() == torch::kBFloat16,
              "C must be contiguous bf16");
  int G = Ac.size(0), M = Ac.size(1), K = Ac.size(2), N = Bc.size(2);
  int mt = ceil_div(M, BM), nt = ceil_div(N, BN);
  int nblocks = G * mt * nt;
  hipLaunchKernelGGL((grouped_gemm_kernel<0, true, false>), dim3(nblocks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)Ac.data_ptr(), (const bf16*)Bc.data_ptr(),
                     nullptr, (bf16*)C.data_ptr(), M, N, K, ACT_LINEAR, mt,
                     nt, nblocks);
  return C;
}

std::vector<torch::Tensor> grouped_linear_wgrad_impl(torch::Tensor X,
                                                     torch::Tensor dZ,
                                                     int64_t tshiftT) {
  CHECK_GPU(X);
  auto Xc = to_bf16c(X);
  auto Zc = to_bf16c(dZ);
  int G = Xc.size(0), M = Zc.size(1), K = Xc.size(-1), N = Zc.size(2);
  if (tshiftT > 0) {
    TORCH_CHECK(M % tshiftT == 0, "rows must be a multiple of T");
    TORCH_CHECK(Xc.numel() == (int64_t)G * M * K, "X/dZ row mismatch");
  }
  int kt = ceil_div(K, BM), nt = ceil_div(N, BN);
  int nblocks = G * kt * nt;
  // choose the M split so the grid comfortably fills 256 CUs
  int target_chunks = std::max(1, 2048 / std::max(nblocks, 1));
  int mchunk = std::max(BK, (int)(((M + target_chunks - 1) / target_chunks
                                   + BK - 1) / BK * BK));
  int nchunks = ceil_div(M, mchunk);
  auto opts = Xc.options().dtype(torch::kFloat32);
  auto dW = nchunks > 1 ? torch::zeros({G, K, N}, opts)
                        : torch::empty({G, K, N}, opts);
  auto db = torch::zeros({G, N}, opts);  // always atomically accumulated
  hipLaunchKernelGGL(grouped_wgrad_kernel, dim3(nblocks, nchunks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)Xc.data_ptr(), (const bf16*)Zc.data_ptr(),
                     dW.data_ptr<float>(), db.data_ptr<float>(), M, N, K,
                     kt, nt, nblocks, mchunk, (int)tshiftT, nullptr, K);
  return {dW, db};
}

std::vector<torch::Tensor> grouped_wgrad_xh(torch::Tensor seq,
                                            torch::Tensor hs,
                                            torch::Tensor dZ, int64_t T) {
  // Combined recurrent weight-grad: {dWx, dWh, db} in one pass —
  // seq [G,M,F] (layer input rows), hs [G,B,T,H] (h rows, read with
  // the in-kernel t-1 shift), dZ [G,M,4H].
  CHECK_GPU(seq);
  auto Sc = to_bf16c(seq);
  auto Hc = to_bf16c(hs.reshape({hs.size(0), -1, hs.size(-1)}));
  auto Zc = to_bf16c(dZ);
  int G = Sc.size(0), M = Zc.size(1), F = Sc.size(2), H = Hc.size(2);
  int N = Zc.size(2);
  TORCH_CHECK(M % T == 0 && Hc.size(1) == M, "row/T mismatch");
  int K = F + H;
  int kt = ceil_div(K, BM), nt = ceil_div(N, BN);
  int nblocks = G * kt * nt;
  int target_chunks = std::max(1, 2048 / std::max(nblocks, 1));
  int mchunk = std::max(BK, (int)(((M + target_chunks - 1) / target_chunks
                                   + BK - 1) / BK * BK));
  int nchunks = ceil_div(M, mchunk);
  auto opts = Sc.options().dtype(torch::kFloat32);
  auto dW = nchunks > 1 ? torch::zeros({G, K, N}, opts)
                        : torch::empty({G, K, N}, opts);
  auto db = torch::zeros({G, N}, opts);
  hipLaunchKernelGGL(grouped_wgrad_kernel, dim3(nblocks, nchunks),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)Sc.data_ptr(), (const bf16*)Zc.data_ptr(),
                     dW.data_ptr<float>(), db.data_ptr<float>(), M, N, K,
                     kt, nt, nblocks, mchunk, (int)T,
                     (const bf16*)Hc.data_ptr(), F);
  return {dW.narrow(1, 0, F), dW.narrow(1, F, H), db};
}

std::vector<torch::Tensor> grouped_linear_wgrad(torch::Tensor X,
                                                torch::Tensor dZ) {
  return grouped_linear_wgrad_impl(X, dZ, 0);
}

std::vector<torch::Tensor> grouped_linear_wgrad_hprev(torch::Tensor hs,
                                                      torch::Tensor dZ,
                                                      int64_t T) {
  // hs: [G, B, T, H] (flattened rows b*T+t); dZ: [G, B*T, 4H].
  // Reads h_{t-1} per row in-kernel — no h_prev_all materialization.
  return grouped_linear_wgrad_impl(hs.reshape({hs.size(0), -1, hs.size(-1)}),
                                   dZ, T);
}

torch::Tensor act_l1_bwd(torch::Tensor dA, torch::Tensor Y, int64_t act,
                         double l1) {
  CHECK_GPU(dA);
  auto dAc = to_bf16c(dA);
  auto Yc = to_bf16c(Y);
  auto dZ = torch::empty_like(dAc);
  size_t n = dAc.numel();
  int blocks = (int)std::min<size_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(act_l1_bwd_kernel, dim3(blocks), dim3(256), 0,
                     cur_stream(), (const bf16*)dAc.data_ptr(),
                     (const bf16*)Yc.data_ptr(), (bf16*)dZ.data_ptr(), n,
                     (int)act, (float)l1);
  return dZ;
}

torch::Tensor window_gather(torch::Tensor X, torch::Tensor idx,
                            int64_t T) {
  // X: [G, N, F] (bf16/fp-castable), idx: [G, B] int32 window starts.
  CHECK_GPU(X);
  auto Xc = to_bf16c(X);
  auto ic = idx.to(torch::kInt32).contiguous();
  int G = Xc.size(0), N = Xc.size(1), F = Xc.size(2);
  int B = ic.size(1);
  auto out = torch::empty({G, B, T, F}, Xc.options());
  hipLaunchKernelGGL(window_gather_kernel, dim3((unsigned)G * B),
                     dim3(256), 0, cur_stream(),
                     (const bf16*)Xc.data_ptr(), ic.data_ptr<int>(),
                     (bf16*)out.data_ptr(), N, F, B, (int)T);
  return out;
}

std::vector<torch::Tensor> mse_bwd(torch::Tensor Y, torch::Tensor T,
                                   int64_t real_n) {
  CHECK_GPU(Y);
  auto Yc = to_bf16c(Y);
  auto Tc = to_bf16c(T);
  int G = Yc.size(0);
  int64_t per_g = Yc.numel() / G;
  if (real_n <= 0) real_n = per_g;
  auto dY = torch::empty_like(Yc);
  auto loss = torch::zeros({G}, Yc.options().dtype(torch::kFloat32));
  int blocks = (int)std::min<int64_t>((per_g + 255) / 256, 512);
  hipLaunchKernelGGL(mse_bwd_kernel, dim3(blocks, G), dim3(256), 0,
                     cur_stream(), (const bf16*)Yc.data_ptr(),
                     (const bf16*)Tc.data_ptr(), (bf16*)dY.data_ptr(),
                     loss.data_ptr<float>(), (int)per_g, (int)real_n);
  return {loss, dY};
}

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, double lr, double b1, double b2, double eps,
               int64_t step, c10::optional<torch::Tensor> plp,
               torch::Tensor step_buf) {
  CHECK_GPU(p);
  TORCH_CHECK(step_buf.is_cuda() && step_buf.scalar_type() == torch::kInt32,
              "step_buf must be an int32 GPU tensor");
  size_t n = p.numel();
  bf16* plp_ptr = nullptr;
  if (plp.has_value()) plp_ptr = (bf16*)plp->data_ptr();
  int blocks = (int)std::min<size_t>((n + 255) / 256, 4096);
  hipLaunchKernelGGL(adam_bump_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     step_buf.data_ptr<int>());
  hipLaunchKernelGGL(adam_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), plp_ptr, n,
                     (float)lr, (float)b1, (float)b2, (float)eps,
                     step_buf.data_ptr<int>());
}

std::vector<torch::Tensor> lstm_pointwise_fwd(torch::Tensor gates,
                                              torch::Tensor c_prev) {
  CHECK_GPU(gates);
  auto gc = to_bf16c(gates);
  auto cp = c_prev.to(torch::kFloat32).contiguous();
  int H4 = gc.size(-1);
  int H = H4 / 4;
  size_t rows = gc.numel() / H4;
  auto sizes = gc.sizes().vec();
  sizes.back() = H;
  auto h = torch::empty(sizes, gc.options());
  auto c = torch::empty(sizes, gc.options().dtype(torch::kFloat32));
  auto gact = torch::empty_like(gc);
  size_t n = rows * H;
  int blocks = (int)std::min<size_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(lstm_pw_fwd_kernel, dim3(blocks), dim3(256), 0,
                     cur_stream(), (const bf16*)gc.data_ptr(),
                     cp.data_ptr<float>(), (bf16*)h.data_ptr(),
                     c.data_ptr<float>(), (bf16*)gact.data_ptr(), rows, H);
  return {h, c, gact};
}

std::vector<torch::Tensor> lstm_pointwise_bwd(torch::Tensor dh,
                                              torch::Tensor dc_next,
                                              torch::Tensor gact,
                                              torch::Tensor c,
                                              torch::Tensor c_prev) {
  CHECK_GPU(dh);
  auto dhc = to_bf16c(dh);
  auto dcn = dc_next.to(torch::kFloat32).contiguous();
  auto gc = to_bf16c(gact);
  auto cc = c.to(torch::kFloat32).contiguous();
  auto cpc = c_prev.to(torch::kFloat32).contiguous();
  int H = dhc.size(-1);
  size_t rows = dhc.numel() / H;
  auto dgates = torch::empty_like(gc);
  auto dc_prev = torch::empty_like(cc);
  size_t n = rows * H;
  int blocks = (int)std::min<size_t>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(lstm_pw_bwd_kernel, dim3(blocks), dim3(256), 0,
                     cur_stream(), (const bf16*)dhc.data_ptr(),
                     dcn.data_ptr<float>(), (const bf16*)gc.data_ptr(),
                     cc.data_ptr<float>(), cpc.data_ptr<float>(),
                     (bf16*)dgates.data_ptr(), dc_prev.data_ptr<float>(),
                     rows, H);
  return {dgates, dc_prev};
}

}  // namespace

// fused anomaly scoring entry point (anomaly.hip)
namespace gordo_anomaly {
std::vector<torch::Tensor> anomaly_score(
    torch::Tensor out, torch::Tensor y, torch::Tensor scale,
    torch::Tensor minv, c10::optional<torch::Tensor> feat_thr,
    double agg_thr);
}  // namespace gordo_anomaly

// fused LSTM sequence-scan entry points (lstm_seq.hip)
namespace gordo_lstm {
std::vector<torch::Tensor> lstm_seq_fwd(torch::Tensor xW, torch::Tensor Wh);
std::vector<torch::Tensor> lstm_seq_fwd_fused(torch::Tensor xseq,
                                              torch::Tensor Wx,
                                              torch::Tensor Wh,
                                              torch::Tensor bias,
                                              bool store_aux);
std::vector<torch::Tensor> lstm_seq_fwd_v3(torch::Tensor xW, torch::Tensor Wh);
torch::Tensor lstm_seq_bwd_v3(torch::Tensor dSeq, torch::Tensor gacts,
                              torch::Tensor cs, torch::Tensor Wh,
                              bool last_only);
std::vector<torch::Tensor> lstm_seq_bwd_fused(
    torch::Tensor dSeq, torch::Tensor gacts, torch::Tensor cs,
    torch::Tensor Wh, torch::Tensor Wx, bool last_only);
torch::Tensor lstm_seq_bwd(torch::Tensor dSeq, torch::Tensor gacts,
                           torch::Tensor cs, torch::Tensor Wh,
                           bool last_only);
}  // namespace gordo_lstm

// device threshold/statistics kernels (thresholds.hip: K10-K12)
namespace gordo_thresholds {
torch::Tensor trail_min_max(torch::Tensor X, int64_t w);
torch::Tensor windowed_quantile(torch::Tensor X, int64_t w, double q);
torch::Tensor row_quantile(torch::Tensor X, double q);
}  // namespace gordo_thresholds

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("trail_min_max", &gordo_thresholds::trail_min_max,
          "rolling(w).min().max() per row (K10)");
  mod.def("windowed_quantile", &gordo_thresholds::windowed_quantile,
          "rolling(w).quantile(q) per row — q=0.5 is the smm smoothing "
          "(K11)");
  mod.def("row_quantile", &gordo_thresholds::row_quantile,
          "NaN-dropping linear-interpolated quantile per row (K12)");
  mod.def("anomaly_score", &gordo_anomaly::anomaly_score,
          "fused DiffBased anomaly scoring (serving hot path)");
  mod.def("lstm_seq_bwd_v3", &gordo_lstm::lstm_seq_bwd_v3,
          "pipelined fused LSTM backward scan (GORDO_LSTM_V3 opt-in)");
  mod.def("lstm_seq_fwd_v3", &gordo_lstm::lstm_seq_fwd_v3,
          "pipelined fused LSTM forward scan (GORDO_LSTM_V3 opt-in)");
  mod.def("lstm_seq_fwd", &gordo_lstm::lstm_seq_fwd,
          "fused LSTM forward sequence scan (Wh resident in LDS)");
  mod.def("lstm_seq_fwd_fused", &gordo_lstm::lstm_seq_fwd_fused,
          "v4 scan with the x-side gate GEMM fused in (halves fwd "
          "HBM traffic); store_aux=false skips cs/gacts (inference)");
  mod.def("lstm_seq_bwd", &gordo_lstm::lstm_seq_bwd,
          "fused LSTM backward (BPTT) sequence scan");
  mod.def("lstm_seq_bwd_fused", &gordo_lstm::lstm_seq_bwd_fused,
          "v5 reverse scan with dSeq = dG@Wx^T fused in");
  mod.def("grouped_linear_fwd", &grouped_linear_fwd,
          "Y = act(X@W + b) per group (MFMA)");
  mod.def("grouped_linear_bwd_data", &grouped_linear_bwd_data,
          "dX = dZ@W^T per group (MFMA)");
  mod.def("grouped_linear_wgrad", &grouped_linear_wgrad,
          "dW = X^T@dZ, db = colsum(dZ) per group (MFMA)");
  mod.def("grouped_linear_wgrad_hprev", &grouped_linear_wgrad_hprev,
          "dWh = h_prev^T@dG with in-kernel t-1 shift (no concat)");
  mod.def("grouped_wgrad_xh", &grouped_wgrad_xh,
          "combined {dWx, dWh, db} — dZ staged once for both");
  mod.def("window_gather", &window_gather,
          "K7 sliding-window featurizer: [G,N,F] -> [G,B,T,F]");
  mod.def("grouped_gemm_acc", &grouped_gemm_acc, "C += A@B per group (MFMA)");
  mod.def("act_l1_bwd", &act_l1_bwd, "fused activation+L1 backward");
  mod.def("mse_bwd", &mse_bwd, py::arg("Y"), py::arg("T"),
          py::arg("real_n") = -1,
          "fused per-model MSE loss + grad (real_n: divisor when the "
          "feature dim is zero-padded)");
  mod.def("adam_step", &adam_step, "fused Adam + bf16 mirror refresh");
  mod.def("lstm_pointwise_fwd", &lstm_pointwise_fwd, "LSTM cell fwd");
  mod.def("lstm_pointwise_bwd", &lstm_pointwise_bwd, "LSTM cell bwd");
}
