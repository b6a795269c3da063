// Fused LSTM sequence-scan kernels for MI355X (gfx950) — v2.
//
// Key observation: the LSTM recurrence is ROW-INDEPENDENT — window row
// b's (h, c) never reads another row's state. So each 64-lane wave
// owns 16 window rows for the ENTIRE sequence and keeps its state in
// wave-private LDS patches: no __syncthreads anywhere in the T loop
// (v1 paid 2 block barriers x 144 steps). Waves on a CU make
// independent progress, hiding each other's MFMA/LDS/HBM latencies.
//
// Per wave, per timestep t:
//   1. gates[16, 4H] = h[16, H] @ Wh   — MFMA 16x16x32, A-fragments
//      read straight from the wave's h patch (row-major [16][LDK],
//      16-lane-group conflict-free b128 reads).
//   2. epilogue adds the precomputed x-side gates xW[.., t, ..]
//      (one grouped MFMA GEMM over all B*T rows, gordo_kernels.hip)
//      and stores to the wave's gate patch.
//   3. fused sigmoid/tanh gate math updates the h/c patches and writes
//      h_t / c_t / activated gates to HBM for BPTT.
// Wh itself is staged once per workgroup into shared LDS (transposed,
// so B-fragment reads are row-contiguous); the stage barrier is the
// only barrier in the kernel. Backward is the mirror image
// (reverse-time, dgates @ Wh^T carry).
//
// Geometry: H <= 64 (4H <= 256); the engine falls back to the
// per-timestep kernels beyond that.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

#define DEV_INLINE __device__ __forceinline__

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

DEV_INLINE float lbf2f(bf16 v) { return __bfloat162float(v); }
DEV_INLINE bf16 lf2bf(float v) { return __float2bfloat16(v); }
DEV_INLINE float sigmoidf_(float x) { return 1.f / (1.f + __expf(-x)); }

constexpr int WROWS = 16;   // rows per wave
constexpr int NWAVES = 4;   // waves per workgroup
constexpr int ROWS = WROWS * NWAVES;  // rows per workgroup
constexpr int LDK = 72;     // padded row length for h patches (bf16)

// ---------------------------------------------------------------------------
// Forward scan.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void lstm_seq_fwd_kernel(
    const bf16* __restrict__ xW, const bf16* __restrict__ Wh,
    bf16* __restrict__ hs, float* __restrict__ cs,
    bf16* __restrict__ gacts, int B, int T, int H, int ldg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  bf16* WhT = reinterpret_cast<bf16*>(smem);  // [H4][LDK] shared
  // per-wave private patches
  bf16* hP_all = WhT + (size_t)H4 * LDK;                  // [NWAVES][16][LDK]
  bf16* gP_all = hP_all + (size_t)NWAVES * WROWS * LDK;   // [NWAVES][16][ldg]
  float* cP_all =
      reinterpret_cast<float*>(gP_all + (size_t)NWAVES * WROWS * ldg);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  bf16* hP = hP_all + (size_t)wid * WROWS * LDK;
  bf16* gP = gP_all + (size_t)wid * WROWS * ldg;
  float* cP = cP_all + (size_t)wid * WROWS * H;

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS + wid * WROWS;
  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* xWg = xW + ((size_t)g * B + r0) * T * H4;
  bf16* hsg = hs + ((size_t)g * B + r0) * T * H;
  float* csg = cs + ((size_t)g * B + r0) * T * H;
  bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const int rows_here = max(0, min(WROWS, B - r0));

  // one-time: stage WhT[n][k=h] (transposed, zero-padded) — all threads
  for (int i = tid; i < H4 * LDK; i += 256) {
    int n = i / LDK, k = i % LDK;
    WhT[i] = (k < H) ? Whg[(size_t)k * H4 + n] : lf2bf(0.f);
  }
  // wave-private init (own patch only)
  for (int i = lane; i < WROWS * LDK; i += 64) hP[i] = lf2bf(0.f);
  for (int i = lane; i < WROWS * H; i += 64) cP[i] = 0.f;
  __syncthreads();  // WhT ready — the only block barrier

  const int FN = (H4 + 15) / 16;
  const int KK = (H + 31) / 32;

  for (int t = 0; t < T; ++t) {
    // 1. MFMA: gates = h @ Wh. A-fragment: lane l -> row l15, k-slice
    //    kslot*8..+8 per K-step (loaded once per t).
    bf16x8 afrag[2];
    #pragma unroll 2
    for (int kk = 0; kk < KK; ++kk)
      afrag[kk] = *reinterpret_cast<const bf16x8*>(
          &hP[l15 * LDK + kk * 32 + kslot * 8]);

    for (int fn = 0; fn < FN; ++fn) {
      int col = fn * 16 + l15;
      int bcol = min(col, H4 - 1);
      f32x4 acc = {};
      #pragma unroll 2
      for (int kk = 0; kk < KK; ++kk) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &WhT[(size_t)bcol * LDK + kk * 32 + kslot * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kk], b, acc, 0,
                                                      0, 0);
      }
      // 2. epilogue: add xW_t, store to the wave's gate patch
      if (col < H4) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = (lane >> 4) * 4 + r;
          float xv = (row < rows_here)
                         ? lbf2f(xWg[((size_t)row * T + t) * H4 + col])
                         : 0.f;
          gP[row * ldg + col] = lf2bf(acc[r] + xv);
        }
      }
    }

    // 3. fused gate math + state update + HBM outputs (wave-local; the
    //    compiler's lgkmcnt ordering covers the gP RAW dependency)
    for (int e = lane; e < WROWS * H; e += 64) {
      int row = e / H, hh = e % H;
      const bf16* grow = &gP[row * ldg];
      float i_g = sigmoidf_(lbf2f(grow[hh]));
      float f_g = sigmoidf_(lbf2f(grow[H + hh]));
      float g_g = tanhf(lbf2f(grow[2 * H + hh]));
      float o_g = sigmoidf_(lbf2f(grow[3 * H + hh]));
      float cc = f_g * cP[row * H + hh] + i_g * g_g;
      float hv = o_g * tanhf(cc);
      cP[row * H + hh] = cc;
      hP[row * LDK + hh] = lf2bf(hv);
      if (row < rows_here) {
        size_t base = ((size_t)row * T + t) * H + hh;
        hsg[base] = lf2bf(hv);
        csg[base] = cc;
        size_t gbase = ((size_t)row * T + t) * H4;
        gag[gbase + hh] = lf2bf(i_g);
        gag[gbase + H + hh] = lf2bf(f_g);
        gag[gbase + 2 * H + hh] = lf2bf(g_g);
        gag[gbase + 3 * H + hh] = lf2bf(o_g);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward scan (BPTT), reverse time. dSeq is [G,B,T,H], or [G,B,H]
// with last_only (grads only on h_{T-1}).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void lstm_seq_bwd_kernel(
    const bf16* __restrict__ dSeq, const bf16* __restrict__ gacts,
    const float* __restrict__ cs, const bf16* __restrict__ Wh,
    bf16* __restrict__ dG, int B, int T, int H, int ldg, int last_only) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  bf16* WhN = reinterpret_cast<bf16*>(smem);  // [H][ldg] native, shared
  bf16* dgP_all = WhN + (size_t)H * ldg;                  // [NWAVES][16][ldg]
  bf16* dhP_all = dgP_all + (size_t)NWAVES * WROWS * ldg; // [NWAVES][16][LDK]
  float* dcP_all =
      reinterpret_cast<float*>(dhP_all + (size_t)NWAVES * WROWS * LDK);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  bf16* dgP = dgP_all + (size_t)wid * WROWS * ldg;
  bf16* dhP = dhP_all + (size_t)wid * WROWS * LDK;
  float* dcP = dcP_all + (size_t)wid * WROWS * H;

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS + wid * WROWS;
  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const float* csg = cs + ((size_t)g * B + r0) * T * H;
  const bf16* dSg = last_only ? dSeq + ((size_t)g * B + r0) * H
                              : dSeq + ((size_t)g * B + r0) * T * H;
  bf16* dGg = dG + ((size_t)g * B + r0) * T * H4;
  const int rows_here = max(0, min(WROWS, B - r0));

  for (int i = tid; i < H * ldg; i += 256) {
    int h = i / ldg, n = i % ldg;
    WhN[i] = (n < H4) ? Whg[(size_t)h * H4 + n] : lf2bf(0.f);
  }
  for (int i = lane; i < WROWS * LDK; i += 64) dhP[i] = lf2bf(0.f);
  for (int i = lane; i < WROWS * H; i += 64) dcP[i] = 0.f;
  // zero dgP once: gate backward writes only cols [0, 4H) but the MFMA
  // K loop reads through the padded columns.
  for (int i = lane; i < WROWS * ldg; i += 64) dgP[i] = lf2bf(0.f);
  __syncthreads();  // WhN ready

  const int KK4 = (H4 + 31) / 32;
  const int FNH = (H + 15) / 16;

  for (int t = T - 1; t >= 0; --t) {
    // 1. fused gate backward (wave-local)
    for (int e = lane; e < WROWS * H; e += 64) {
      int row = e / H, hh = e % H;
      float dh = lbf2f(dhP[row * LDK + hh]);
      if (row < rows_here) {
        if (last_only) {
          if (t == T - 1) dh += lbf2f(dSg[(size_t)row * H + hh]);
        } else {
          dh += lbf2f(dSg[((size_t)row * T + t) * H + hh]);
        }
      }
      size_t gbase = ((size_t)row * T + t) * H4;
      size_t cbase = ((size_t)row * T + t) * H + hh;
      float i_g = 0.5f, f_g = 0.5f, g_g = 0.f, o_g = 0.5f, cc = 0.f,
            cp = 0.f;
      if (row < rows_here) {
        i_g = lbf2f(gag[gbase + hh]);
        f_g = lbf2f(gag[gbase + H + hh]);
        g_g = lbf2f(gag[gbase + 2 * H + hh]);
        o_g = lbf2f(gag[gbase + 3 * H + hh]);
        cc = csg[cbase];
        cp = (t > 0) ? csg[cbase - H] : 0.f;
      }
      float tc = tanhf(cc);
      float dc = dcP[row * H + hh] + dh * o_g * (1.f - tc * tc);
      float di = dc * g_g;
      float df = dc * cp;
      float dg = dc * i_g;
      float do_ = dh * tc;
      dcP[row * H + hh] = dc * f_g;
      float vi = di * i_g * (1.f - i_g);
      float vf = df * f_g * (1.f - f_g);
      float vg = dg * (1.f - g_g * g_g);
      float vo = do_ * o_g * (1.f - o_g);
      dgP[row * ldg + hh] = lf2bf(vi);
      dgP[row * ldg + H + hh] = lf2bf(vf);
      dgP[row * ldg + 2 * H + hh] = lf2bf(vg);
      dgP[row * ldg + 3 * H + hh] = lf2bf(vo);
      if (row < rows_here) {
        dGg[gbase + hh] = lf2bf(vi);
        dGg[gbase + H + hh] = lf2bf(vf);
        dGg[gbase + 2 * H + hh] = lf2bf(vg);
        dGg[gbase + 3 * H + hh] = lf2bf(vo);
      }
    }

    // 2. dh_carry = dgates @ Wh^T (MFMA, wave-local)
    for (int fn = 0; fn < FNH; ++fn) {
      int col = fn * 16 + l15;  // h index
      int bcol = min(col, H - 1);
      f32x4 acc = {};
      for (int kk = 0; kk < KK4; ++kk) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &dgP[l15 * ldg + kk * 32 + kslot * 8]);
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &WhN[(size_t)bcol * ldg + kk * 32 + kslot * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      }
      if (col < H) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = (lane >> 4) * 4 + r;
          dhP[row * LDK + col] = lf2bf(acc[r]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
namespace gordo_lstm {

// pad 4H up to the MFMA K-step (32) plus 8: the K loop's last fragment
// read may touch columns [H4, ru32(H4)); they must exist and be zero.
inline int pad_ldg(int h4) { return ((h4 + 31) & ~31) + 8; }

std::vector<torch::Tensor> lstm_seq_fwd(torch::Tensor xW, torch::Tensor Wh) {
  TORCH_CHECK(xW.is_cuda() && xW.dim() == 4, "xW must be [G,B,T,4H] on GPU");
  auto xc = xW.to(torch::kBFloat16).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = xc.size(0), B = xc.size(1), T = xc.size(2), H4 = xc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(H <= 64, "lstm_seq_fwd supports H <= 64");
  int ldg = pad_ldg(H4);
  auto hs = torch::empty({G, B, T, H}, xc.options());
  auto cs = torch::empty({G, B, T, H}, xc.options().dtype(torch::kFloat32));
  auto gacts = torch::empty({G, B, T, H4}, xc.options());
  size_t lds = (size_t)H4 * LDK * 2 +
               (size_t)NWAVES * WROWS * (LDK * 2 + ldg * 2 + H * 4);
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded");
  int blocks = G * ((B + ROWS - 1) / ROWS);
  hipLaunchKernelGGL(lstm_seq_fwd_kernel, dim3(blocks), dim3(256), lds,
                     at::cuda::getCurrentCUDAStream().stream(),
                     (const bf16*)xc.data_ptr(), (const bf16*)Whc.data_ptr(),
                     (bf16*)hs.data_ptr(), cs.data_ptr<float>(),
                     (bf16*)gacts.data_ptr(), B, T, H, ldg);
  return {hs, cs, gacts};
}

torch::Tensor lstm_seq_bwd(torch::Tensor dSeq, torch::Tensor gacts,
                           torch::Tensor cs, torch::Tensor Wh,
                           bool last_only) {
  TORCH_CHECK(gacts.is_cuda() && gacts.dim() == 4, "gacts must be [G,B,T,4H]");
  auto dc = dSeq.to(torch::kBFloat16).contiguous();
  auto gc = gacts.to(torch::kBFloat16).contiguous();
  auto cc = cs.to(torch::kFloat32).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = gc.size(0), B = gc.size(1), T = gc.size(2), H4 = gc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(H <= 64, "lstm_seq_bwd supports H <= 64");
  int ldg = pad_ldg(H4);
  auto dG = torch::empty_like(gc);
  size_t lds = (size_t)H * ldg * 2 +
               (size_t)NWAVES * WROWS * (ldg * 2 + LDK * 2 + H * 4);
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded");
  int blocks = G * ((B + ROWS - 1) / ROWS);
  hipLaunchKernelGGL(lstm_seq_bwd_kernel, dim3(blocks), dim3(256), lds,
                     at::cuda::getCurrentCUDAStream().stream(),
                     (const bf16*)dc.data_ptr(), (const bf16*)gc.data_ptr(),
                     cc.data_ptr<float>(), (const bf16*)Whc.data_ptr(),
                     (bf16*)dG.data_ptr(), B, T, H, ldg, last_only ? 1 : 0);
  return dG;
}

}  // namespace gordo_lstm
