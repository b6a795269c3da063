// Fused LSTM sequence-scan kernels for MI355X (gfx950).
//
// The reference's per-timestep Keras LSTM maps
// to a sequential h/c recurrence that launch-per-timestep execution
// makes launch-bound (~576 launches per batch at lookback 144). These
// kernels run the ENTIRE sequence loop on-device:
//
//   lstm_seq_fwd : per (model g, 64-window row tile) workgroup, the
//     recurrent weight matrix Wh is staged TRANSPOSED into LDS once,
//     h/c live in LDS across timesteps, and each step is
//     {MFMA h@Wh -> gates (+= precomputed xW_t from HBM) -> fused
//     sigmoid/tanh gate math -> h/c update}, writing the h sequence,
//     c sequence and activated gates needed by BPTT.
//   lstm_seq_bwd : the reverse-time scan: fused gate backward +
//     MFMA dgates@Wh^T carry, emitting pre-activation gate grads for
//     the batched wgrad GEMMs.
//
// Supported geometry: H <= 64 (4H <= 256); the Python engine falls
// back to the per-timestep kernels beyond that. One workgroup = 4
// waves; the x-side GEMM (x@Wx + b over all B*T rows) stays a single
// big grouped MFMA GEMM in gordo_kernels.hip.

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
DEV_INLINE unsigned short bf16_bits(bf16 v) {
  return __builtin_bit_cast(unsigned short, v);
}
DEV_INLINE bf16 bits_bf16(unsigned short b) {
  return __builtin_bit_cast(bf16, b);
}
DEV_INLINE bf16 lf2bf(float v) { return __float2bfloat16(v); }
DEV_INLINE float sigmoidf_(float x) { return 1.f / (1.f + __expf(-x)); }
DEV_INLINE float fast_tanhf_(float x) {
  float e = __expf(2.f * x);
  return 1.f - 2.f / (e + 1.f);  // saturates to 1 when e overflows to inf
}

constexpr int LDK = 72;    // padded K-row length for h / WhT tiles (bf16)
// ROWS (window rows per workgroup) is a template parameter: 64 rows for
// big launches, 32 when the grid would underfill the 256 CUs (doubles
// the number of independent workgroups on the bench's G*ceil(B/64)
// ~ 250-workgroup shape).

// ---------------------------------------------------------------------------
// Forward scan.
//   xW    [G, B, T, 4H]  bf16 (x@Wx + b, precomputed)
//   Wh    [G, H, 4H]     bf16
//   hs    [G, B, T, H]   bf16 out
//   cs    [G, B, T, H]   f32  out
//   gacts [G, B, T, 4H]  bf16 out (activated i,f,g,o)
// grid.x = G * ceil(B/64); dynamic LDS.
// ---------------------------------------------------------------------------
template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_fwd_kernel(
    const bf16* __restrict__ xW, const bf16* __restrict__ Wh,
    bf16* __restrict__ hs, float* __restrict__ cs,
    bf16* __restrict__ gacts, int B, int T, int H, int ldg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  // carves (all 16B aligned: LDK and ldg are multiples of 8 bf16)
  bf16* WhT = reinterpret_cast<bf16*>(smem);              // [H4][LDK]
  bf16* hS = WhT + (size_t)H4 * LDK;                      // [ROWS][LDK]
  bf16* gS = hS + (size_t)ROWS * LDK;                     // [ROWS][ldg]
  float* cS = reinterpret_cast<float*>(gS + (size_t)ROWS * ldg);  // [ROWS][H]

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* xWg = xW + ((size_t)g * B + r0) * T * H4;
  bf16* hsg = hs + ((size_t)g * B + r0) * T * H;
  float* csg = cs + ((size_t)g * B + r0) * T * H;
  bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  // ---- one-time: stage WhT[n][k=h] (transposed, zero-padded) ----
  for (int i = tid; i < H4 * LDK; i += 256) {
    int n = i / LDK, k = i % LDK;
    WhT[i] = (k < H) ? Whg[(size_t)k * H4 + n] : lf2bf(0.f);
  }
  // zero h, c
  for (int i = tid; i < ROWS * LDK; i += 256) hS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) cS[i] = 0.f;
  __syncthreads();

  const int wcol0 = wid * 64;       // this wave's 64-column slice of 4H
  const bool wave_active = wcol0 < H4;

  for (int t = 0; t < T; ++t) {
    // ---- gates = h @ Wh (MFMA) + xW_t ----
    if (wave_active) {
      constexpr int FM = ROWS / 16;
      // prefetch this step's x-side gate values BEFORE the MFMA loop so
      // the HBM latency hides under the matrix work (the epilogue then
      // reads registers, not memory)
      bf16 xv[FM][4][4];
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + (lane >> 4) * 4 + r;
            xv[fm][fn][r] =
                (col < H4 && row < rows_here)
                    ? xWg[((size_t)row * T + t) * H4 + col]
                    : lf2bf(0.f);
          }
        }
      }
      f32x4 acc[FM][4] = {};
      for (int kk = 0; kk < H; kk += 32) {
        #pragma unroll
        for (int fm = 0; fm < FM; ++fm) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &hS[(fm * 16 + l15) * LDK + kk + kslot * 8]);
          #pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = wcol0 + fn * 16 + l15;
            bf16x8 b = *reinterpret_cast<const bf16x8*>(
                &WhT[(size_t)min(col, H4 - 1) * LDK + kk + kslot * 8]);
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[fm][fn], 0, 0, 0);
          }
        }
      }
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          if (col >= H4) continue;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + (lane >> 4) * 4 + r;
            gS[row * ldg + col] =
                lf2bf(acc[fm][fn][r] + lbf2f(xv[fm][fn][r]));
          }
        }
      }
    }
    __syncthreads();

    // ---- fused gate math + h/c update + outputs ----
    for (int e = tid; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      const bf16* grow = &gS[row * ldg];
      float i_g = sigmoidf_(lbf2f(grow[hh]));
      float f_g = sigmoidf_(lbf2f(grow[H + hh]));
      float g_g = fast_tanhf_(lbf2f(grow[2 * H + hh]));
      float o_g = sigmoidf_(lbf2f(grow[3 * H + hh]));
      float cc = f_g * cS[row * H + hh] + i_g * g_g;
      float hv = o_g * fast_tanhf_(cc);
      cS[row * H + hh] = cc;
      hS[row * LDK + hh] = lf2bf(hv);
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
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Backward scan (BPTT).
//   dSeq    [G, B, T, H] bf16 — upstream grads on the h sequence, or
//           when last_only: [G, B, H], grads only on h_{T-1}.
//   gacts   [G, B, T, 4H] bf16, cs [G, B, T, H] f32 (from forward)
//   Wh      [G, H, 4H] bf16
//   dG      [G, B, T, 4H] bf16 out — pre-activation gate grads.
// ---------------------------------------------------------------------------
template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_bwd_kernel(
    const bf16* __restrict__ dSeq, const bf16* __restrict__ gacts,
    const float* __restrict__ cs, const bf16* __restrict__ Wh,
    bf16* __restrict__ dG, int B, int T, int H, int ldg, int last_only) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  bf16* WhN = reinterpret_cast<bf16*>(smem);               // [H][ldg] native
  bf16* dgS = WhN + (size_t)H * ldg;                       // [ROWS][ldg]
  bf16* dhS = dgS + (size_t)ROWS * ldg;                    // [ROWS][LDK]
  float* dcS = reinterpret_cast<float*>(dhS + (size_t)ROWS * LDK);  // [ROWS][H]

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const float* csg = cs + ((size_t)g * B + r0) * T * H;
  const bf16* dSg = last_only ? dSeq + ((size_t)g * B + r0) * H
                              : dSeq + ((size_t)g * B + r0) * T * H;
  bf16* dGg = dG + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  // stage Wh native [H][4H] zero-padded to [H][ldg]
  for (int i = tid; i < H * ldg; i += 256) {
    int h = i / ldg, n = i % ldg;
    WhN[i] = (n < H4) ? Whg[(size_t)h * H4 + n] : lf2bf(0.f);
  }
  for (int i = tid; i < ROWS * LDK; i += 256) dhS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) dcS[i] = 0.f;
  // zero dgS once: the gate-backward phase writes only cols [0, 4H),
  // but the MFMA K loop reads through the padded columns.
  for (int i = tid; i < ROWS * ldg; i += 256) dgS[i] = lf2bf(0.f);
  __syncthreads();

  for (int t = T - 1; t >= 0; --t) {
    // ---- fused gate backward ----
    for (int e = tid; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      float dh = lbf2f(dhS[row * LDK + hh]);
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
      float tc = fast_tanhf_(cc);
      float dc = dcS[row * H + hh] + dh * o_g * (1.f - tc * tc);
      float di = dc * g_g;
      float df = dc * cp;
      float dg = dc * i_g;
      float do_ = dh * tc;
      dcS[row * H + hh] = dc * f_g;
      float vi = di * i_g * (1.f - i_g);
      float vf = df * f_g * (1.f - f_g);
      float vg = dg * (1.f - g_g * g_g);
      float vo = do_ * o_g * (1.f - o_g);
      dgS[row * ldg + hh] = lf2bf(vi);
      dgS[row * ldg + H + hh] = lf2bf(vf);
      dgS[row * ldg + 2 * H + hh] = lf2bf(vg);
      dgS[row * ldg + 3 * H + hh] = lf2bf(vo);
      if (row < rows_here) {
        dGg[gbase + hh] = lf2bf(vi);
        dGg[gbase + H + hh] = lf2bf(vf);
        dGg[gbase + 2 * H + hh] = lf2bf(vg);
        dGg[gbase + 3 * H + hh] = lf2bf(vo);
      }
    }
    // zero the LDS pad columns once is unnecessary: ldg cols >= H4 are
    // never written but also never read as MFMA K (K loop runs over H4).
    __syncthreads();

    // ---- dh_carry = dgates @ Wh^T (MFMA): out [ROWS][H] ----
    // wave w owns rows w*16..w*16+15 (fm = w), cols 0..63 (fn 0..3).
    if (wid < ROWS / 16) {
      f32x4 acc[4] = {};
      for (int kk = 0; kk < H4; kk += 32) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &dgS[(wid * 16 + l15) * ldg + kk + kslot * 8]);
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = fn * 16 + l15;  // h index
          bf16x8 b = *reinterpret_cast<const bf16x8*>(
              &WhN[(size_t)min(col, H - 1) * ldg + kk + kslot * 8]);
          acc[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[fn], 0, 0, 0);
        }
      }
      #pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = fn * 16 + l15;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = wid * 16 + (lane >> 4) * 4 + r;
          if (col < H) dhS[row * LDK + col] = lf2bf(acc[fn][r]);
        }
      }
    }
    __syncthreads();
  }
}


// ===========================================================================
// v5 backward scan with FUSED dSeq — the dG consumer GEMM
// dSeq_t = dG_t @ Wx^T runs inside the reverse scan (WxT LDS-resident,
// the carry MFMA's column range widened from H to H+F), so the
// 384 MB/layer dG re-read of the separate bwd-data GEMM disappears
// (the fleet's BPTT is bandwidth-bound like the forward — see the v4
// note). Structure is the v1 barriered scan; any H <= 64 (the v2/v3
// variants keep the non-fused path for the bottom layer, which needs
// no dSeq).
// ===========================================================================
template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_bwd_v5_kernel(
    const bf16* __restrict__ dSeq, const bf16* __restrict__ gacts,
    const float* __restrict__ cs, const bf16* __restrict__ Wh,
    const bf16* __restrict__ Wx,   // [G, F, 4H] native
    bf16* __restrict__ dG, bf16* __restrict__ dX,  // [G, B, T, F]
    int B, int T, int H, int F, int ldg, int last_only) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  bf16* WhN = reinterpret_cast<bf16*>(smem);               // [H][ldg]
  bf16* WxN = WhN + (size_t)H * ldg;                       // [F][ldg]
  bf16* dgS = WxN + (size_t)F * ldg;                       // [ROWS][ldg]
  bf16* dhS = dgS + (size_t)ROWS * ldg;                    // [ROWS][LDK]
  float* dcS = reinterpret_cast<float*>(dhS + (size_t)ROWS * LDK);

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* Wxg = Wx + (size_t)g * F * H4;
  const bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const float* csg = cs + ((size_t)g * B + r0) * T * H;
  const bf16* dSg = last_only ? dSeq + ((size_t)g * B + r0) * H
                              : dSeq + ((size_t)g * B + r0) * T * H;
  bf16* dGg = dG + ((size_t)g * B + r0) * T * H4;
  bf16* dXg = dX + ((size_t)g * B + r0) * T * F;
  const int rows_here = min(ROWS, B - r0);

  for (int i = tid; i < H * ldg; i += 256) {
    int h = i / ldg, n = i % ldg;
    WhN[i] = (n < H4) ? Whg[(size_t)h * H4 + n] : lf2bf(0.f);
  }
  for (int i = tid; i < F * ldg; i += 256) {
    int f = i / ldg, n = i % ldg;
    WxN[i] = (n < H4) ? Wxg[(size_t)f * H4 + n] : lf2bf(0.f);
  }
  for (int i = tid; i < ROWS * LDK; i += 256) dhS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) dcS[i] = 0.f;
  for (int i = tid; i < ROWS * ldg; i += 256) dgS[i] = lf2bf(0.f);
  __syncthreads();

  constexpr int FM = ROWS / 16;
  const int HF = H + F;

  // v3-style register prefetch of the next (earlier) step's gate
  // activations + cell states: their loads depend only on t, so they
  // issue before the MFMA carry phase and the HBM latency hides under
  // the matrix work (buffer capped at 8 slots — v3 note).
  constexpr int MAXE = (ROWS * 64 + 255) / 256;
  constexpr int PF = MAXE > 8 ? 8 : MAXE;
  unsigned int p_if[PF];
  unsigned int p_go[PF];
  float p_cc[PF];
  float p_cp[PF];

  auto prefetch_step = [&](int t) {
    #pragma unroll
    for (int s2 = 0; s2 < PF; ++s2) {
      int e = tid + s2 * 256;
      unsigned int v_if = 0x3F003F00u;
      unsigned int v_go = 0x3F000000u;
      float cc = 0.f, cp = 0.f;
      if (e < ROWS * H) {
        int row = e / H, hh = e % H;
        if (row < rows_here) {
          size_t gbase = ((size_t)row * T + t) * H4;
          v_if = (unsigned int)bf16_bits(gag[gbase + hh]) |
                 ((unsigned int)bf16_bits(gag[gbase + H + hh]) << 16);
          v_go = (unsigned int)bf16_bits(gag[gbase + 2 * H + hh]) |
                 ((unsigned int)bf16_bits(gag[gbase + 3 * H + hh]) << 16);
          size_t cbase = ((size_t)row * T + t) * H + hh;
          cc = csg[cbase];
          cp = (t > 0) ? csg[cbase - H] : 0.f;
        }
      }
      p_if[s2] = v_if;
      p_go[s2] = v_go;
      p_cc[s2] = cc;
      p_cp[s2] = cp;
    }
  };

  prefetch_step(T - 1);

  for (int t = T - 1; t >= 0; --t) {
    // ---- fused gate backward (prefetched slots + direct tail) ----
    auto gate_bwd = [&](int e, int t_, float i_g, float f_g, float g_g,
                        float o_g, float cc, float cp) {
      int row = e / H, hh = e % H;
      float dh = lbf2f(dhS[row * LDK + hh]);
      if (row < rows_here) {
        if (last_only) {
          if (t_ == T - 1) dh += lbf2f(dSg[(size_t)row * H + hh]);
        } else {
          dh += lbf2f(dSg[((size_t)row * T + t_) * H + hh]);
        }
      }
      float tc = fast_tanhf_(cc);
      float dc = dcS[row * H + hh] + dh * o_g * (1.f - tc * tc);
      float di = dc * g_g;
      float df = dc * cp;
      float dg = dc * i_g;
      float do_ = dh * tc;
      dcS[row * H + hh] = dc * f_g;
      float vi = di * i_g * (1.f - i_g);
      float vf = df * f_g * (1.f - f_g);
      float vg = dg * (1.f - g_g * g_g);
      float vo = do_ * o_g * (1.f - o_g);
      dgS[row * ldg + hh] = lf2bf(vi);
      dgS[row * ldg + H + hh] = lf2bf(vf);
      dgS[row * ldg + 2 * H + hh] = lf2bf(vg);
      dgS[row * ldg + 3 * H + hh] = lf2bf(vo);
      if (row < rows_here) {
        size_t gbase = ((size_t)row * T + t_) * H4;
        dGg[gbase + hh] = lf2bf(vi);
        dGg[gbase + H + hh] = lf2bf(vf);
        dGg[gbase + 2 * H + hh] = lf2bf(vg);
        dGg[gbase + 3 * H + hh] = lf2bf(vo);
      }
    };
    #pragma unroll
    for (int s2 = 0; s2 < PF; ++s2) {
      int e = tid + s2 * 256;
      if (e >= ROWS * H) break;
      gate_bwd(e, t,
               lbf2f(bits_bf16((unsigned short)p_if[s2])),
               lbf2f(bits_bf16((unsigned short)(p_if[s2] >> 16))),
               lbf2f(bits_bf16((unsigned short)p_go[s2])),
               lbf2f(bits_bf16((unsigned short)(p_go[s2] >> 16))),
               p_cc[s2], p_cp[s2]);
    }
    for (int e = tid + PF * 256; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      float i_g = 0.5f, f_g = 0.5f, g_g = 0.f, o_g = 0.5f, cc = 0.f,
            cp = 0.f;
      if (row < rows_here) {
        size_t gbase = ((size_t)row * T + t) * H4;
        i_g = lbf2f(gag[gbase + hh]);
        f_g = lbf2f(gag[gbase + H + hh]);
        g_g = lbf2f(gag[gbase + 2 * H + hh]);
        o_g = lbf2f(gag[gbase + 3 * H + hh]);
        size_t cbase = ((size_t)row * T + t) * H + hh;
        cc = csg[cbase];
        cp = (t > 0) ? csg[cbase - H] : 0.f;
      }
      gate_bwd(e, t, i_g, f_g, g_g, o_g, cc, cp);
    }
    if (t > 0) prefetch_step(t - 1);
    __syncthreads();

    // ---- [dh_carry | dSeq_t] = dgates @ [Wh ; Wx]^T ----
    for (int col0 = wid * 64; col0 < HF; col0 += 256) {
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        f32x4 acc[4] = {};
        for (int kk = 0; kk < H4; kk += 32) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &dgS[(fm * 16 + l15) * ldg + kk + kslot * 8]);
          #pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = col0 + fn * 16 + l15;
            int cc2 = min(col, HF - 1);
            const bf16* Brow =
                (cc2 < H) ? &WhN[(size_t)cc2 * ldg]
                          : &WxN[(size_t)(cc2 - H) * ldg];
            bf16x8 b = *reinterpret_cast<const bf16x8*>(
                &Brow[kk + kslot * 8]);
            acc[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[fn], 0, 0, 0);
          }
        }
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = col0 + fn * 16 + l15;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + kslot * 4 + r;
            if (col < H) {
              dhS[row * LDK + col] = lf2bf(acc[fn][r]);
            } else if (col < HF && row < rows_here) {
              dXg[((size_t)row * T + t) * F + (col - H)] =
                  lf2bf(acc[fn][r]);
            }
          }
        }
      }
    }
    __syncthreads();
  }
}

// ===========================================================================
// v3 backward scan — v1 structure plus a software-pipelined prefetch of
// the NEXT (earlier) timestep's gate activations and cell states: those
// loads depend only on t, so they issue before the MFMA dh-carry phase
// and their HBM latency hides under the matrix work + barrier. The bwd
// workgroup is LDS-bound to ~2 waves/SIMD, so hardware occupancy can't
// hide the 7 loads per element the gate-backward phase makes — the
// prefetch buffer (2 packed uints + 2 floats per element slot, <=64
// VGPRs at the 64-row tile, within the 128-VGPR/2-wave budget) does it
// in software. dSeq stays un-prefetched: in the flagship autoencoder
// path (last_only) it is read only at t=T-1. DORMANT until
// GPU-validated: dispatched via GORDO_LSTM_V3=1.
// ===========================================================================

template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_bwd_v3_kernel(
    const bf16* __restrict__ dSeq, const bf16* __restrict__ gacts,
    const float* __restrict__ cs, const bf16* __restrict__ Wh,
    bf16* __restrict__ dG, int B, int T, int H, int ldg, int last_only) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  bf16* WhN = reinterpret_cast<bf16*>(smem);               // [H][ldg] native
  bf16* dgS = WhN + (size_t)H * ldg;                       // [ROWS][ldg]
  bf16* dhS = dgS + (size_t)ROWS * ldg;                    // [ROWS][LDK]
  float* dcS = reinterpret_cast<float*>(dhS + (size_t)ROWS * LDK);  // [ROWS][H]

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const float* csg = cs + ((size_t)g * B + r0) * T * H;
  const bf16* dSg = last_only ? dSeq + ((size_t)g * B + r0) * H
                              : dSeq + ((size_t)g * B + r0) * T * H;
  bf16* dGg = dG + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  for (int i = tid; i < H * ldg; i += 256) {
    int h = i / ldg, n = i % ldg;
    WhN[i] = (n < H4) ? Whg[(size_t)h * H4 + n] : lf2bf(0.f);
  }
  for (int i = tid; i < ROWS * LDK; i += 256) dhS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) dcS[i] = 0.f;
  for (int i = tid; i < ROWS * ldg; i += 256) dgS[i] = lf2bf(0.f);
  __syncthreads();

  // element slots this thread owns in the strided e-loop (H <= 64).
  // The prefetch buffer is capped at 8 slots (32 VGPRs): the full
  // 16-slot buffer at the 64-row tile pushed the kernel to 360
  // combined V+AGPRs -> 1 wave/SIMD, LOSING the LDS-bound 2-wave
  // occupancy. 8 slots cover the whole e-loop at the 32-row tile and
  // ~70% of it at 64 rows/H=42; tail slots load v1-style.
  constexpr int MAXE = (ROWS * 64 + 255) / 256;
  constexpr int PF = MAXE > 8 ? 8 : MAXE;
  unsigned int p_if[PF];   // gacts i,f packed
  unsigned int p_go[PF];   // gacts g,o packed
  float p_cc[PF];
  float p_cp[PF];

  auto prefetch_step = [&](int t) {
    #pragma unroll
    for (int s = 0; s < PF; ++s) {
      int e = tid + s * 256;
      unsigned int v_if = 0x3F003F00u;  // two bf16(0.5)
      unsigned int v_go = 0x3F000000u;  // low: bf16(0)=g, high: bf16(0.5)=o
      float cc = 0.f, cp = 0.f;
      if (e < ROWS * H) {
        int row = e / H, hh = e % H;
        if (row < rows_here) {
          size_t gbase = ((size_t)row * T + t) * H4;
          v_if = (unsigned int)bf16_bits(gag[gbase + hh]) |
                 ((unsigned int)bf16_bits(gag[gbase + H + hh]) << 16);
          v_go = (unsigned int)bf16_bits(gag[gbase + 2 * H + hh]) |
                 ((unsigned int)bf16_bits(gag[gbase + 3 * H + hh]) << 16);
          size_t cbase = ((size_t)row * T + t) * H + hh;
          cc = csg[cbase];
          cp = (t > 0) ? csg[cbase - H] : 0.f;
        }
      }
      p_if[s] = v_if;
      p_go[s] = v_go;
      p_cc[s] = cc;
      p_cp[s] = cp;
    }
  };

  prefetch_step(T - 1);

  for (int t = T - 1; t >= 0; --t) {
    // ---- fused gate backward: prefetched slots (register-indexed,
    // unrolled), then the tail slots with v1-style direct loads ----
    auto gate_bwd = [&](int e, int t_, float i_g, float f_g, float g_g,
                        float o_g, float cc, float cp) {
      int row = e / H, hh = e % H;
      float dh = lbf2f(dhS[row * LDK + hh]);
      if (row < rows_here) {
        if (last_only) {
          if (t_ == T - 1) dh += lbf2f(dSg[(size_t)row * H + hh]);
        } else {
          dh += lbf2f(dSg[((size_t)row * T + t_) * H + hh]);
        }
      }
      float tc = fast_tanhf_(cc);
      float dc = dcS[row * H + hh] + dh * o_g * (1.f - tc * tc);
      float di = dc * g_g;
      float df = dc * cp;
      float dg = dc * i_g;
      float do_ = dh * tc;
      dcS[row * H + hh] = dc * f_g;
      float vi = di * i_g * (1.f - i_g);
      float vf = df * f_g * (1.f - f_g);
      float vg = dg * (1.f - g_g * g_g);
      float vo = do_ * o_g * (1.f - o_g);
      dgS[row * ldg + hh] = lf2bf(vi);
      dgS[row * ldg + H + hh] = lf2bf(vf);
      dgS[row * ldg + 2 * H + hh] = lf2bf(vg);
      dgS[row * ldg + 3 * H + hh] = lf2bf(vo);
      if (row < rows_here) {
        size_t gbase = ((size_t)row * T + t_) * H4;
        dGg[gbase + hh] = lf2bf(vi);
        dGg[gbase + H + hh] = lf2bf(vf);
        dGg[gbase + 2 * H + hh] = lf2bf(vg);
        dGg[gbase + 3 * H + hh] = lf2bf(vo);
      }
    };
    #pragma unroll
    for (int s = 0; s < PF; ++s) {
      int e = tid + s * 256;
      if (e >= ROWS * H) break;
      gate_bwd(e, t,
               lbf2f(bits_bf16((unsigned short)p_if[s])),
               lbf2f(bits_bf16((unsigned short)(p_if[s] >> 16))),
               lbf2f(bits_bf16((unsigned short)p_go[s])),
               lbf2f(bits_bf16((unsigned short)(p_go[s] >> 16))),
               p_cc[s], p_cp[s]);
    }
    for (int e = tid + PF * 256; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      float i_g = 0.5f, f_g = 0.5f, g_g = 0.f, o_g = 0.5f, cc = 0.f,
            cp = 0.f;
      if (row < rows_here) {
        size_t gbase = ((size_t)row * T + t) * H4;
        i_g = lbf2f(gag[gbase + hh]);
        f_g = lbf2f(gag[gbase + H + hh]);
        g_g = lbf2f(gag[gbase + 2 * H + hh]);
        o_g = lbf2f(gag[gbase + 3 * H + hh]);
        size_t cbase = ((size_t)row * T + t) * H + hh;
        cc = csg[cbase];
        cp = (t > 0) ? csg[cbase - H] : 0.f;
      }
      gate_bwd(e, t, i_g, f_g, g_g, o_g, cc, cp);
    }
    // issue the next (earlier) step's loads before the MFMA phase
    if (t > 0) prefetch_step(t - 1);
    __syncthreads();

    // ---- dh_carry = dgates @ Wh^T (MFMA): out [ROWS][H] ----
    if (wid < ROWS / 16) {
      f32x4 acc[4] = {};
      for (int kk = 0; kk < H4; kk += 32) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &dgS[(wid * 16 + l15) * ldg + kk + kslot * 8]);
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = fn * 16 + l15;  // h index
          bf16x8 b = *reinterpret_cast<const bf16x8*>(
              &WhN[(size_t)min(col, H - 1) * ldg + kk + kslot * 8]);
          acc[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[fn], 0, 0, 0);
        }
      }
      #pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = fn * 16 + l15;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = wid * 16 + (lane >> 4) * 4 + r;
          if (col < H) dhS[row * LDK + col] = lf2bf(acc[fn][r]);
        }
      }
    }
    __syncthreads();
  }
}

// ===========================================================================
// v3 forward scan — v1 structure plus a software-pipelined double
// buffer for the x-side gate tile: while the pointwise phase of step t
// runs (LDS + global stores, no dependence on xW[t+1]), the loads of
// xW[t+1] are already in flight into registers, so their HBM latency
// hides under the gate math and the barrier instead of stalling the
// MFMA epilogue of step t+1. The tile is kept packed two-bf16-per-u32
// to halve the register footprint (2 buffers x FM x 4 cols x 2 pairs).
// DORMANT until GPU-validated: dispatched only via GORDO_LSTM_V3=1
// (ROADMAP round-2 lever #1).
// ===========================================================================

template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_fwd_v3_kernel(
    const bf16* __restrict__ xW, const bf16* __restrict__ Wh,
    bf16* __restrict__ hs, float* __restrict__ cs,
    bf16* __restrict__ gacts, int B, int T, int H, int ldg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  bf16* WhT = reinterpret_cast<bf16*>(smem);              // [H4][LDK]
  bf16* hS = WhT + (size_t)H4 * LDK;                      // [ROWS][LDK]
  bf16* gS = hS + (size_t)ROWS * LDK;                     // [ROWS][ldg]
  float* cS = reinterpret_cast<float*>(gS + (size_t)ROWS * ldg);  // [ROWS][H]

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* xWg = xW + ((size_t)g * B + r0) * T * H4;
  bf16* hsg = hs + ((size_t)g * B + r0) * T * H;
  float* csg = cs + ((size_t)g * B + r0) * T * H;
  bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  for (int i = tid; i < H4 * LDK; i += 256) {
    int n = i / LDK, k = i % LDK;
    WhT[i] = (k < H) ? Whg[(size_t)k * H4 + n] : lf2bf(0.f);
  }
  for (int i = tid; i < ROWS * LDK; i += 256) hS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) cS[i] = 0.f;
  __syncthreads();

  const int wcol0 = wid * 64;
  const bool wave_active = wcol0 < H4;
  constexpr int FM = ROWS / 16;

  // packed x-gate tile: [fm][fn][pair] = (row 2p, row 2p+1) as 2x bf16
  unsigned int xv[FM][4][2];

  // prologue: prefetch t = 0
  if (wave_active) {
    #pragma unroll
    for (int fm = 0; fm < FM; ++fm) {
      #pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = wcol0 + fn * 16 + l15;
        #pragma unroll
        for (int p = 0; p < 2; ++p) {
          unsigned int pk = 0;
          #pragma unroll
          for (int q = 0; q < 2; ++q) {
            int row = fm * 16 + kslot * 4 + p * 2 + q;
            bf16 v = (col < H4 && row < rows_here)
                         ? xWg[((size_t)row * T + 0) * H4 + col]
                         : lf2bf(0.f);
            pk |= (unsigned int)bf16_bits(v) << (16 * q);
          }
          xv[fm][fn][p] = pk;
        }
      }
    }
  }

  for (int t = 0; t < T; ++t) {
    if (wave_active) {
      f32x4 acc[FM][4] = {};
      for (int kk = 0; kk < H; kk += 32) {
        #pragma unroll
        for (int fm = 0; fm < FM; ++fm) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &hS[(fm * 16 + l15) * LDK + kk + kslot * 8]);
          #pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = wcol0 + fn * 16 + l15;
            bf16x8 b = *reinterpret_cast<const bf16x8*>(
                &WhT[(size_t)min(col, H4 - 1) * LDK + kk + kslot * 8]);
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[fm][fn], 0, 0, 0);
          }
        }
      }
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          if (col >= H4) continue;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + kslot * 4 + r;
            bf16 xb = bits_bf16(
                (unsigned short)(xv[fm][fn][r >> 1] >> (16 * (r & 1))));
            gS[row * ldg + col] = lf2bf(acc[fm][fn][r] + lbf2f(xb));
          }
        }
      }
    }
    __syncthreads();

    // ---- prefetch t+1 BEFORE the pointwise phase: these loads have no
    // dependence on this step's gate math, so they overlap it ----
    if (wave_active && t + 1 < T) {
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          #pragma unroll
          for (int p = 0; p < 2; ++p) {
            unsigned int pk = 0;
            #pragma unroll
            for (int q = 0; q < 2; ++q) {
              int row = fm * 16 + kslot * 4 + p * 2 + q;
              bf16 v = (col < H4 && row < rows_here)
                           ? xWg[((size_t)row * T + (t + 1)) * H4 + col]
                           : lf2bf(0.f);
              pk |= (unsigned int)bf16_bits(v) << (16 * q);
            }
            xv[fm][fn][p] = pk;
          }
        }
      }
    }

    for (int e = tid; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      const bf16* grow = &gS[row * ldg];
      float i_g = sigmoidf_(lbf2f(grow[hh]));
      float f_g = sigmoidf_(lbf2f(grow[H + hh]));
      float g_g = fast_tanhf_(lbf2f(grow[2 * H + hh]));
      float o_g = sigmoidf_(lbf2f(grow[3 * H + hh]));
      float cc = f_g * cS[row * H + hh] + i_g * g_g;
      float hv = o_g * fast_tanhf_(cc);
      cS[row * H + hh] = cc;
      hS[row * LDK + hh] = lf2bf(hv);
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
    __syncthreads();
  }
}

// ===========================================================================
// v2 scan kernels — require H % 16 == 0 (the engine pads hidden sizes
// to 16; pad units are mathematically inert). With 16-aligned gate
// blocks, the four gate values of hidden unit hh land in the SAME lane
// of the MFMA D-fragments, so the entire gate pointwise phase runs in
// registers: no gate LDS round trip and no barriers in the T loop
// (the only LDS use is the Wh stage plus a wave-private 16-row h/dg
// bounce to re-shape D-layout -> A-layout between steps). Waves are
// fully independent -> 4 blocks/CU of free-running waves.
// ===========================================================================

template <int HF>  // HF = H/16; H = 16*HF <= 64
__global__ __launch_bounds__(256) void lstm_seq_fwd_v2_kernel(
    const bf16* __restrict__ xW, const bf16* __restrict__ Wh,
    bf16* __restrict__ hs, float* __restrict__ cs,
    bf16* __restrict__ gacts, int B, int T) {
  constexpr int H = 16 * HF;
  constexpr int H4 = 4 * H;
  constexpr int FN = 4 * HF;
  constexpr int KK = (H + 31) / 32;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* WhT = reinterpret_cast<bf16*>(smem);            // [H4][LDK] shared
  bf16* hP_all = WhT + (size_t)H4 * LDK;                // [4][16][LDK]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;
  bf16* hP = hP_all + (size_t)wid * 16 * LDK;

  const int nb = (B + 63) / 64;
  const int g = blockIdx.x / nb;
  const int r0 = (blockIdx.x % nb) * 64 + wid * 16;
  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* xWg = xW + ((size_t)g * B + r0) * T * H4;
  bf16* hsg = hs + ((size_t)g * B + r0) * T * H;
  float* csg = cs + ((size_t)g * B + r0) * T * H;
  bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const int rows_here = max(0, min(16, B - r0));

  for (int i = tid; i < H4 * LDK; i += 256) {
    int n = i / LDK, k = i % LDK;
    WhT[i] = (k < H) ? Whg[(size_t)k * H4 + n] : lf2bf(0.f);
  }
  for (int i = lane; i < 16 * LDK; i += 64) hP[i] = lf2bf(0.f);
  __syncthreads();  // WhT ready — the only barrier

  float cReg[HF][4] = {};
  const int myrow = kslot * 4;  // this lane's 4 rows: myrow..myrow+3

  for (int t = 0; t < T; ++t) {
    // prefetch the x-side gate tile for this step (in-lane positions)
    bf16 xv[FN][4];
    #pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      int col = fn * 16 + l15;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = myrow + r;
        xv[fn][r] = (row < rows_here)
                        ? xWg[((size_t)row * T + t) * H4 + col]
                        : lf2bf(0.f);
      }
    }
    // h A-fragments from the wave-private bounce buffer
    bf16x8 afrag[KK];
    #pragma unroll
    for (int kk = 0; kk < KK; ++kk)
      afrag[kk] = *reinterpret_cast<const bf16x8*>(
          &hP[l15 * LDK + kk * 32 + kslot * 8]);
    // gates = h @ Wh
    f32x4 acc[FN];
    #pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      f32x4 a = {};
      #pragma unroll
      for (int kk = 0; kk < KK; ++kk) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &WhT[(size_t)(fn * 16 + l15) * LDK + kk * 32 + kslot * 8]);
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[kk], b, a, 0, 0,
                                                    0);
      }
      acc[fn] = a;
    }
    // fused gate math — entirely in registers (same lane holds all 4
    // gate values of each (row, hh))
    #pragma unroll
    for (int hf = 0; hf < HF; ++hf) {
      int hh = hf * 16 + l15;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = myrow + r;
        float i_g = sigmoidf_(acc[hf][r] + lbf2f(xv[hf][r]));
        float f_g = sigmoidf_(acc[HF + hf][r] + lbf2f(xv[HF + hf][r]));
        float g_g =
            fast_tanhf_(acc[2 * HF + hf][r] + lbf2f(xv[2 * HF + hf][r]));
        float o_g = sigmoidf_(acc[3 * HF + hf][r] + lbf2f(xv[3 * HF + hf][r]));
        float cc = f_g * cReg[hf][r] + i_g * g_g;
        float hv = o_g * fast_tanhf_(cc);
        cReg[hf][r] = cc;
        hP[row * LDK + hh] = lf2bf(hv);
        if (row < rows_here) {
          size_t base = ((size_t)row * T + t) * H + hh;
          hsg[base] = lf2bf(hv);
          csg[base] = cc;
          size_t gb = ((size_t)row * T + t) * H4;
          gag[gb + hh] = lf2bf(i_g);
          gag[gb + H + hh] = lf2bf(f_g);
          gag[gb + 2 * H + hh] = lf2bf(g_g);
          gag[gb + 3 * H + hh] = lf2bf(o_g);
        }
      }
    }
  }
}

template <int HF>
__global__ __launch_bounds__(256) void lstm_seq_bwd_v2_kernel(
    const bf16* __restrict__ dSeq, const bf16* __restrict__ gacts,
    const float* __restrict__ cs, const bf16* __restrict__ Wh,
    bf16* __restrict__ dG, int B, int T, int ldg, int last_only) {
  constexpr int H = 16 * HF;
  constexpr int H4 = 4 * H;
  constexpr int KK4 = (H4 + 31) / 32;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* WhN = reinterpret_cast<bf16*>(smem);             // [H][ldg] shared
  bf16* dgP_all = WhN + (size_t)H * ldg;                 // [4][16][ldg]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;
  bf16* dgP = dgP_all + (size_t)wid * 16 * ldg;

  const int nb = (B + 63) / 64;
  const int g = blockIdx.x / nb;
  const int r0 = (blockIdx.x % nb) * 64 + wid * 16;
  const bf16* Whg = Wh + (size_t)g * H * H4;
  const bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const float* csg = cs + ((size_t)g * B + r0) * T * H;
  const bf16* dSg = last_only ? dSeq + ((size_t)g * B + r0) * H
                              : dSeq + ((size_t)g * B + r0) * T * H;
  bf16* dGg = dG + ((size_t)g * B + r0) * T * H4;
  const int rows_here = max(0, min(16, B - r0));

  for (int i = tid; i < H * ldg; i += 256) {
    int h = i / ldg, n = i % ldg;
    WhN[i] = (n < H4) ? Whg[(size_t)h * H4 + n] : lf2bf(0.f);
  }
  for (int i = lane; i < 16 * ldg; i += 64) dgP[i] = lf2bf(0.f);
  __syncthreads();  // WhN ready

  float dhReg[HF][4] = {};
  float dcReg[HF][4] = {};
  const int myrow = kslot * 4;

  for (int t = T - 1; t >= 0; --t) {
    // fused gate backward — in registers, in-lane
    #pragma unroll
    for (int hf = 0; hf < HF; ++hf) {
      int hh = hf * 16 + l15;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = myrow + r;
        float dh = dhReg[hf][r];
        float i_g = 0.5f, f_g = 0.5f, g_g = 0.f, o_g = 0.5f, cc = 0.f,
              cp = 0.f;
        if (row < rows_here) {
          if (last_only) {
            if (t == T - 1) dh += lbf2f(dSg[(size_t)row * H + hh]);
          } else {
            dh += lbf2f(dSg[((size_t)row * T + t) * H + hh]);
          }
          size_t gb = ((size_t)row * T + t) * H4;
          i_g = lbf2f(gag[gb + hh]);
          f_g = lbf2f(gag[gb + H + hh]);
          g_g = lbf2f(gag[gb + 2 * H + hh]);
          o_g = lbf2f(gag[gb + 3 * H + hh]);
          size_t cb = ((size_t)row * T + t) * H + hh;
          cc = csg[cb];
          cp = (t > 0) ? csg[cb - H] : 0.f;
        }
        float tc = fast_tanhf_(cc);
        float dc = dcReg[hf][r] + dh * o_g * (1.f - tc * tc);
        float vi = dc * g_g * i_g * (1.f - i_g);
        float vf = dc * cp * f_g * (1.f - f_g);
        float vg = dc * i_g * (1.f - g_g * g_g);
        float vo = dh * tc * o_g * (1.f - o_g);
        dcReg[hf][r] = dc * f_g;
        dgP[row * ldg + hh] = lf2bf(vi);
        dgP[row * ldg + H + hh] = lf2bf(vf);
        dgP[row * ldg + 2 * H + hh] = lf2bf(vg);
        dgP[row * ldg + 3 * H + hh] = lf2bf(vo);
        if (row < rows_here) {
          size_t gb = ((size_t)row * T + t) * H4;
          dGg[gb + hh] = lf2bf(vi);
          dGg[gb + H + hh] = lf2bf(vf);
          dGg[gb + 2 * H + hh] = lf2bf(vg);
          dGg[gb + 3 * H + hh] = lf2bf(vo);
        }
      }
    }
    // dh_carry = dgates @ Wh^T (wave-local; D-layout output stays in
    // registers for the next step's pointwise)
    #pragma unroll
    for (int hf = 0; hf < HF; ++hf) {
      f32x4 a = {};
      #pragma unroll
      for (int kk = 0; kk < KK4; ++kk) {
        bf16x8 af = *reinterpret_cast<const bf16x8*>(
            &dgP[l15 * ldg + kk * 32 + kslot * 8]);
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &WhN[(size_t)(hf * 16 + l15) * ldg + kk * 32 + kslot * 8]);
        a = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b, a, 0, 0, 0);
      }
      #pragma unroll
      for (int r = 0; r < 4; ++r) dhReg[hf][r] = a[r];
    }
  }
}

// ===========================================================================
// Big-H scan kernels (64 < H <= 256, H % 8 == 0) — the reference's
// DEFAULT LSTM dims are (256, 128, 64) (reference
// gordo/machine/model/factories/lstm_autoencoder.py:112), which the
// H<=64 kernels cannot serve: WhT for H=128 is 139 KB of LDS alone.
// Design: Wh stays in GLOBAL memory (one model's Wh is 128 KB-512 KB —
// L2-resident per XCD after the first timestep; B-fragments are
// 16-byte k-contiguous loads), LDS holds only the per-tile h / c /
// gate state, and each wave loops over 64-column tiles of the 4H gate
// dimension. Compute per step is O(H^2) while the serial dependence
// stays 1 step, so the latency-boundness of the small-H scans fades:
// at H=256 each workgroup issues 2048 MFMAs per timestep.
// ===========================================================================

template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_fwd_big_kernel(
    const bf16* __restrict__ xW, const bf16* __restrict__ WhT_g,
    bf16* __restrict__ hs, float* __restrict__ cs,
    bf16* __restrict__ gacts, int B, int T, int H, int ldg, int Kp) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  const int LDH = Kp + 8;  // padded h-row length (bank-conflict pad)
  bf16* hS = reinterpret_cast<bf16*>(smem);               // [ROWS][LDH]
  bf16* gS = hS + (size_t)ROWS * LDH;                     // [ROWS][ldg]
  float* cS = reinterpret_cast<float*>(gS + (size_t)ROWS * ldg);  // [ROWS][H]

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* WhTg = WhT_g + (size_t)g * H4 * Kp;
  const bf16* xWg = xW + ((size_t)g * B + r0) * T * H4;
  bf16* hsg = hs + ((size_t)g * B + r0) * T * H;
  float* csg = cs + ((size_t)g * B + r0) * T * H;
  bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  // zero h (incl. the [H, Kp) pad columns the MFMA K loop reads), c
  for (int i = tid; i < ROWS * LDH; i += 256) hS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) cS[i] = 0.f;
  __syncthreads();

  constexpr int FM = ROWS / 16;

  for (int t = 0; t < T; ++t) {
    // ---- gates = h @ Wh (MFMA, Wh streamed from L2) + xW_t ----
    for (int wcol0 = wid * 64; wcol0 < H4; wcol0 += 256) {
      // prefetch this tile's x-side gate values (register-resident;
      // HBM latency hides under the MFMA loop)
      bf16 xv[FM][4][4];
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + kslot * 4 + r;
            xv[fm][fn][r] =
                (col < H4 && row < rows_here)
                    ? xWg[((size_t)row * T + t) * H4 + col]
                    : lf2bf(0.f);
          }
        }
      }
      f32x4 acc[FM][4] = {};
      for (int kk = 0; kk < Kp; kk += 32) {
        #pragma unroll
        for (int fm = 0; fm < FM; ++fm) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &hS[(fm * 16 + l15) * LDH + kk + kslot * 8]);
          #pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = wcol0 + fn * 16 + l15;
            bf16x8 b = *reinterpret_cast<const bf16x8*>(
                &WhTg[(size_t)min(col, H4 - 1) * Kp + kk + kslot * 8]);
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[fm][fn], 0, 0, 0);
          }
        }
      }
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          if (col >= H4) continue;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + kslot * 4 + r;
            gS[row * ldg + col] =
                lf2bf(acc[fm][fn][r] + lbf2f(xv[fm][fn][r]));
          }
        }
      }
    }
    __syncthreads();

    // ---- fused gate math + h/c update + outputs (identical to v1) ----
    for (int e = tid; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      const bf16* grow = &gS[row * ldg];
      float i_g = sigmoidf_(lbf2f(grow[hh]));
      float f_g = sigmoidf_(lbf2f(grow[H + hh]));
      float g_g = fast_tanhf_(lbf2f(grow[2 * H + hh]));
      float o_g = sigmoidf_(lbf2f(grow[3 * H + hh]));
      float cc = f_g * cS[row * H + hh] + i_g * g_g;
      float hv = o_g * fast_tanhf_(cc);
      cS[row * H + hh] = cc;
      hS[row * LDH + hh] = lf2bf(hv);
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
    __syncthreads();
  }
}

template <int ROWS>
__global__ __launch_bounds__(256) void lstm_seq_bwd_big_kernel(
    const bf16* __restrict__ dSeq, const bf16* __restrict__ gacts,
    const float* __restrict__ cs, const bf16* __restrict__ WhN_g,
    bf16* __restrict__ dG, int B, int T, int H, int ldg, int last_only) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;          // K length of the carry GEMM (32-mult)
  const int LDH = ((H + 31) & ~31) + 8;
  bf16* dgS = reinterpret_cast<bf16*>(smem);              // [ROWS][ldg]
  bf16* dhS = dgS + (size_t)ROWS * ldg;                   // [ROWS][LDH]
  float* dcS = reinterpret_cast<float*>(dhS + (size_t)ROWS * LDH);  // [ROWS][H]

  const int g = blockIdx.x / ((B + ROWS - 1) / ROWS);
  const int r0 = (blockIdx.x % ((B + ROWS - 1) / ROWS)) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* WhNg = WhN_g + (size_t)g * H * H4;  // native [H][4H]
  const bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const float* csg = cs + ((size_t)g * B + r0) * T * H;
  const bf16* dSg = last_only ? dSeq + ((size_t)g * B + r0) * H
                              : dSeq + ((size_t)g * B + r0) * T * H;
  bf16* dGg = dG + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  for (int i = tid; i < ROWS * LDH; i += 256) dhS[i] = lf2bf(0.f);
  for (int i = tid; i < ROWS * H; i += 256) dcS[i] = 0.f;
  for (int i = tid; i < ROWS * ldg; i += 256) dgS[i] = lf2bf(0.f);
  __syncthreads();

  constexpr int FM = ROWS / 16;

  for (int t = T - 1; t >= 0; --t) {
    // ---- fused gate backward (identical to v1) ----
    for (int e = tid; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      float dh = lbf2f(dhS[row * LDH + hh]);
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
      float tc = fast_tanhf_(cc);
      float dc = dcS[row * H + hh] + dh * o_g * (1.f - tc * tc);
      float di = dc * g_g;
      float df = dc * cp;
      float dg = dc * i_g;
      float do_ = dh * tc;
      dcS[row * H + hh] = dc * f_g;
      float vi = di * i_g * (1.f - i_g);
      float vf = df * f_g * (1.f - f_g);
      float vg = dg * (1.f - g_g * g_g);
      float vo = do_ * o_g * (1.f - o_g);
      dgS[row * ldg + hh] = lf2bf(vi);
      dgS[row * ldg + H + hh] = lf2bf(vf);
      dgS[row * ldg + 2 * H + hh] = lf2bf(vg);
      dgS[row * ldg + 3 * H + hh] = lf2bf(vo);
      if (row < rows_here) {
        dGg[gbase + hh] = lf2bf(vi);
        dGg[gbase + H + hh] = lf2bf(vf);
        dGg[gbase + 2 * H + hh] = lf2bf(vg);
        dGg[gbase + 3 * H + hh] = lf2bf(vo);
      }
    }
    __syncthreads();

    // ---- dh_carry = dgates @ Wh^T (MFMA; WhN from L2): out [ROWS][H] ----
    for (int col0 = wid * 64; col0 < H; col0 += 256) {
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        f32x4 acc[4] = {};
        for (int kk = 0; kk < H4; kk += 32) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &dgS[(fm * 16 + l15) * ldg + kk + kslot * 8]);
          #pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = col0 + fn * 16 + l15;  // h index
            bf16x8 b = *reinterpret_cast<const bf16x8*>(
                &WhNg[(size_t)min(col, H - 1) * H4 + kk + kslot * 8]);
            acc[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[fn], 0, 0, 0);
          }
        }
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = col0 + fn * 16 + l15;
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + kslot * 4 + r;
            if (col < H) dhS[row * LDH + col] = lf2bf(acc[fn][r]);
          }
        }
      }
    }
    __syncthreads();
  }
}

// ===========================================================================
// v4 fused-xW forward scan — computes the x-side gate GEMM INSIDE the
// recurrence instead of precomputing xW = x@Wx + b into a [G,B,T,4H]
// HBM tensor: in the fleet the scans are HBM-BANDWIDTH bound (the xW
// round trip is ~770 MB per layer pass at the bench shape), so folding
// the two GEMMs into one K = [h | x_t] x [Wh ; Wx] MFMA loop over an
// LDS-resident concatenated weight tile roughly halves forward-path
// traffic and removes the separate grouped-GEMM launch. The x_{t+1}
// row tile is register-prefetched during the pointwise phase (the v3
// pattern) and written into the LDS K-buffer before the barrier.
// STORE_AUX=false additionally skips the cs/gacts stores for
// inference (predict/CV-validation passes need only hs).
// Geometry: H <= 64, H % 8 == 0, F % 8 == 0 (the pad8 engine layout).
// ===========================================================================
template <int ROWS, bool STORE_AUX>
__global__ __launch_bounds__(256) void lstm_seq_fwd_v4_kernel(
    const bf16* __restrict__ xseq,   // [G, B, T, F]
    const bf16* __restrict__ WcatT,  // [G, 4H, Kc] (Kc = H + F)
    const bf16* __restrict__ bias,   // [G, 4H]
    bf16* __restrict__ hs, float* __restrict__ cs,
    bf16* __restrict__ gacts, int B, int T, int H, int F, int ldg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H4 = 4 * H;
  const int Kc = H + F;          // both % 8 == 0
  // pad the K row to the MFMA K step so the last fragment read stays
  // in bounds (zeros staged beyond Kc)
  const int LDKc = ((Kc + 31) & ~31) + 8;
  bf16* WT = reinterpret_cast<bf16*>(smem);               // [H4][LDKc]
  bf16* hxS = WT + (size_t)H4 * LDKc;                     // [ROWS][LDKc]
  bf16* gS = hxS + (size_t)ROWS * LDKc;                   // [ROWS][ldg]
  bf16* bS = gS + (size_t)ROWS * ldg;                     // [ldg]
  float* cS = reinterpret_cast<float*>(bS + ldg);         // [ROWS][H]

  const int nb = (B + ROWS - 1) / ROWS;
  const int g = blockIdx.x / nb;
  const int r0 = (blockIdx.x % nb) * ROWS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int kslot = lane >> 4;

  const bf16* WTg = WcatT + (size_t)g * H4 * Kc;
  const bf16* xg = xseq + ((size_t)g * B + r0) * T * F;
  bf16* hsg = hs + ((size_t)g * B + r0) * T * H;
  float* csg = cs + ((size_t)g * B + r0) * T * H;
  bf16* gag = gacts + ((size_t)g * B + r0) * T * H4;
  const int rows_here = min(ROWS, B - r0);

  // stage the concatenated weight tile (zero-padded K tail) + bias
  for (int i = tid; i < H4 * LDKc; i += 256) {
    int n = i / LDKc, k = i % LDKc;
    WT[i] = (k < Kc) ? WTg[(size_t)n * Kc + k] : lf2bf(0.f);
  }
  for (int i = tid; i < ldg; i += 256)
    bS[i] = (i < H4) ? bias[(size_t)g * H4 + i] : lf2bf(0.f);
  // zero the h section and the K-tail padding ONLY: the x section
  // [H, H+F) is written concurrently by the x_0 staging loop below
  // (different threads own the same address in the two loops, so
  // zeroing it here would race with that staging — seen as corrupted
  // t=0 gates under co-residency before this guard)
  for (int i = tid; i < ROWS * LDKc; i += 256) {
    int k = i % LDKc;
    if (k < H || k >= H + F) hxS[i] = lf2bf(0.f);
  }
  for (int i = tid; i < ROWS * H; i += 256) cS[i] = 0.f;
  // x_0 into the K-buffer's x section (16-byte lanes: F % 8 == 0)
  for (int i = tid; i < ROWS * (F / 8); i += 256) {
    int row = i / (F / 8), f8 = i % (F / 8);
    bf16x8 v = {};
    if (row < rows_here)
      v = *reinterpret_cast<const bf16x8*>(
          &xg[((size_t)row * T + 0) * F + f8 * 8]);
    *reinterpret_cast<bf16x8*>(&hxS[row * LDKc + H + f8 * 8]) = v;
  }
  __syncthreads();

  constexpr int FM = ROWS / 16;
  const int wcol0 = wid * 64;
  const bool wave_active = wcol0 < H4;
  // per-thread x-prefetch slots (strided like the staging loop)
  constexpr int NPF = (ROWS * 16 + 255) / 256;  // F/8 <= 16 at F<=128
  bf16x8 xpf[NPF];

  for (int t = 0; t < T; ++t) {
    // ---- gates = [h | x_t] @ [Wh ; Wx] + b (MFMA over Kc) ----
    if (wave_active) {
      f32x4 acc[FM][4] = {};
      for (int kk = 0; kk < Kc; kk += 32) {
        #pragma unroll
        for (int fm = 0; fm < FM; ++fm) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &hxS[(fm * 16 + l15) * LDKc + kk + kslot * 8]);
          #pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = wcol0 + fn * 16 + l15;
            bf16x8 b = *reinterpret_cast<const bf16x8*>(
                &WT[(size_t)min(col, H4 - 1) * LDKc + kk + kslot * 8]);
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, b, acc[fm][fn], 0, 0, 0);
          }
        }
      }
      #pragma unroll
      for (int fm = 0; fm < FM; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wcol0 + fn * 16 + l15;
          if (col >= H4) continue;
          float bv = lbf2f(bS[col]);
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = fm * 16 + kslot * 4 + r;
            gS[row * ldg + col] = lf2bf(acc[fm][fn][r] + bv);
          }
        }
      }
    }
    __syncthreads();

    // ---- register-prefetch x_{t+1} (no dependence on this step) ----
    const int nx = ROWS * (F / 8);
    if (t + 1 < T) {
      #pragma unroll
      for (int s = 0; s < NPF; ++s) {
        int i = tid + s * 256;
        if (i < nx) {
          int row = i / (F / 8), f8 = i % (F / 8);
          bf16x8 v = {};
          if (row < rows_here)
            v = *reinterpret_cast<const bf16x8*>(
                &xg[((size_t)row * T + (t + 1)) * F + f8 * 8]);
          xpf[s] = v;
        }
      }
    }

    // ---- fused gate math + h/c update + outputs ----
    for (int e = tid; e < ROWS * H; e += 256) {
      int row = e / H, hh = e % H;
      const bf16* grow = &gS[row * ldg];
      float i_g = sigmoidf_(lbf2f(grow[hh]));
      float f_g = sigmoidf_(lbf2f(grow[H + hh]));
      float g_g = fast_tanhf_(lbf2f(grow[2 * H + hh]));
      float o_g = sigmoidf_(lbf2f(grow[3 * H + hh]));
      float cc = f_g * cS[row * H + hh] + i_g * g_g;
      float hv = o_g * fast_tanhf_(cc);
      cS[row * H + hh] = cc;
      hxS[row * LDKc + hh] = lf2bf(hv);
      if (row < rows_here) {
        size_t base = ((size_t)row * T + t) * H + hh;
        hsg[base] = lf2bf(hv);
        if (STORE_AUX) {
          csg[base] = cc;
          size_t gbase = ((size_t)row * T + t) * H4;
          gag[gbase + hh] = lf2bf(i_g);
          gag[gbase + H + hh] = lf2bf(f_g);
          gag[gbase + 2 * H + hh] = lf2bf(g_g);
          gag[gbase + 3 * H + hh] = lf2bf(o_g);
        }
      }
    }
    // write the prefetched x_{t+1} into the K-buffer before the
    // barrier (its x section is not read by the pointwise phase)
    if (t + 1 < T) {
      #pragma unroll
      for (int s = 0; s < NPF; ++s) {
        int i = tid + s * 256;
        if (i < nx) {
          int row = i / (F / 8), f8 = i % (F / 8);
          *reinterpret_cast<bf16x8*>(
              &hxS[row * LDKc + H + f8 * 8]) = xpf[s];
        }
      }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
namespace gordo_lstm {

// pad 4H up to the MFMA K-step (32) plus 8: the K loop's last
// fragment read may touch columns [H4, ru32(H4)); they must exist in
// the row and be zero.
inline int pad_ldg(int h4) { return ((h4 + 31) & ~31) + 8; }

inline int pick_rows(int G, int B) {
  // smallest row tile whose grid reaches ~2 workgroups per CU: the
  // scans are latency-bound chains, so co-resident workgroups on one
  // CU overlap each other's stalls (WhT is staged once per launch, so
  // smaller tiles do not re-read Wh). GORDO_LSTM_ROWS forces a tile
  // for A/B measurement.
  if (const char* e = getenv("GORDO_LSTM_ROWS")) {
    int r = atoi(e);
    if (r == 16 || r == 32 || r == 64) return r;
  }
  if (G * ((B + 63) / 64) >= 512) return 64;
  if (G * ((B + 31) / 32) >= 512) return 32;
  return 16;
}

// ---- big-H path (64 < H <= 256, H % 8 == 0) ----
inline bool big_ok(int H) { return H > 64 && H <= 256 && H % 8 == 0; }

inline size_t big_lds(int rows, int H, int ldg, int Kp) {
  int LDH = Kp + 8;
  return (size_t)rows * LDH * 2 + (size_t)rows * ldg * 2 +
         (size_t)rows * H * 4;
}

inline int pick_rows_big(int B, int H, int ldg, int Kp) {
  // 64-row tiles halve the per-step Wh L2 re-reads (each workgroup
  // streams the whole Wh every timestep) — use them whenever the LDS
  // fits AND the batch actually fills the tile.
  if (B >= 48 && big_lds(64, H, ldg, Kp) <= 160 * 1024) return 64;
  return 32;
}

std::vector<torch::Tensor> lstm_seq_fwd_big(torch::Tensor xW,
                                            torch::Tensor Wh) {
  TORCH_CHECK(xW.is_cuda() && xW.dim() == 4, "xW must be [G,B,T,4H] on GPU");
  auto xc = xW.to(torch::kBFloat16).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = xc.size(0), B = xc.size(1), T = xc.size(2), H4 = xc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(big_ok(H), "lstm_seq_fwd_big needs 64 < H <= 256, H % 8 == 0");
  int Kp = (H + 31) & ~31;
  int ldg = pad_ldg(H4);
  // pre-transpose Wh -> [G, 4H, Kp] (zero-padded K) once per launch so
  // B-fragment loads are 16-byte k-contiguous
  auto WhT = torch::zeros({G, H4, Kp},
                          Whc.options().dtype(torch::kBFloat16));
  WhT.narrow(2, 0, H).copy_(Whc.transpose(1, 2));
  WhT = WhT.contiguous();
  auto hs = torch::empty({G, B, T, H}, xc.options());
  auto cs = torch::empty({G, B, T, H}, xc.options().dtype(torch::kFloat32));
  auto gacts = torch::empty({G, B, T, H4}, xc.options());
  int rows = pick_rows_big(B, H, ldg, Kp);
  size_t lds = big_lds(rows, H, ldg, Kp);
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded (big fwd)");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_fwd_big_kernel<64>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)WhT.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg, Kp);
  else
    hipLaunchKernelGGL(lstm_seq_fwd_big_kernel<32>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)WhT.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg, Kp);
  return {hs, cs, gacts};
}

torch::Tensor lstm_seq_bwd_big(torch::Tensor dSeq, torch::Tensor gacts,
                               torch::Tensor cs, torch::Tensor Wh,
                               bool last_only) {
  TORCH_CHECK(gacts.is_cuda() && gacts.dim() == 4, "gacts must be [G,B,T,4H]");
  auto dc = dSeq.to(torch::kBFloat16).contiguous();
  auto gc = gacts.to(torch::kBFloat16).contiguous();
  auto ccs = cs.to(torch::kFloat32).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = gc.size(0), B = gc.size(1), T = gc.size(2), H4 = gc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(big_ok(H), "lstm_seq_bwd_big needs 64 < H <= 256, H % 8 == 0");
  int Kp = (H + 31) & ~31;
  int ldg = pad_ldg(H4);
  auto dG = torch::empty_like(gc);
  int rows = pick_rows_big(B, H, ldg, Kp);
  size_t lds = big_lds(rows, H, ldg, Kp);
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded (big bwd)");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_bwd_big_kernel<64>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), ccs.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  else
    hipLaunchKernelGGL(lstm_seq_bwd_big_kernel<32>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), ccs.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  return dG;
}

inline bool v4_ok(int H, int F) {
  return H <= 64 && H % 8 == 0 && F % 8 == 0 && F <= 128;
}

std::vector<torch::Tensor> lstm_seq_fwd_fused(torch::Tensor xseq,
                                              torch::Tensor Wx,
                                              torch::Tensor Wh,
                                              torch::Tensor bias,
                                              bool store_aux) {
  // xseq [G,B,T,F]; Wx [G,F,4H]; Wh [G,H,4H]; bias [G,4H].
  // Computes the x-side gate GEMM inside the scan (see kernel note).
  TORCH_CHECK(xseq.is_cuda() && xseq.dim() == 4,
              "xseq must be [G,B,T,F] on GPU");
  auto xc = xseq.to(torch::kBFloat16).contiguous();
  auto Wxc = Wx.to(torch::kBFloat16);
  auto Whc = Wh.to(torch::kBFloat16);
  auto bc = bias.to(torch::kBFloat16).contiguous();
  int G = xc.size(0), B = xc.size(1), T = xc.size(2), F = xc.size(3);
  int H = Whc.size(1), H4 = 4 * H;
  TORCH_CHECK(v4_ok(H, F), "lstm_seq_fwd_fused geometry unsupported");
  int Kc = H + F;
  int LDKc = ((Kc + 31) & ~31) + 8;
  int ldg = pad_ldg(H4);
  // concatenated transposed weights [G, 4H, Kc] = [Wh ; Wx]^T
  auto Wcat = torch::cat({Whc, Wxc}, 1).transpose(1, 2).contiguous();
  auto hs = torch::empty({G, B, T, H}, xc.options());
  auto cs = store_aux
                ? torch::empty({G, B, T, H},
                               xc.options().dtype(torch::kFloat32))
                : torch::empty({1}, xc.options().dtype(torch::kFloat32));
  auto gacts = store_aux ? torch::empty({G, B, T, H4}, xc.options())
                         : torch::empty({1}, xc.options());
  int rows = pick_rows(G, B);
  auto lds_for = [&](int r) {
    return (size_t)H4 * LDKc * 2 + (size_t)r * LDKc * 2 +
           (size_t)r * ldg * 2 + (size_t)ldg * 2 + (size_t)r * H * 4;
  };
  while (rows > 16 && lds_for(rows) > 160 * 1024) rows /= 2;
  size_t lds = lds_for(rows);
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded (v4 fwd)");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  auto xp = (const bf16*)xc.data_ptr();
  auto wp = (const bf16*)Wcat.data_ptr();
  auto bp = (const bf16*)bc.data_ptr();
  auto hp = (bf16*)hs.data_ptr();
  auto cp = cs.data_ptr<float>();
  auto gp = (bf16*)gacts.data_ptr();
  #define V4_LAUNCH(R, S)                                              \
    hipLaunchKernelGGL((lstm_seq_fwd_v4_kernel<R, S>), dim3(blocks),   \
                       dim3(256), lds, stream, xp, wp, bp, hp, cp, gp, \
                       B, T, H, F, ldg)
  if (store_aux) {
    if (rows == 64) V4_LAUNCH(64, true);
    else if (rows == 32) V4_LAUNCH(32, true);
    else V4_LAUNCH(16, true);
  } else {
    if (rows == 64) V4_LAUNCH(64, false);
    else if (rows == 32) V4_LAUNCH(32, false);
    else V4_LAUNCH(16, false);
  }
  #undef V4_LAUNCH
  if (store_aux) return {hs, cs, gacts};
  return {hs};
}

std::vector<torch::Tensor> lstm_seq_fwd(torch::Tensor xW, torch::Tensor Wh) {
  TORCH_CHECK(xW.is_cuda() && xW.dim() == 4, "xW must be [G,B,T,4H] on GPU");
  if (xW.size(3) / 4 > 64) return lstm_seq_fwd_big(xW, Wh);
  auto xc = xW.to(torch::kBFloat16).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = xc.size(0), B = xc.size(1), T = xc.size(2), H4 = xc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(H <= 64, "lstm_seq_fwd supports H <= 64");
  int ldg = pad_ldg(H4);
  auto hs = torch::empty({G, B, T, H}, xc.options());
  auto cs = torch::empty({G, B, T, H}, xc.options().dtype(torch::kFloat32));
  auto gacts = torch::empty({G, B, T, H4}, xc.options());
  int rows = pick_rows(G, B);
  size_t lds = (size_t)H4 * LDK * 2 + (size_t)rows * LDK * 2 +
               (size_t)rows * ldg * 2 + (size_t)rows * H * 4;
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  if (H % 16 == 0) {
    // v2: register-resident gate math, no barriers in the T loop
    int blocks2 = G * ((B + 63) / 64);
    size_t lds2 = (size_t)H4 * LDK * 2 + (size_t)4 * 16 * LDK * 2;
    auto xp = (const bf16*)xc.data_ptr();
    auto wp = (const bf16*)Whc.data_ptr();
    auto hp = (bf16*)hs.data_ptr();
    auto cp = cs.data_ptr<float>();
    auto gp = (bf16*)gacts.data_ptr();
    switch (H / 16) {
      case 1: hipLaunchKernelGGL(lstm_seq_fwd_v2_kernel<1>, dim3(blocks2),
                                 dim3(256), lds2, stream, xp, wp, hp, cp, gp,
                                 B, T); break;
      case 2: hipLaunchKernelGGL(lstm_seq_fwd_v2_kernel<2>, dim3(blocks2),
                                 dim3(256), lds2, stream, xp, wp, hp, cp, gp,
                                 B, T); break;
      case 3: hipLaunchKernelGGL(lstm_seq_fwd_v2_kernel<3>, dim3(blocks2),
                                 dim3(256), lds2, stream, xp, wp, hp, cp, gp,
                                 B, T); break;
      default: hipLaunchKernelGGL(lstm_seq_fwd_v2_kernel<4>, dim3(blocks2),
                                 dim3(256), lds2, stream, xp, wp, hp, cp, gp,
                                 B, T); break;
    }
    return {hs, cs, gacts};
  }
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_fwd_kernel<64>, dim3(blocks), dim3(256), lds,
                       stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)Whc.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg);
  else if (rows == 32)
    hipLaunchKernelGGL(lstm_seq_fwd_kernel<32>, dim3(blocks), dim3(256), lds,
                       stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)Whc.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg);
  else
    hipLaunchKernelGGL(lstm_seq_fwd_kernel<16>, dim3(blocks), dim3(256), lds,
                       stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)Whc.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg);
  return {hs, cs, gacts};
}

std::vector<torch::Tensor> lstm_seq_fwd_v3(torch::Tensor xW,
                                            torch::Tensor Wh) {
  // identical contract to lstm_seq_fwd, pipelined kernel (always the
  // barriered layout, never the v2 path: v3 is measured against v1 on
  // the same shapes). H > 64 routes to the big-H kernel (no v3 there).
  TORCH_CHECK(xW.is_cuda() && xW.dim() == 4, "xW must be [G,B,T,4H] on GPU");
  if (xW.size(3) / 4 > 64) return lstm_seq_fwd_big(xW, Wh);
  auto xc = xW.to(torch::kBFloat16).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = xc.size(0), B = xc.size(1), T = xc.size(2), H4 = xc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(H <= 64, "lstm_seq_fwd_v3 supports H <= 64");
  int ldg = pad_ldg(H4);
  auto hs = torch::empty({G, B, T, H}, xc.options());
  auto cs = torch::empty({G, B, T, H}, xc.options().dtype(torch::kFloat32));
  auto gacts = torch::empty({G, B, T, H4}, xc.options());
  int rows = pick_rows(G, B);
  size_t lds = (size_t)H4 * LDK * 2 + (size_t)rows * LDK * 2 +
               (size_t)rows * ldg * 2 + (size_t)rows * H * 4;
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_fwd_v3_kernel<64>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)Whc.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg);
  else if (rows == 32)
    hipLaunchKernelGGL(lstm_seq_fwd_v3_kernel<32>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)Whc.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg);
  else
    hipLaunchKernelGGL(lstm_seq_fwd_v3_kernel<16>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)xc.data_ptr(),
                       (const bf16*)Whc.data_ptr(), (bf16*)hs.data_ptr(),
                       cs.data_ptr<float>(), (bf16*)gacts.data_ptr(), B, T, H,
                       ldg);
  return {hs, cs, gacts};
}


std::vector<torch::Tensor> lstm_seq_bwd_fused(
    torch::Tensor dSeq, torch::Tensor gacts, torch::Tensor cs,
    torch::Tensor Wh, torch::Tensor Wx, bool last_only) {
  // returns {dG, dX}: the reverse scan with dSeq_t = dG_t @ Wx^T
  // fused in (see kernel note). Geometry: H <= 64, F <= 128.
  TORCH_CHECK(gacts.is_cuda() && gacts.dim() == 4, "gacts must be [G,B,T,4H]");
  auto dc = dSeq.to(torch::kBFloat16).contiguous();
  auto gc = gacts.to(torch::kBFloat16).contiguous();
  auto ccs = cs.to(torch::kFloat32).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  auto Wxc = Wx.to(torch::kBFloat16).contiguous();
  int G = gc.size(0), B = gc.size(1), T = gc.size(2), H4 = gc.size(3);
  int H = H4 / 4, F = Wxc.size(1);
  TORCH_CHECK(H <= 64 && F <= 128, "lstm_seq_bwd_fused geometry");
  int ldg = pad_ldg(H4);
  auto dG = torch::empty_like(gc);
  auto dX = torch::empty({G, B, T, F}, gc.options());
  int rows = pick_rows(G, B);
  auto lds_for = [&](int r) {
    return (size_t)H * ldg * 2 + (size_t)F * ldg * 2 +
           (size_t)r * ldg * 2 + (size_t)r * LDK * 2 + (size_t)r * H * 4;
  };
  while (rows > 16 && lds_for(rows) > 160 * 1024) rows /= 2;
  size_t lds = lds_for(rows);
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded (v5 bwd)");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  auto dp = (const bf16*)dc.data_ptr();
  auto gp = (const bf16*)gc.data_ptr();
  auto cp = ccs.data_ptr<float>();
  auto whp = (const bf16*)Whc.data_ptr();
  auto wxp = (const bf16*)Wxc.data_ptr();
  auto dgp = (bf16*)dG.data_ptr();
  auto dxp = (bf16*)dX.data_ptr();
  int lo = last_only ? 1 : 0;
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_bwd_v5_kernel<64>, dim3(blocks), dim3(256),
                       lds, stream, dp, gp, cp, whp, wxp, dgp, dxp, B, T,
                       H, F, ldg, lo);
  else if (rows == 32)
    hipLaunchKernelGGL(lstm_seq_bwd_v5_kernel<32>, dim3(blocks), dim3(256),
                       lds, stream, dp, gp, cp, whp, wxp, dgp, dxp, B, T,
                       H, F, ldg, lo);
  else
    hipLaunchKernelGGL(lstm_seq_bwd_v5_kernel<16>, dim3(blocks), dim3(256),
                       lds, stream, dp, gp, cp, whp, wxp, dgp, dxp, B, T,
                       H, F, ldg, lo);
  return {dG, dX};
}

torch::Tensor lstm_seq_bwd_v3(torch::Tensor dSeq, torch::Tensor gacts,
                              torch::Tensor cs, torch::Tensor Wh,
                              bool last_only) {
  // identical contract to lstm_seq_bwd; pipelined kernel, always the
  // barriered layout (A/B-tested against v1 on the same shapes).
  // H > 64 routes to the big-H kernel (no v3 there).
  TORCH_CHECK(gacts.is_cuda() && gacts.dim() == 4, "gacts must be [G,B,T,4H]");
  if (gacts.size(3) / 4 > 64)
    return lstm_seq_bwd_big(dSeq, gacts, cs, Wh, last_only);
  auto dc = dSeq.to(torch::kBFloat16).contiguous();
  auto gc = gacts.to(torch::kBFloat16).contiguous();
  auto cc = cs.to(torch::kFloat32).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = gc.size(0), B = gc.size(1), T = gc.size(2), H4 = gc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(H <= 64, "lstm_seq_bwd_v3 supports H <= 64");
  int ldg = pad_ldg(H4);
  auto dG = torch::empty_like(gc);
  int rows = pick_rows(G, B);
  size_t lds = (size_t)H * ldg * 2 + (size_t)rows * ldg * 2 +
               (size_t)rows * LDK * 2 + (size_t)rows * H * 4;
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_bwd_v3_kernel<64>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), cc.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  else if (rows == 32)
    hipLaunchKernelGGL(lstm_seq_bwd_v3_kernel<32>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), cc.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  else
    hipLaunchKernelGGL(lstm_seq_bwd_v3_kernel<16>, dim3(blocks), dim3(256),
                       lds, stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), cc.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  return dG;
}

torch::Tensor lstm_seq_bwd(torch::Tensor dSeq, torch::Tensor gacts,
                           torch::Tensor cs, torch::Tensor Wh,
                           bool last_only) {
  TORCH_CHECK(gacts.is_cuda() && gacts.dim() == 4, "gacts must be [G,B,T,4H]");
  if (gacts.size(3) / 4 > 64)
    return lstm_seq_bwd_big(dSeq, gacts, cs, Wh, last_only);
  auto dc = dSeq.to(torch::kBFloat16).contiguous();
  auto gc = gacts.to(torch::kBFloat16).contiguous();
  auto cc = cs.to(torch::kFloat32).contiguous();
  auto Whc = Wh.to(torch::kBFloat16).contiguous();
  int G = gc.size(0), B = gc.size(1), T = gc.size(2), H4 = gc.size(3);
  int H = H4 / 4;
  TORCH_CHECK(H <= 64, "lstm_seq_bwd supports H <= 64");
  int ldg = pad_ldg(H4);
  auto dG = torch::empty_like(gc);
  int rows = pick_rows(G, B);
  size_t lds = (size_t)H * ldg * 2 + (size_t)rows * ldg * 2 +
               (size_t)rows * LDK * 2 + (size_t)rows * H * 4;
  TORCH_CHECK(lds <= 160 * 1024, "LDS budget exceeded");
  int blocks = G * ((B + rows - 1) / rows);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  if (H % 16 == 0) {
    int blocks2 = G * ((B + 63) / 64);
    size_t lds2 = (size_t)H * ldg * 2 + (size_t)4 * 16 * ldg * 2;
    auto dp = (const bf16*)dc.data_ptr();
    auto gp = (const bf16*)gc.data_ptr();
    auto cp = cc.data_ptr<float>();
    auto wp = (const bf16*)Whc.data_ptr();
    auto op = (bf16*)dG.data_ptr();
    int lo = last_only ? 1 : 0;
    switch (H / 16) {
      case 1: hipLaunchKernelGGL(lstm_seq_bwd_v2_kernel<1>, dim3(blocks2),
                                 dim3(256), lds2, stream, dp, gp, cp, wp, op,
                                 B, T, ldg, lo); break;
      case 2: hipLaunchKernelGGL(lstm_seq_bwd_v2_kernel<2>, dim3(blocks2),
                                 dim3(256), lds2, stream, dp, gp, cp, wp, op,
                                 B, T, ldg, lo); break;
      case 3: hipLaunchKernelGGL(lstm_seq_bwd_v2_kernel<3>, dim3(blocks2),
                                 dim3(256), lds2, stream, dp, gp, cp, wp, op,
                                 B, T, ldg, lo); break;
      default: hipLaunchKernelGGL(lstm_seq_bwd_v2_kernel<4>, dim3(blocks2),
                                 dim3(256), lds2, stream, dp, gp, cp, wp, op,
                                 B, T, ldg, lo); break;
    }
    return dG;
  }
  if (rows == 64)
    hipLaunchKernelGGL(lstm_seq_bwd_kernel<64>, dim3(blocks), dim3(256), lds,
                       stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), cc.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  else if (rows == 32)
    hipLaunchKernelGGL(lstm_seq_bwd_kernel<32>, dim3(blocks), dim3(256), lds,
                       stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), cc.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  else
    hipLaunchKernelGGL(lstm_seq_bwd_kernel<16>, dim3(blocks), dim3(256), lds,
                       stream, (const bf16*)dc.data_ptr(),
                       (const bf16*)gc.data_ptr(), cc.data_ptr<float>(),
                       (const bf16*)Whc.data_ptr(), (bf16*)dG.data_ptr(), B,
                       T, H, ldg, last_only ? 1 : 0);
  return dG;
}

}  // namespace gordo_lstm
