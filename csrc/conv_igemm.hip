// MI355X (gfx950/CDNA4) implicit-GEMM convolution kernels — NHWC bf16.
//
// Replaces the reference's dependency on cuDNN conv kernels (SURVEY.md §2.2
// N1: 3x3 s1/s2 and 1x1 convs of the CIFAR ResNet, fwd + dgrad + wgrad).
// Written MFMA-first for CDNA4: v_mfma_f32_16x16x32_bf16 tiles, fp32
// accumulation, LDS-staged operand tiles sized for 64-wide wavefronts.
//
// GEMM views (all NHWC, reduction in fp32):
//   fwd  : out[N*P*Q][Kout] = im2col(x)[M][R*S*C]   @ w[Kout][R*S*C]^T
//   dgrad: dx[N*H*W][C]     = im2col'(dy)[M][R*S*K] @ wT[C][R*S*K]^T
//          (wT = weight rotated 180° and transposed to [C][R][S][K], built
//           host-side — a few KB)
//   wgrad: dw[Kout][R*S*C]  = dy^T[Kout][N*P*Q]     @ im2col(x)[N*P*Q][R*S*C]
//          (split-K over N*P*Q chunks, fp32 atomics into a dw accumulator)
//
// FAST path: when the contiguous channel count is a multiple of BK(=64) a
// BK-sized reduction chunk lies inside ONE (r,s) filter tap with a contiguous
// channel run — every A-tile row is one 64-byte-aligned 128 B global read
// (or zero-fill). This covers every ResNet conv except the 3-channel stem.
//
// PADC path (the stem): C==3 inputs are padded once per call to 4 channels
// (x -> NHWC C=4; weight -> [K][K_pad] with K_pad = R*S*4 rounded up to BK,
// zero-filled), which makes every filter tap a naturally-aligned 8 B dwordx2
// gather — 4x fewer loads and ~8x less address math than the per-element
// generic gather that round 1 used for the stem.
#include <torch/extension.h>
#include <ATen/ATen.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 128;   // GEMM M tile (output pixels)
constexpr int BN = 64;    // GEMM N tile (output channels / rsc)
constexpr int BK = 64;    // reduction tile
constexpr int LDK = BK + 8;  // padded LDS row stride: (m*LDK) covers all 64
                             // banks across a 16-lane ds_read_b128 group

struct ConvDims {
  int Nb;                // batch
  int OH, OW;            // spatial dims of the GEMM-M tensor (out / in)
  int GH, GW, GC;        // dims of the gathered tensor (x for fwd, dy for dgrad)
  int R, S;              // filter
  int stride, pad;
  int M, N, K;           // GEMM sizes: K = R*S*GC
};

// Decode a flat GEMM-M row into (image, row, col).
__device__ __forceinline__ void decode_m(int m, const ConvDims& d,
                                         int& n, int& oh, int& ow) {
  ow = m % d.OW;
  const int t = m / d.OW;
  oh = t % d.OH;
  n = t / d.OH;
}

// Input row coordinate for filter tap i. MODE 0 = fwd (oh is an output pixel,
// gather from x); MODE 1/2 = dgrad (oh is an input pixel, gather from dy with
// the transpose-conv index relation; tap index i is the 180°-rotated r).
// MODE 2 is the stride-2 parity-decomposed dgrad: each block handles one
// (h%2, w%2) output class, so filter taps whose divisibility test fails for
// the whole class are skipped wholesale (see the k-loop).
template <int MODE>
__device__ __forceinline__ bool tap_coord(int oh, int i, int filt, int stride,
                                          int pad, int lim, int& ih) {
  if (MODE == 0) {
    ih = oh * stride + i - pad;
    return ih >= 0 && ih < lim;
  }
  const int t = oh + pad - filt + 1 + i;
  if (t < 0 || t % stride != 0) return false;
  ih = t / stride;
  return ih < lim;
}

// MODE 2 row decode: class-local m -> real (n, oh, ow) for parity (pa, pb).
__device__ __forceinline__ void decode_m2(int m, const ConvDims& d,
                                          int pa, int pb,
                                          int& n, int& oh, int& ow) {
  const int ow2 = d.OW >> 1, oh2 = d.OH >> 1;
  ow = (m % ow2) * 2 + pb;
  const int t = m / ow2;
  oh = (t % oh2) * 2 + pa;
  n = t / oh2;
}

// ---------------------------------------------------------------------------
// fwd / dgrad kernel: out[M][N] = gatherA[M][K] @ B[N][K]^T
// 256 threads = 4 waves as a 2x2 wave grid; per wave 64x32 via 4x2 MFMA tiles.
// ---------------------------------------------------------------------------
// BMT = GEMM-M tile (128 default; 64 doubles the block count for the late
// small-M stages so they still fill 256 CUs). MI = BMT/32 MFMA row-tiles per
// wave; EPT = BMT/4 = A-tile elements each thread stages per BK chunk.
// PADC: gathered tensor has GC==4 (channel-padded stem); d.K is R*S*4 rounded
// up to BK and the B operand is the [N][d.K] zero-padded weight, so every
// A-tile tap is one aligned dwordx2 and every B row a pair of float4s.
// BNT = GEMM-N tile: 64 default; 128 halves the number of N-tiles — and
// with it the per-pass re-reads of the gathered A operand — for the wide-N
// high-C shapes (ResNet50's 1x1 expansions), where fwd/dgrad is A-traffic
// bound.
template <int MODE, bool FAST, int BMT = BM, bool PADC = false, int BNT = BN>
__global__ __launch_bounds__(256)
void conv_igemm_kernel(const __bf16* __restrict__ Ag,
                       const __bf16* __restrict__ Bg,
                       __bf16* __restrict__ out, ConvDims d) {
  constexpr int MI = BMT / 32;
  constexpr int NI = BNT / 32;
  constexpr int EPT = BMT / 4;          // 32 or 16 elems per loader thread
  constexpr int PER_ROW = 64 / EPT;     // loader threads per A row
  constexpr int BEPT = BK * BNT / 256;  // B elems per loader thread
  __shared__ __bf16 sA[BMT * LDK];
  __shared__ __bf16 sB[BNT * LDK];

  const int tid = threadIdx.x;
  const int m0 = blockIdx.y * BMT;
  const int n0 = blockIdx.x * BNT;
  const int pa = (MODE == 2) ? (int)(blockIdx.z & 1) : 0;
  const int pb = (MODE == 2) ? (int)(blockIdx.z >> 1) : 0;

  // A loader: thread -> (row, EPT-element slice of the BK chunk)
  const int a_row = tid / PER_ROW;
  const int a_off = (tid % PER_ROW) * EPT;
  int a_n, a_oh, a_ow;
  {
    int m = m0 + a_row;
    if (m >= d.M) m = d.M - 1;  // clamped rows only feed predicated-out outputs
    if (MODE == 2) decode_m2(m, d, pa, pb, a_n, a_oh, a_ow);
    else decode_m(m, d, a_n, a_oh, a_ow);
  }
  // B loader: thread -> (row, BEPT-element slice of the BK chunk)
  const int b_row = tid & (BNT - 1);
  const int b_off = (tid / BNT) * BEPT;
  const long b_base = (long)min(n0 + b_row, d.N - 1) * d.K;

  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave >> 1) * (BMT / 2);
  const int wn = (wave & 1) * (BNT / 2);
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;

  f32x4 acc[MI][NI] = {};

  const int SC = d.S * d.GC;

  // ---- FAST-path staging helpers (explicit scalars: an array here is
  // demoted to scratch — an 80 B/lane spill measured 2-3x on the kernel) ----
  auto issue_A = [&](int kk0, float4& v0, float4& v1, float4& v2, float4& v3) {
    const int i = kk0 / SC;
    const int rem = kk0 - i * SC;
    const int j = rem / d.GC;
    const int c0 = rem - j * d.GC + a_off;
    int ih, iw;
    const bool ok = tap_coord<MODE>(a_oh, i, d.R, d.stride, d.pad, d.GH, ih)
                  & tap_coord<MODE>(a_ow, j, d.S, d.stride, d.pad, d.GW, iw);
    v0 = v1 = v2 = v3 = float4{};
    if (ok) {
      const float4* src = (const float4*)(Ag +
          (((long)a_n * d.GH + ih) * d.GW + iw) * d.GC + c0);
      v0 = src[0];
      v1 = src[1];
      if constexpr (EPT == 32) {
        v2 = src[2];
        v3 = src[3];
      }
    }
  };
  auto write_A = [&](float4 v0, float4 v1, float4 v2, float4 v3) {
    float4* dst = (float4*)(sA + a_row * LDK + a_off);
    dst[0] = v0;
    dst[1] = v1;
    if constexpr (EPT == 32) {
      dst[2] = v2;
      dst[3] = v3;
    }
  };
  // explicit scalar refs, NOT an array or struct: anything indexable (or
  // aggregate) here is demoted to scratch — the r2c3 BNT refactor's
  // float4[] cost the pipelined kernels ~1.6x (144 B/lane spill)
  auto issue_B = [&](int kk0, float4& v0, float4& v1, float4& v2, float4& v3) {
    const float4* src = (const float4*)(Bg + b_base + kk0 + b_off);
    v0 = src[0];
    v1 = src[1];
    if constexpr (BEPT == 32) {
      v2 = src[2];
      v3 = src[3];
    }
  };
  auto write_B = [&](float4 v0, float4 v1, float4 v2, float4 v3) {
    float4* dst = (float4*)(sB + b_row * LDK + b_off);
    dst[0] = v0;
    dst[1] = v1;
    if constexpr (BEPT == 32) {
      dst[2] = v2;
      dst[3] = v3;
    }
  };
  auto mfma_tile = [&]() {
    #pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      bf16x8 af[MI], bf[NI];
      #pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        af[mi] = *(const bf16x8*)&sA[(wm + mi * 16 + fr) * LDK + ks + fk];
      #pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        bf[ni] = *(const bf16x8*)&sB[(wn + ni * 16 + fr) * LDK + ks + fk];
      #pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        #pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  if constexpr (PADC) {
    const int RS = d.R * d.S;
    for (int kk0 = 0; kk0 < d.K; kk0 += BK) {
      // A tile: one aligned 8 B dwordx2 per filter tap (4 padded channels)
      #pragma unroll
      for (int e = 0; e < EPT / 4; ++e) {
        const int t4 = a_off + e * 4;
        const int tap = (kk0 + t4) >> 2;
        float2 v = {};
        if (tap < RS) {
          const int i = tap / d.S, j = tap - (tap / d.S) * d.S;
          int ih, iw;
          if (tap_coord<MODE>(a_oh, i, d.R, d.stride, d.pad, d.GH, ih) &&
              tap_coord<MODE>(a_ow, j, d.S, d.stride, d.pad, d.GW, iw))
            v = *(const float2*)(Ag +
                (((long)a_n * d.GH + ih) * d.GW + iw) * 4);
        }
        *(float2*)(sA + a_row * LDK + t4) = v;
      }
      // B tile: rows are d.K-long (BK multiple) -> always-aligned float4s
      {
        float4 b0, b1, b2, b3;
        issue_B(kk0, b0, b1, b2, b3);
        write_B(b0, b1, b2, b3);
      }
      __syncthreads();
      mfma_tile();
      __syncthreads();
    }
  } else if constexpr (FAST && MODE != 2) {
    // Register-pipelined: tile k+1's global loads issue before tile k's
    // MFMAs, so HBM/L2 latency overlaps compute; the waits land at the LDS
    // write after the barrier (write-after-barrier form).
    {
      float4 a0, a1, a2, a3, b0, b1, b2, b3;
      issue_A(0, a0, a1, a2, a3);
      issue_B(0, b0, b1, b2, b3);
      write_A(a0, a1, a2, a3);
      write_B(b0, b1, b2, b3);
    }
    __syncthreads();
    for (int kk0 = 0; kk0 < d.K; kk0 += BK) {
      const bool has_next = kk0 + BK < d.K;
      float4 a0 = {}, a1 = {}, a2 = {}, a3 = {};
      float4 b0 = {}, b1 = {}, b2 = {}, b3 = {};
      if (has_next) {
        issue_A(kk0 + BK, a0, a1, a2, a3);
        issue_B(kk0 + BK, b0, b1, b2, b3);
      }
      mfma_tile();
      if (has_next) {
        __syncthreads();
        write_A(a0, a1, a2, a3);
        write_B(b0, b1, b2, b3);
        __syncthreads();
      }
    }
  } else {
    for (int kk0 = 0; kk0 < d.K; kk0 += BK) {
      // ---- stage A tile ---------------------------------------------
      if (FAST) {  // MODE == 2 only (other FAST modes take the pipeline)
        if (MODE == 2) {
          // stride-2 divisibility is class-uniform: skip the whole tap when
          // either axis has the wrong parity (3/4 of taps for 3x3 s2 dgrad)
          const int i = kk0 / SC;
          const int rem = kk0 - i * SC;
          const int j = rem / d.GC;
          const int ta = pa + d.pad - d.R + 1 + i;
          const int tb = pb + d.pad - d.S + 1 + j;
          if ((ta & 1) || (tb & 1)) continue;
        }
        float4 a0, a1, a2, a3, b0, b1, b2, b3;
        issue_A(kk0, a0, a1, a2, a3);
        issue_B(kk0, b0, b1, b2, b3);
        write_A(a0, a1, a2, a3);
        write_B(b0, b1, b2, b3);
      } else {
        // generic gather: one element at a time (stem conv only)
        for (int e = tid; e < BMT * BK; e += 256) {
          const int row = e >> 6, kk = kk0 + (e & 63);
          __bf16 v = (__bf16)0.f;
          if (kk < d.K) {
            int m = m0 + row;
            if (m >= d.M) m = d.M - 1;
            int n, oh, ow;
            decode_m(m, d, n, oh, ow);
            const int i = kk / SC, rem = kk - i * SC;
            const int j = rem / d.GC, c = rem - j * d.GC;
            int ih, iw;
            if (tap_coord<MODE>(oh, i, d.R, d.stride, d.pad, d.GH, ih) &&
                tap_coord<MODE>(ow, j, d.S, d.stride, d.pad, d.GW, iw))
              v = Ag[(((long)n * d.GH + ih) * d.GW + iw) * d.GC + c];
          }
          sA[row * LDK + (e & 63)] = v;
        }
        for (int e = tid; e < BNT * BK; e += 256) {
          const int row = e >> 6, kk = kk0 + (e & 63);
          sB[row * LDK + (e & 63)] = (kk < d.K)
              ? Bg[(long)min(n0 + row, d.N - 1) * d.K + kk] : (__bf16)0.f;
        }
      }
      __syncthreads();
      mfma_tile();
      __syncthreads();
    }
  }

  // ---- epilogue: D row = (lane>>4)*4 + reg, col = lane&15 ----------------
  const int dm = (lane >> 4) * 4;
  const int dn = lane & 15;
  #pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      const int n = n0 + wn + ni * 16 + dn;
      if (n >= d.N) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + mi * 16 + dm + r;
        if (m >= d.M) continue;
        long om = m;
        if (MODE == 2) {
          int on, ooh, oow;
          decode_m2(m, d, pa, pb, on, ooh, oow);
          om = ((long)on * d.OH + ooh) * d.OW + oow;
        }
        out[om * d.N + n] = (__bf16)acc[mi][ni][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad kernel: dw[Kout][RSC] += dy^T[Kout][NPQ-chunk] @ im2col(x)[chunk][RSC]
// 64x64 output tile per block, 2x2 waves of 32x32; split-K over NPQ chunks.
// Each grid.z slice accumulates ITS chunks in registers and stores a private
// fp32 partial slab (no atomics — round 1's fp32 atomicAdd contention was the
// main wgrad cost at small tm*tn); wgrad_reduce_kernel sums the slabs and
// emits the bf16 channels_last weight grad in one pass. DIRECT (splits==1,
// no channel padding): the slab and reduce collapse away — the kernel writes
// bf16 dw itself. Both LDS tiles are written transposed (reduction index
// contiguous per row) so MFMA fragment reads are 16-byte ds_read_b128.
// PADC: x has 4 padded channels (stem) — per-tap dwordx2 gathers.
// ---------------------------------------------------------------------------
// ATOMIC: all grid.z slices atomicAdd into ONE fp32 slab (round 1's scheme)
// instead of private slabs — wins on small dw where the slab+reduce pass
// costs more than the contention; kept as autotune variant wtile=4.
template <bool FAST, bool PADC = false, bool DIRECT = false,
          bool ATOMIC = false>
__global__ __launch_bounds__(256)
void conv_wgrad_kernel(const __bf16* __restrict__ dy,
                       const __bf16* __restrict__ x,
                       float* __restrict__ dwp,
                       __bf16* __restrict__ dwb, ConvDims d) {
  // d: M = Kout, N = R*S*C, K = Nb*P*Q; OH/OW = P,Q; GH/GW/GC = H,W,C
  // 64(Kout) x 128(rsc) tile: each loaded byte feeds twice the MFMA work of
  // the 64x64 tile. FAST needs C % 64 == 0 so every 32-wide rsc sub-chunk
  // stays inside one (r,s) filter tap.
  __shared__ __bf16 sA[64 * LDK];   // [kout][npq]
  __shared__ __bf16 sB[128 * LDK];  // [rsc][npq]

  const int tid = threadIdx.x;
  const int m0 = blockIdx.y * 64;
  const int n0 = blockIdx.x * 128;
  const int row = tid & 63;        // npq row within the chunk
  const int grp = tid >> 6;        // 0..3

  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 64;
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;

  const int SC = d.S * d.GC;
  const int nchunks = (d.K + BK - 1) / BK;
  f32x4 acc[2][4] = {};

  for (int kc = blockIdx.z; kc < nchunks; kc += gridDim.z) {
    const int kk0 = kc * BK;
    const int npq = kk0 + row;
    int xn, xp, xq;
    decode_m(min(npq, d.K - 1), d, xn, xp, xq);
    const bool npq_ok = npq < d.K;

    // ---- dy tile, transposed into sA[kout][npq] ------------------------
    {
      const int cg = grp * 16;
      __bf16 vals[16];
      // the row bound also guards FAST: K % 16 == 0 admits K % 64 != 0
      // (e.g. K=80 via the public conv2d API), where the last tile's 16-wide
      // float4 loads would run past dy
      if (npq_ok && m0 + cg + 16 <= d.M) {
        const float4* src = (const float4*)(dy + (long)npq * d.M + m0 + cg);
        *(float4*)&vals[0] = src[0];
        *(float4*)&vals[8] = src[1];
      } else {
        #pragma unroll
        for (int e = 0; e < 16; ++e)
          vals[e] = (npq_ok && m0 + cg + e < d.M)
              ? dy[(long)npq * d.M + m0 + cg + e] : (__bf16)0.f;
      }
      #pragma unroll
      for (int e = 0; e < 16; ++e)
        sA[(cg + e) * LDK + row] = vals[e];
    }
    // ---- x tile (128 rsc cols = 4 x 32-col sub-chunks), transposed -----
    {
      const int cg = grp * 32;     // this thread's 32-col sub-chunk
      __bf16 vals[32];
      if (PADC) {
        // stem: 8 taps of 4 padded channels each, aligned dwordx2 gathers
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int nn = n0 + cg + e * 4;
          float2 v = {};
          if (npq_ok && nn < d.N) {
            const int tap = nn >> 2;
            const int i = tap / d.S, j = tap - (tap / d.S) * d.S;
            const int ih = xp * d.stride + i - d.pad;
            const int iw = xq * d.stride + j - d.pad;
            if (ih >= 0 && ih < d.GH && iw >= 0 && iw < d.GW)
              v = *(const float2*)(x +
                  (((long)xn * d.GH + ih) * d.GW + iw) * 4);
          }
          *(float2*)&vals[e * 4] = v;
        }
      } else if (FAST) {
        const int nn = n0 + cg;
        const int i = nn / SC;
        const int rem = nn - i * SC;
        const int j = rem / d.GC;
        const int c0 = rem - j * d.GC;
        const int ih = xp * d.stride + i - d.pad;
        const int iw = xq * d.stride + j - d.pad;
        if (npq_ok && ih >= 0 && ih < d.GH && iw >= 0 && iw < d.GW) {
          const float4* src = (const float4*)(x +
              (((long)xn * d.GH + ih) * d.GW + iw) * d.GC + c0);
          *(float4*)&vals[0] = src[0];
          *(float4*)&vals[8] = src[1];
          *(float4*)&vals[16] = src[2];
          *(float4*)&vals[24] = src[3];
        } else {
          #pragma unroll
          for (int e = 0; e < 32; ++e) vals[e] = (__bf16)0.f;
        }
      } else {
        #pragma unroll
        for (int e = 0; e < 32; ++e) {
          const int nn = n0 + cg + e;
          __bf16 v = (__bf16)0.f;
          if (npq_ok && nn < d.N) {
            const int i = nn / SC, rem = nn - i * SC;
            const int j = rem / d.GC, c = rem - j * d.GC;
            const int ih = xp * d.stride + i - d.pad;
            const int iw = xq * d.stride + j - d.pad;
            if (ih >= 0 && ih < d.GH && iw >= 0 && iw < d.GW)
              v = x[(((long)xn * d.GH + ih) * d.GW + iw) * d.GC + c];
          }
          vals[e] = v;
        }
      }
      #pragma unroll
      for (int e = 0; e < 32; ++e)
        sB[(cg + e) * LDK + row] = vals[e];
    }
    __syncthreads();

    #pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      bf16x8 af[2], bf[4];
      #pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        af[mi] = *(const bf16x8*)&sA[(wm + mi * 16 + fr) * LDK + ks + fk];
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = *(const bf16x8*)&sB[(wn + ni * 16 + fr) * LDK + ks + fk];
      #pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  const int dm = (lane >> 4) * 4;
  const int dn = lane & 15;
  float* slab = DIRECT ? nullptr
              : dwp + (ATOMIC ? 0L : (long)blockIdx.z * d.M * d.N);
  #pragma unroll
  for (int mi = 0; mi < 2; ++mi)
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int n = n0 + wn + ni * 16 + dn;
      if (n >= d.N) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + mi * 16 + dm + r;
        if (m >= d.M) continue;
        if (DIRECT) dwb[(long)m * d.N + n] = (__bf16)acc[mi][ni][r];
        else if (ATOMIC) atomicAdd(&slab[(long)m * d.N + n], acc[mi][ni][r]);
        else slab[(long)m * d.N + n] = acc[mi][ni][r];
      }
    }
}

// ---------------------------------------------------------------------------
// wgrad v2: large-tile variant for the big-spatial / big-channel shapes
// (ResNet50 @ 224px). Tile = (WGM*64) x (WGN*64) per 256-thread block
// ((2,2) -> 128x128, (4,1) -> 256x64): fewer K-passes over dy/x than v1's
// 64x128 (cross-tile re-reads scale with tile count), and the LDS transpose
// is done 4x4 in REGISTERS (8 v_perm per block) with ds_write_b64 stores —
// v1 staged with 48 scalar b16 LDS writes per thread per chunk, which
// rivalled the MFMA issue time. FAST-path shapes only (C % 64 == 0).
// ---------------------------------------------------------------------------

// transpose a 4x4 bf16 block: in[i] = row i as uint2 {e[i][0..1], e[i][2..3]},
// out[c] = column c. 8 v_perm_b32: selector picks [b0,b1,a0,a1]/[b2,b3,a2,a3]
// bytes of the {src1(b) low, src0(a) high} concatenation.
__device__ __forceinline__ void tr4x4_bf16(const uint2 in[4], uint2 out[4]) {
  const unsigned LO = 0x05040100u, HI = 0x07060302u;
  out[0].x = __builtin_amdgcn_perm(in[1].x, in[0].x, LO);
  out[0].y = __builtin_amdgcn_perm(in[3].x, in[2].x, LO);
  out[1].x = __builtin_amdgcn_perm(in[1].x, in[0].x, HI);
  out[1].y = __builtin_amdgcn_perm(in[3].x, in[2].x, HI);
  out[2].x = __builtin_amdgcn_perm(in[1].y, in[0].y, LO);
  out[2].y = __builtin_amdgcn_perm(in[3].y, in[2].y, LO);
  out[3].x = __builtin_amdgcn_perm(in[1].y, in[0].y, HI);
  out[3].y = __builtin_amdgcn_perm(in[3].y, in[2].y, HI);
}

// CH = BK-chunks per pipeline stage. CH=2 stages 128 npq rows so the MFMA
// phase (~64 MFMAs/wave, ~1100 cyc) covers the ~900-cyc load latency that
// left CH=1 at 54-57% SQ_WAIT_ANY (profiles/r2c13): LDS doubles (69.6 KB
// for (2,2) -> 2 blocks/CU) and staging registers double, trading
// block-level overlap for complete within-wave latency cover.
template <int WGM, int WGN, bool DIRECT, int CH = 1>
__global__ __launch_bounds__(256, 2)
void conv_wgrad_v2_kernel(const __bf16* __restrict__ dy,
                          const __bf16* __restrict__ x,
                          float* __restrict__ dwp,
                          __bf16* __restrict__ dwb, ConvDims d) {
  constexpr int TM = WGM * 64;          // kout tile
  constexpr int TN = WGN * 64;          // rsc tile
  constexpr int TA = TM / 64;
  constexpr int TB = TN / 64;
  constexpr int SK = CH * BK;           // npq rows per stage
  constexpr int SLDK = SK + 8;          // padded LDS row length
  __shared__ __bf16 sA[TM * SLDK];      // [kout][npq]
  __shared__ __bf16 sB[TN * SLDK];      // [rsc][npq]
  // DOUBLE-BUFFERED per-chunk row metadata: for npq row r, the x base
  // offset of tap (0,0), the (ih0, iw0) coords for bounds tests, and a
  // validity flag (NOT derived from the offset sign — that offset is
  // legitimately negative for pad-boundary rows of image 0). Buffer b
  // holds the NEXT chunk's rows so its gathers can issue before the
  // current chunk's MFMAs (register pipeline, as in the LIN kernel).
  __shared__ int sRowOff[2][SK];        // (xn*GH + ih0)*GW + iw0
  __shared__ short sIh0[2][SK], sIw0[2][SK];
  __shared__ unsigned char sOk[2][SK];  // npq < d.K

  const int tid = threadIdx.x;
  const int m0 = blockIdx.y * TM;
  const int n0 = blockIdx.x * TN;

  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave / WGN) * 64;
  const int wn = (wave % WGN) * 64;
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;

  const int SC = d.S * d.GC;
  const int nstages = (d.K + SK - 1) / SK;
  f32x4 acc[4][4] = {};

  uint2 sta[TA * CH][4], stb[TB * CH][4];  // staged 4x4 blocks for one stage

  auto compute_meta = [&](int kc, int b) {
    if (tid < SK) {
      const int npq = kc * SK + tid;
      int off = 0, ih0 = 0, iw0 = 0;
      const bool ok = npq < d.K && kc < nstages;
      if (ok) {
        int xn, xp, xq;
        decode_m(npq, d, xn, xp, xq);
        ih0 = xp * d.stride - d.pad;
        iw0 = xq * d.stride - d.pad;
        off = ((xn * d.GH + ih0) * d.GW + iw0);
      }
      sRowOff[b][tid] = off;
      sIh0[b][tid] = (short)ih0;
      sIw0[b][tid] = (short)iw0;
      sOk[b][tid] = ok;
    }
  };
  // thread -> 4x4 block mapping is COLUMN-group-fastest so a wave's global
  // loads are contiguous; the transposed b64 stores are XOR-swizzled on the
  // npq quad (even bits only — 8-element fragment reads stay intact).
  auto issue_all = [&](int kk0, int b) {
    #pragma unroll
    for (int t = 0; t < TA * CH; ++t) {
      const int bid = t * 256 + tid;
      const int c4 = bid & (TM / 4 - 1);
      const int r4 = bid / (TM / 4);    // 0 .. 16*CH-1
      const int npq0 = kk0 + r4 * 4;
      const int mcol = m0 + c4 * 4;
      if (npq0 + 4 <= d.K && mcol + 4 <= d.M) {
        #pragma unroll
        for (int i = 0; i < 4; ++i)
          sta[t][i] = *(const uint2*)(dy + (long)(npq0 + i) * d.M + mcol);
      } else {
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
          __bf16 e[4] = {};
          if (npq0 + i < d.K)
            #pragma unroll
            for (int c = 0; c < 4; ++c)
              if (mcol + c < d.M)
                e[c] = dy[(long)(npq0 + i) * d.M + mcol + c];
          sta[t][i] = *(uint2*)e;
        }
      }
    }
    #pragma unroll
    for (int t = 0; t < TB * CH; ++t) {
      const int bid = t * 256 + tid;
      const int c4 = bid & (TN / 4 - 1);
      const int r4 = bid / (TN / 4);
      const int nn = n0 + c4 * 4;       // rsc col of this 4-col group
      // FAST contract (C % 64 == 0): nn..nn+3 sit inside ONE filter tap
      const int i_tap = nn / SC;
      const int rem = nn - i_tap * SC;
      const int j_tap = rem / d.GC;
      const int cch = rem - j_tap * d.GC;
      const bool ncol_ok = nn + 4 <= d.N;
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int r = r4 * 4 + i;
        uint2 v = {0u, 0u};
        if (ncol_ok && sOk[b][r]) {
          const int ih = sIh0[b][r] + i_tap;
          const int iw = sIw0[b][r] + j_tap;
          if (ih >= 0 && ih < d.GH && iw >= 0 && iw < d.GW)
            v = *(const uint2*)(x +
                ((long)sRowOff[b][r] + i_tap * d.GW + j_tap) * d.GC + cch);
        }
        stb[t][i] = v;
      }
    }
  };
  // swizzle note: q ranges over 16*CH quads; the XOR touches only bits 1-3
  // (even, <16), so swizzled quads stay inside their row for any CH
  auto write_all = [&]() {
    #pragma unroll
    for (int t = 0; t < TA * CH; ++t) {
      const int bid = t * 256 + tid;
      const int c4 = bid & (TM / 4 - 1);
      const int r4 = bid / (TM / 4);
      uint2 out[4];
      tr4x4_bf16(sta[t], out);
      const int q = r4 ^ (c4 & 14);
      #pragma unroll
      for (int i = 0; i < 4; ++i)
        *(uint2*)(sA + (c4 * 4 + i) * SLDK + q * 4) = out[i];
    }
    #pragma unroll
    for (int t = 0; t < TB * CH; ++t) {
      const int bid = t * 256 + tid;
      const int c4 = bid & (TN / 4 - 1);
      const int r4 = bid / (TN / 4);
      uint2 out[4];
      tr4x4_bf16(stb[t], out);
      const int q = r4 ^ (c4 & 14);
      #pragma unroll
      for (int i = 0; i < 4; ++i)
        *(uint2*)(sB + (c4 * 4 + i) * SLDK + q * 4) = out[i];
    }
  };
  auto mfma_all = [&]() {
    #pragma unroll
    for (int ks = 0; ks < SK; ks += 32) {
      bf16x8 af[4], bf[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int row = wm + mi * 16 + fr;
        const int q = ((ks + fk) >> 2) ^ ((row >> 2) & 14);
        af[mi] = *(const bf16x8*)&sA[row * SLDK + q * 4];
      }
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int row = wn + ni * 16 + fr;
        const int q = ((ks + fk) >> 2) ^ ((row >> 2) & 14);
        bf[ni] = *(const bf16x8*)&sB[row * SLDK + q * 4];
      }
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  // pipeline: meta(k) -> loads(k) -> tiles(k); then per chunk k:
  //   loads(k+1) [covered by MFMA(k)] / MFMA(k) / write tiles(k+1) +
  //   meta(k+2) between two barriers
  bool first = true;
  int buf = 0;
  for (int kc = blockIdx.z; kc < nstages; kc += gridDim.z) {
    const int kn = kc + gridDim.z;
    if (first) {
      compute_meta(kc, 0);
      __syncthreads();
      issue_all(kc * SK, 0);
      write_all();
      compute_meta(kn, 1);
      __syncthreads();
      first = false;
      buf = 1;                          // meta[1] holds stage kn
    }
    const bool has_next = kn < nstages;
    if (has_next) issue_all(kn * SK, buf);
    mfma_all();
    if (has_next) {
      __syncthreads();
      write_all();
      compute_meta(kn + gridDim.z, buf ^ 1);
      __syncthreads();
      buf ^= 1;
    }
  }

  const int dm = (lane >> 4) * 4;
  const int dn = lane & 15;
  float* slab = DIRECT ? nullptr : dwp + (long)blockIdx.z * d.M * d.N;
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi)
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int n = n0 + wn + ni * 16 + dn;
      if (n >= d.N) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + mi * 16 + dm + r;
        if (m >= d.M) continue;
        if (DIRECT) dwb[(long)m * d.N + n] = (__bf16)acc[mi][ni][r];
        else slab[(long)m * d.N + n] = acc[mi][ni][r];
      }
    }
}

// ---------------------------------------------------------------------------
// wgrad v3 (LIN): 1x1 stride-1 pad-0 shapes — the bulk of ResNet50's wgrad
// time. im2col(x) IS x, so BOTH operands are plain row-major [npq][C]:
// no gather, no row metadata, and the whole chunk pipeline can be
// register-double-buffered like the fwd kernel (loads for chunk k+1 issue
// before chunk k's MFMAs, hiding the ~900-cycle HBM latency that left v2 at
// ~38% of peak bandwidth). Tile = (WGM*64) x (WGN*64), 2 barriers/chunk.
// ---------------------------------------------------------------------------
template <int WGM, int WGN, bool DIRECT>
__global__ __launch_bounds__(256, 2)  // 2 waves/SIMD: the double buffer
                                      // needs a partner wave per SIMD
void conv_wgrad_lin_kernel(const __bf16* __restrict__ dy,
                           const __bf16* __restrict__ x,
                           float* __restrict__ dwp,
                           __bf16* __restrict__ dwb, ConvDims d) {
  constexpr int TM = WGM * 64;
  constexpr int TN = WGN * 64;
  constexpr int TA = TM / 64;
  constexpr int TB = TN / 64;
  __shared__ __bf16 sA[TM * LDK];
  __shared__ __bf16 sB[TN * LDK];

  const int tid = threadIdx.x;
  const int m0 = blockIdx.y * TM;
  const int n0 = blockIdx.x * TN;

  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave / WGN) * 64;
  const int wn = (wave % WGN) * 64;
  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;

  const int nchunks = (d.K + BK - 1) / BK;
  f32x4 acc[4][4] = {};

  // staged 4x4 blocks for one chunk: [block][row] uint2, fully unrolled
  // (verify ScratchSize stays 0 — indexable locals here have spilled before)
  uint2 sta[TA][4], stb[TB][4];

  // per-pass block decomposition (c4-fastest => coalesced loads)
  auto load_op = [&](const __bf16* src, int ncols, int c4n, int base,
                     int kk0, int bid, uint2 (&in)[4]) {
    const int c4 = bid & (c4n - 1);
    const int r4 = bid / c4n;
    const int row0 = kk0 + r4 * 4;
    const int col = base + c4 * 4;
    if (row0 + 4 <= d.K && col + 4 <= ncols) {
      #pragma unroll
      for (int i = 0; i < 4; ++i)
        in[i] = *(const uint2*)(src + (long)(row0 + i) * ncols + col);
    } else {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        __bf16 e[4] = {};
        if (row0 + i < d.K)
          #pragma unroll
          for (int c = 0; c < 4; ++c)
            if (col + c < ncols) e[c] = src[(long)(row0 + i) * ncols + col + c];
        in[i] = *(uint2*)e;
      }
    }
  };
  auto issue_all = [&](int kk0) {
    #pragma unroll
    for (int t = 0; t < TA; ++t)
      load_op(dy, d.M, TM / 4, m0, kk0, t * 256 + tid, sta[t]);
    #pragma unroll
    for (int t = 0; t < TB; ++t)
      load_op(x, d.N, TN / 4, n0, kk0, t * 256 + tid, stb[t]);
  };
  auto write_all = [&]() {
    #pragma unroll
    for (int t = 0; t < TA; ++t) {
      const int bid = t * 256 + tid;
      const int c4 = bid & (TM / 4 - 1);
      const int r4 = bid / (TM / 4);
      uint2 out[4];
      tr4x4_bf16(sta[t], out);
      const int q = r4 ^ (c4 & 14);
      #pragma unroll
      for (int i = 0; i < 4; ++i)
        *(uint2*)(sA + (c4 * 4 + i) * LDK + q * 4) = out[i];
    }
    #pragma unroll
    for (int t = 0; t < TB; ++t) {
      const int bid = t * 256 + tid;
      const int c4 = bid & (TN / 4 - 1);
      const int r4 = bid / (TN / 4);
      uint2 out[4];
      tr4x4_bf16(stb[t], out);
      const int q = r4 ^ (c4 & 14);
      #pragma unroll
      for (int i = 0; i < 4; ++i)
        *(uint2*)(sB + (c4 * 4 + i) * LDK + q * 4) = out[i];
    }
  };
  auto mfma_all = [&]() {
    #pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      bf16x8 af[4], bf[4];
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int row = wm + mi * 16 + fr;
        const int q = ((ks + fk) >> 2) ^ ((row >> 2) & 14);
        af[mi] = *(const bf16x8*)&sA[row * LDK + q * 4];
      }
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int row = wn + ni * 16 + fr;
        const int q = ((ks + fk) >> 2) ^ ((row >> 2) & 14);
        bf[ni] = *(const bf16x8*)&sB[row * LDK + q * 4];
      }
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  bool first = true;
  for (int kc = blockIdx.z; kc < nchunks; kc += gridDim.z) {
    const int kk0 = kc * BK;
    if (first) {
      issue_all(kk0);
      write_all();
      __syncthreads();
      first = false;
    }
    const int kn = kc + gridDim.z;
    const bool has_next = kn < nchunks;
    // chunk kn's global loads issue before chunk kc's MFMAs (latency cover)
    if (has_next) issue_all(kn * BK);
    mfma_all();
    if (has_next) {
      __syncthreads();
      write_all();
      __syncthreads();
    }
  }

  const int dm = (lane >> 4) * 4;
  const int dn = lane & 15;
  float* slab = DIRECT ? nullptr : dwp + (long)blockIdx.z * d.M * d.N;
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi)
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int n = n0 + wn + ni * 16 + dn;
      if (n >= d.N) continue;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + mi * 16 + dm + r;
        if (m >= d.M) continue;
        if (DIRECT) dwb[(long)m * d.N + n] = (__bf16)acc[mi][ni][r];
        else slab[(long)m * d.N + n] = acc[mi][ni][r];
      }
    }
}

// Sum the split-K partial slabs and emit the bf16 channels_last weight grad:
// out[k][tap*Cout + c] (= memory layout of channels_last (K, Cout, R, S)).
// Cpad==Cout -> identity column map; the stem maps Cpad=4 -> Cout=3 and drops
// the zero-pad channel and any tap >= R*S.
__global__ void wgrad_reduce_kernel(const float* __restrict__ part,
                                    __bf16* __restrict__ dw,
                                    long MN, int N, int Nout, int splits,
                                    int Cpad, int Cout) {
  for (long o = (long)blockIdx.x * blockDim.x + threadIdx.x; o < MN;
       o += (long)gridDim.x * blockDim.x) {
    const long m = o / N;
    const int n = (int)(o - m * N);
    int nout = n;
    if (Cpad != Cout) {
      const int tap = n / Cpad, c = n - tap * Cpad;
      if (c >= Cout || tap * Cout >= Nout) continue;
      nout = tap * Cout + c;
    }
    float s = 0.f;
    for (int z = 0; z < splits; ++z) s += part[(long)z * MN + o];
    dw[m * Nout + nout] = (__bf16)s;
  }
}

// wT[c][i][j][k] = w[k][R-1-i][S-1-j][c] — the 180°-rotated transposed filter
// the dgrad GEMM consumes, built in ONE kernel (the ATen flip+permute+copy
// chain was 3 launches per conv backward).
__global__ void build_wT_kernel(const __bf16* __restrict__ w,
                                __bf16* __restrict__ wT,
                                int K, int R, int S, int C) {
  const long total = (long)K * R * S * C;
  for (long o = (long)blockIdx.x * blockDim.x + threadIdx.x; o < total;
       o += (long)gridDim.x * blockDim.x) {
    const int k = (int)(o % K);
    long t = o / K;
    const int j = (int)(t % S); t /= S;
    const int i = (int)(t % R);
    const int c = (int)(t / R);
    wT[o] = w[(((long)k * R + (R - 1 - i)) * S + (S - 1 - j)) * C + c];
  }
}

// x4[p][0..3] = {x[p][0..2], 0} — NHWC C=3 -> C=4 (stem fast path). Two
// pixels per thread: 3 aligned dword loads -> 2 aligned dwordx2 stores.
__global__ void pad_c3_to_c4_kernel(const __bf16* __restrict__ x,
                                    __bf16* __restrict__ x4, long npix) {
  typedef __attribute__((ext_vector_type(2))) float f32x2;
  const long pairs = (npix + 1) / 2;
  for (long o = (long)blockIdx.x * blockDim.x + threadIdx.x; o < pairs;
       o += (long)gridDim.x * blockDim.x) {
    const long p = o * 2;
    __bf16 v[8] = {};
    if (p + 2 <= npix) {
      const float* src = (const float*)(x + p * 3);   // 12 B, 4 B aligned
      *(float*)&v[0] = src[0];
      *(float*)&v[2] = src[1];   // v[2],v[3] = {x[p][2], x[p+1][0]}
      *(float*)&v[6] = src[2];
      v[4] = v[3]; v[5] = v[6]; v[6] = v[7]; v[3] = (__bf16)0.f;
      v[7] = (__bf16)0.f;
    } else {  // odd tail pixel
      for (int c = 0; c < 3; ++c) v[c] = x[p * 3 + c];
    }
    float2* dst = (float2*)(x4 + p * 4);
    dst[0] = *(float2*)&v[0];
    if (p + 2 <= npix) dst[1] = *(float2*)&v[4];
  }
}

// w4[k][tap*4 + c] = w[k][tap*3 + c] for c<3, tap<R*S; else 0.
// Kpad = R*S*4 rounded up to BK. w is channels_last (K,3,R,S) = [K][R][S][3].
__global__ void pad_w_c4_kernel(const __bf16* __restrict__ w,
                                __bf16* __restrict__ w4,
                                int K, int RS, int Kpad) {
  const long total = (long)K * Kpad;
  for (long o = (long)blockIdx.x * blockDim.x + threadIdx.x; o < total;
       o += (long)gridDim.x * blockDim.x) {
    const int k = (int)(o / Kpad);
    const int n = (int)(o - (long)k * Kpad);
    const int tap = n >> 2, c = n & 3;
    w4[o] = (tap < RS && c < 3) ? w[((long)k * RS + tap) * 3 + c]
                                : (__bf16)0.f;
  }
}

hipStream_t conv_stream() { return c10::hip::getCurrentHIPStream().stream(); }

inline void check_nhwc_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::ScalarType::BFloat16,
              name, " must be a bf16 HIP tensor");
  TORCH_CHECK(t.dim() == 4 &&
              (t.is_contiguous(at::MemoryFormat::ChannelsLast) ||
               t.is_contiguous()),
              name, " must be 4-D and dense");
}

inline const __bf16* bf16_ptr(const at::Tensor& t) {
  return reinterpret_cast<const __bf16*>(t.data_ptr());
}

}  // namespace

at::Tensor conv_build_wT(at::Tensor w) {
  check_nhwc_bf16(w, "w");
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  auto wT = at::empty({C, R, S, K}, w.options().memory_format(
                                        at::MemoryFormat::Contiguous));
  const long total = (long)K * R * S * C;
  const int blocks = (int)std::min((total + 255) / 256, (long)2048);
  hipLaunchKernelGGL(build_wT_kernel, dim3(blocks), dim3(256), 0,
                     conv_stream(), bf16_ptr(w),
                     reinterpret_cast<__bf16*>(wT.data_ptr()), K, R, S, C);
  return wT;
}

// tile: 0 = size heuristic, 64 or 128 = force that GEMM-M tile (the
// autotune cache in ops/conv.py measures both and pins the winner per shape).
namespace {
inline int pick_bmt(const ConvDims& d, long tile) {
  if (tile == 64 || tile == 128) return (int)tile;
  const long tiles128 = (long)((d.M + BM - 1) / BM) * ((d.N + BN - 1) / BN);
  return tiles128 < 384 ? 64 : 128;  // small-M: halve tile, double blocks
}
}  // namespace

// out (N,K,P,Q) channels_last <- x (N,C,H,W) channels_last, w (K,C,R,S)
// channels_last (memory [K][R][S][C]).
at::Tensor conv_fwd_igemm(at::Tensor x, at::Tensor w, long stride, long pad,
                          long tile) {
  check_nhwc_bf16(x, "x"); check_nhwc_bf16(w, "w");
  const int Nb = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C, "channel mismatch");
  const int P = (H + 2 * (int)pad - R) / (int)stride + 1;
  const int Q = (W + 2 * (int)pad - S) / (int)stride + 1;
  auto out = at::empty({Nb, K, P, Q},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = conv_stream();

  if (C == 3) {
    // stem fast path: pad x and w to 4 channels, tap-wise dwordx2 gathers
    const int RS = R * S;
    const int Kpad = ((RS * 4 + BK - 1) / BK) * BK;
    const long npix = (long)Nb * H * W;
    auto x4 = at::empty({npix * 4}, x.options());
    auto w4 = at::empty({(long)K * Kpad}, w.options());
    {
      const int blocks = (int)std::min((npix / 2 + 255) / 256, (long)4096);
      hipLaunchKernelGGL(pad_c3_to_c4_kernel, dim3(std::max(blocks, 1)),
                         dim3(256), 0, stream, bf16_ptr(x),
                         reinterpret_cast<__bf16*>(x4.data_ptr()), npix);
      const long wtotal = (long)K * Kpad;
      hipLaunchKernelGGL(pad_w_c4_kernel,
                         dim3((int)std::min((wtotal + 255) / 256, (long)1024)),
                         dim3(256), 0, stream, bf16_ptr(w),
                         reinterpret_cast<__bf16*>(w4.data_ptr()), K, RS, Kpad);
    }
    ConvDims d{Nb, P, Q, H, W, 4, R, S, (int)stride, (int)pad,
               Nb * P * Q, K, Kpad};
    const int bmt = pick_bmt(d, tile);
    const dim3 grid((d.N + BN - 1) / BN, (d.M + bmt - 1) / bmt);
    auto* kern = bmt == 64 ? conv_igemm_kernel<0, false, 64, true>
                           : conv_igemm_kernel<0, false, 128, true>;
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, stream,
                       reinterpret_cast<const __bf16*>(x4.data_ptr()),
                       reinterpret_cast<const __bf16*>(w4.data_ptr()),
                       reinterpret_cast<__bf16*>(out.data_ptr()), d);
    return out;
  }

  ConvDims d{Nb, P, Q, H, W, C, R, S, (int)stride, (int)pad,
             Nb * P * Q, K, R * S * C};
  const bool fast = (C % BK == 0);
  if (tile == 228 && fast && d.N >= 96) {
    // 128x128 tile: halves the N-tile count and with it the A re-reads
    const dim3 grid((d.N + 127) / 128, (d.M + 127) / 128);
    hipLaunchKernelGGL((conv_igemm_kernel<0, true, 128, false, 128>), grid,
                       dim3(256), 0, stream, bf16_ptr(x), bf16_ptr(w),
                       reinterpret_cast<__bf16*>(out.data_ptr()), d);
    return out;
  }
  const int bmt = pick_bmt(d, tile);
  const dim3 grid((d.N + BN - 1) / BN, (d.M + bmt - 1) / bmt);
  auto* kern = bmt == 64
      ? (fast ? conv_igemm_kernel<0, true, 64> : conv_igemm_kernel<0, false, 64>)
      : (fast ? conv_igemm_kernel<0, true> : conv_igemm_kernel<0, false>);
  hipLaunchKernelGGL(kern, grid, dim3(256), 0, stream,
                     bf16_ptr(x), bf16_ptr(w),
                     reinterpret_cast<__bf16*>(out.data_ptr()), d);
  return out;
}

// dx (N,C,H,W) channels_last <- dy (N,K,P,Q) channels_last,
// wT (C,R,S,K) CONTIGUOUS with taps 180°-rotated (built by the Python side).
at::Tensor conv_dgrad_igemm(at::Tensor dy, at::Tensor wT, long H, long W,
                            long stride, long pad, long tile) {
  check_nhwc_bf16(dy, "dy");
  TORCH_CHECK(wT.is_cuda() && wT.scalar_type() == at::ScalarType::BFloat16 &&
              wT.is_contiguous(), "wT must be contiguous bf16");
  const int Nb = dy.size(0), K = dy.size(1), P = dy.size(2), Q = dy.size(3);
  const int C = wT.size(0), R = wT.size(1), S = wT.size(2);
  TORCH_CHECK(wT.size(3) == K, "channel mismatch");
  auto dx = at::empty({Nb, C, H, W},
                      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  ConvDims d{Nb, (int)H, (int)W, P, Q, K, R, S, (int)stride, (int)pad,
             Nb * (int)H * (int)W, C, R * S * K};
  const bool fast = (K % BK == 0);
  if (fast && stride == 2 && H % 2 == 0 && W % 2 == 0) {
    // parity-decomposed: one grid.z slice per (h%2, w%2) output class,
    // class-local M; wrong-parity taps are skipped inside the kernel.
    ConvDims d2 = d;
    d2.M = Nb * (int)(H / 2) * (int)(W / 2);
    // 1x1 s2: only class (0,0) receives anything — 3/4 of dx is zero.
    // memset + one class beats 4 classes computing mostly nothing.
    const int nclass = (R == 1 && S == 1 && pad == 0) ? 1 : 4;
    if (nclass == 1) {
      const hipError_t err = hipMemsetAsync(
          dx.data_ptr(), 0, dx.numel() * dx.element_size(), conv_stream());
      TORCH_CHECK(err == hipSuccess, "hipMemsetAsync failed: ",
                  hipGetErrorString(err));
    }
    const long t128 = (long)((d2.M + BM - 1) / BM) *
                      ((d2.N + BN - 1) / BN) * nclass;
    const int bmt = (tile == 64 || tile == 128) ? (int)tile
                                                : (t128 < 384 ? 64 : 128);
    const dim3 grid((d2.N + BN - 1) / BN, (d2.M + bmt - 1) / bmt, nclass);
    auto* kern = bmt == 64 ? conv_igemm_kernel<2, true, 64>
                           : conv_igemm_kernel<2, true, 128>;
    hipLaunchKernelGGL(kern, grid, dim3(256), 0,
                       conv_stream(), bf16_ptr(dy), bf16_ptr(wT),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), d2);
    return dx;
  }
  if (tile == 228 && fast && d.N >= 96) {
    const dim3 grid((d.N + 127) / 128, (d.M + 127) / 128);
    hipLaunchKernelGGL((conv_igemm_kernel<1, true, 128, false, 128>), grid,
                       dim3(256), 0, conv_stream(), bf16_ptr(dy), bf16_ptr(wT),
                       reinterpret_cast<__bf16*>(dx.data_ptr()), d);
    return dx;
  }
  const int bmt = pick_bmt(d, tile);
  const dim3 grid((d.N + BN - 1) / BN, (d.M + bmt - 1) / bmt);
  auto* kern = bmt == 64
      ? (fast ? conv_igemm_kernel<1, true, 64> : conv_igemm_kernel<1, false, 64>)
      : (fast ? conv_igemm_kernel<1, true> : conv_igemm_kernel<1, false>);
  hipLaunchKernelGGL(kern, grid, dim3(256), 0, conv_stream(),
                     bf16_ptr(dy), bf16_ptr(wT),
                     reinterpret_cast<__bf16*>(dx.data_ptr()), d);
  return dx;
}

// dw bf16 (K,C,R,S) channels_last <- dy, x. Two-stage split-K: private fp32
// partial slabs per grid.z slice + one reduce pass (no atomics); splits==1
// non-stem collapses to a DIRECT bf16 store. `splits`: 0 = heuristic.
// `wtile`: 0 = shape heuristic, 1 = v1 64x128, 2 = v2 128x128, 3 = v2 256x64
// (the autotune cache measures and pins these per shape).
at::Tensor conv_wgrad_igemm(at::Tensor dy, at::Tensor x, long R, long S,
                            long stride, long pad, long splits_arg,
                            long wtile) {
  check_nhwc_bf16(dy, "dy"); check_nhwc_bf16(x, "x");
  const int Nb = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = dy.size(1), P = dy.size(2), Q = dy.size(3);
  auto stream = conv_stream();
  auto dw = at::empty({K, C, R, S},
                      x.options().memory_format(at::MemoryFormat::ChannelsLast));

  const bool stem = (C == 3);
  int Cpad = C, Npad = (int)(R * S * C);
  at::Tensor x4;
  const __bf16* xp = bf16_ptr(x);
  if (stem) {
    const long npix = (long)Nb * H * W;
    x4 = at::empty({npix * 4}, x.options());
    const int blocks = (int)std::min((npix / 2 + 255) / 256, (long)4096);
    hipLaunchKernelGGL(pad_c3_to_c4_kernel, dim3(std::max(blocks, 1)),
                       dim3(256), 0, stream, bf16_ptr(x),
                       reinterpret_cast<__bf16*>(x4.data_ptr()), npix);
    xp = reinterpret_cast<const __bf16*>(x4.data_ptr());
    Cpad = 4;
    Npad = (int)(R * S * 4);
  }

  ConvDims d{Nb, P, Q, H, W, Cpad, (int)R, (int)S, (int)stride, (int)pad,
             K, Npad, Nb * P * Q};
  const bool fast = (C % BK == 0) && (K % 16 == 0);

  // tile pick: v2's larger tiles read dy/x fewer times (traffic per pass
  // scales with the tile count along the other axis) and stage with wide
  // LDS writes; LIN (1x1 s1 p0: both operands plain row-major) adds the
  // register-pipelined double buffer; v1 remains for small-M shapes and
  // the generic/stem paths
  const bool lin_ok = fast && !stem && R == 1 && S == 1 &&
                      stride == 1 && pad == 0;
  int tile = (int)wtile;
  if (tile == 0) {
    if (lin_ok && d.M >= 128 && d.N >= 96) tile = 5;
    else if (lin_ok && d.M >= 256 && d.N <= 64) tile = 6;
    else if (fast && !stem && d.M >= 256 && d.N <= 64) tile = 3;
    else if (fast && !stem && d.M >= 128 && d.N >= 96) tile = 2;
    else tile = 1;
  }
  if ((tile == 5 || tile == 6) && !lin_ok) tile = tile - 3;  // 2 / 3
  if (!fast || stem) tile = (tile == 4) ? 4 : 1;  // v2 needs the FAST layout
  const int TM = (tile == 3 || tile == 6 || tile == 9) ? 256
               : (tile == 2 || tile == 5 || tile == 8) ? 128 : 64;
  const int TN = (tile == 2 || tile == 5 || tile == 8) ? 128
               : (tile == 3 || tile == 6 || tile == 9) ? 64 : 128;

  const int tm = (d.M + TM - 1) / TM, tn = (d.N + TN - 1) / TN;
  const int nchunks = (d.K + BK - 1) / BK;
  // wide-stage v2 (wtile 8/9) strides 128-row stages: a z-slice beyond the
  // stage count would leave its partial slab unwritten
  const int nzmax = (tile == 8 || tile == 9) ? (nchunks + 1) / 2 : nchunks;
  // heuristic: ~1024 blocks fills the chip, but keep >=8 chunks per block
  // (fewer and the per-block fill/drain + slab traffic dominates — measured
  // on the CIFAR shapes, where over-splitting cost 2-4x)
  int splits = (splits_arg > 0) ? (int)splits_arg
                                : std::min(1024 / (tm * tn), nchunks / 8);
  // bound the partial-slab workspace to ~96 MB
  const long max_ws = 96L * 1024 * 1024 / ((long)d.M * d.N * 4);
  splits = std::max(1, (int)std::min({(long)splits, (long)nzmax,
                                      std::max(max_ws, 1L), 1024L}));
  const dim3 grid(tn, tm, splits);

  if (splits == 1 && !stem && tile != 4) {
    auto* kern =
        tile == 9 ? conv_wgrad_v2_kernel<4, 1, true, 2>
        : tile == 8 ? conv_wgrad_v2_kernel<2, 2, true, 2>
        : tile == 6 ? conv_wgrad_lin_kernel<4, 1, true>
        : tile == 5 ? conv_wgrad_lin_kernel<2, 2, true>
        : tile == 3 ? conv_wgrad_v2_kernel<4, 1, true>
        : tile == 2 ? conv_wgrad_v2_kernel<2, 2, true>
        : (fast ? conv_wgrad_kernel<true, false, true>
                : conv_wgrad_kernel<false, false, true>);
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, stream, bf16_ptr(dy), xp,
                       nullptr, reinterpret_cast<__bf16*>(dw.data_ptr()), d);
    return dw;
  }

  const bool atomic = (tile == 4);
  const long MN = (long)d.M * d.N;
  auto part = atomic
      ? at::zeros({MN}, x.options().dtype(at::kFloat))
      : at::empty({(long)splits * MN}, x.options().dtype(at::kFloat));
  auto* kern =
      tile == 9 ? conv_wgrad_v2_kernel<4, 1, false, 2>
      : tile == 8 ? conv_wgrad_v2_kernel<2, 2, false, 2>
      : tile == 6 ? conv_wgrad_lin_kernel<4, 1, false>
      : tile == 5 ? conv_wgrad_lin_kernel<2, 2, false>
      : tile == 3 ? conv_wgrad_v2_kernel<4, 1, false>
      : tile == 2 ? conv_wgrad_v2_kernel<2, 2, false>
      : atomic ? (stem ? conv_wgrad_kernel<false, true, false, true>
                 : fast ? conv_wgrad_kernel<true, false, false, true>
                        : conv_wgrad_kernel<false, false, false, true>)
      : stem ? conv_wgrad_kernel<false, true, false>
             : (fast ? conv_wgrad_kernel<true, false, false>
                     : conv_wgrad_kernel<false, false, false>);
  hipLaunchKernelGGL(kern, grid, dim3(256), 0, stream, bf16_ptr(dy), xp,
                     part.data_ptr<float>(), nullptr, d);
  hipLaunchKernelGGL(wgrad_reduce_kernel,
                     dim3((int)std::min((MN + 255) / 256, (long)4096)),
                     dim3(256), 0, stream, part.data_ptr<float>(),
                     reinterpret_cast<__bf16*>(dw.data_ptr()), MN, d.N,
                     (int)(R * S * C), atomic ? 1 : splits, Cpad, C);
  return dw;
}
