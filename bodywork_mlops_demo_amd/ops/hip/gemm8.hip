// 256x256-tile 8-phase software-pipelined NT MFMA GEMM (gfx950).
//
// The deep-pipeline upgrade over gemm.hip's 128^2 kernel (which sits at
// the documented plain-HIP 2-phase ceiling, ~1000 TF): 8 waves per
// 512-thread block (2x4), per-wave C = 128x64 = 8x4 fragments; one
// K-tile (BK=64) is computed as 4 "quadrant" phases of 16 MFMAs each,
// with the operand traffic software-pipelined at HALF-TILE (16 KiB)
// granularity:
//
//   phase (t,q):  [q==3: s_waitcnt vmcnt(6)]   <- once per K-tile
//                 s_barrier                      <- ONE barrier per phase
//                 issue 1 half-tile glds (schedule below)
//                 ds_read the next quadrant's A fragments (4x b128)
//                   [q==3: + ALL of tile t+1's B fragments (8) and its
//                    quadrant-0 A fragments -> the 12-read phase]
//                 s_setprio(1); 16 x mfma_f32_16x16x32_bf16; s_setprio(0)
//
// Issue schedule (1 half-tile per phase, 2 glds each at 512 threads):
//   (t,0): A1(t+1)   (t,1): B0(t+2)   (t,2): B1(t+2)   (t,3): A0(t+2)
// Every half-tile therefore has >=3 phases of flight before its first
// ds_read, and the single vmcnt(6) per tile (newest 3 half-tiles = 6
// wave-glds stay IN FLIGHT) guarantees everything older has landed —
// the pipeline never drains.  Slot-reuse safety: a slot's last reads are
// >=1 phase before the glds that overwrites it, separated by a barrier.
//
// LDS: per operand 2 buffers x 2 halves x [128][64] shorts = 64 KiB;
// A+B = 128 KiB -> 1 block/CU, 8 waves (2/SIMD).  Same 16-byte-chunk
// XOR swizzle as gemm.hip (chunk (row,c) holds logical (row, c^(row&7))).
// The K-loop is unrolled two tiles per iteration so every register-set
// index is compile-time (guide rule 20) — hence "8 phases per iteration".
//
// Requirements: K % 128 == 0 (even tile count); M/N edges handled by
// source clamping + epilogue guards.  Dispatched for M >= 256 (training /
// batch-scoring shapes); smaller M stays on the 128^2 kernel.
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "bf16_utils.h"

#define G8_BM 256
#define G8_BN 256
#define G8_BK 64
#define G8_THREADS 512
#define G8_HT (128 * 64)  // shorts per half-tile slot

typedef __attribute__((ext_vector_type(4))) float g8_f32x4;

#define G8_EPI_NONE 0
#define G8_EPI_BIAS_RELU 1
#define G8_EPI_MASK 2

__device__ __forceinline__ void g8_glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

// stage one [128][64]-bf16 half-tile: 1024 16-B chunks, 512 threads -> 2
// wave-level glds per wave.  Lane-linear LDS; XOR swizzle on the source.
// The per-lane source BYTE offsets (row*K + swizzled column chunk) are
// tile-invariant — computed once at kernel entry (off0/off1, 32-bit) so
// the hot loop's staging is {uniform base + int offset} with no 64-bit
// per-lane math (this kernel is dispatched only for M,N % 256 == 0, so
// no row clamping is needed).
__device__ __forceinline__ void g8_stage_half(short* __restrict__ slot,
                                              const char* __restrict__ base,
                                              int off0, int off1) {
  const int wave = threadIdx.x >> 6;
  g8_glds16(base + off0, (char*)slot + (0 * 8 + wave) * 1024);
  g8_glds16(base + off1, (char*)slot + (1 * 8 + wave) * 1024);
}

template <int EPI, bool HAS_BIAS, bool OUT_FP32, bool EMIT_MASK>
__device__ __forceinline__ void g8_epilogue(
    g8_f32x4 (&acc)[8][4], const float* __restrict__ bias,
    const unsigned char* __restrict__ mask,
    unsigned char* __restrict__ mask_out, void* __restrict__ C, long long M,
    long long N, long long m0, long long n0, int wm, int wn, int fl, int kg) {
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    unsigned long long mrow[4];
    if (EPI == G8_EPI_MASK) {
      const long long stripe = n0 + wn * 64;
      const bool full = stripe + 64 <= N;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long row = m0 + wm * 128 + i * 16 + kg * 4 + r;
        if (row >= M) {
          mrow[r] = 0ull;
        } else if (full) {
          mrow[r] = *(const unsigned long long*)(mask + row * (N >> 3) +
                                                 (stripe >> 3));
        } else {
          unsigned long long v = 0;
          for (int b8 = 0; b8 < 8; ++b8)
            if (stripe + b8 * 8 < N)
              v |= (unsigned long long)
                       mask[row * (N >> 3) + ((stripe >> 3) + b8)]
                   << (8 * b8);
          mrow[r] = v;
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long long col = n0 + wn * 64 + j * 16 + fl;
      bool col_ok = col < N;
      float bval =
          (EPI == G8_EPI_BIAS_RELU && HAS_BIAS && col_ok) ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long row = m0 + wm * 128 + i * 16 + kg * 4 + r;
        bool row_ok = row < M;
        float v = acc[i][j][r];
        if (EPI == G8_EPI_BIAS_RELU) {
          v += bval;
          v = fmaxf(v, 0.0f);
        } else if (EPI == G8_EPI_MASK) {
          v = (mrow[r] >> (j * 16 + fl)) & 1 ? v : 0.0f;
        }
        if (EMIT_MASK) {
          unsigned long long b = __ballot(v > 0.0f);
          if (fl == 0 && row_ok && col < N) {
            unsigned short bits = (unsigned short)((b >> (kg * 16)) & 0xFFFF);
            *(unsigned short*)(mask_out + row * (N >> 3) + (col >> 3)) = bits;
          }
        }
        if (row_ok && col_ok) {
          if (OUT_FP32)
            ((float*)C)[row * N + col] = v;
          else
            ((bf16_t*)C)[row * N + col] = f32_to_bf16(v);
        }
      }
    }
  }
}

template <int EPI, bool HAS_BIAS, bool OUT_FP32, bool EMIT_MASK = false>
__launch_bounds__(G8_THREADS)
__global__ void gemm_nt_8phase_kernel(
    const bf16_t* __restrict__ A,  // [M,K]
    const bf16_t* __restrict__ B,  // [N,K]
    const float* __restrict__ bias, const unsigned char* __restrict__ mask,
    unsigned char* __restrict__ mask_out, void* __restrict__ C, long long M,
    long long N, long long K) {
  // ONE __shared__ object (guide §5 trap (a))
  __shared__ short lds[8 * G8_HT];  // [op A=0/B=1][buf][half][128][64]
  const long long m0 = (long long)blockIdx.y * G8_BM;
  const long long n0 = (long long)blockIdx.x * G8_BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 2;  // 0..1
  const int wn = wave & 3;   // 0..3
  const int fl = lane & 15;
  const int kg = lane >> 4;
  const int swz = fl & 7;
  const long long nt = K / G8_BK;

  // slot base (in shorts): op*4HT + buf*2HT + half*HT
#define G8_ASLOT(buf, half) (lds + ((buf) * 2 + (half)) * G8_HT)
#define G8_BSLOT(buf, half) (lds + 4 * G8_HT + ((buf) * 2 + (half)) * G8_HT)

  // wave-local read bases: this wave only ever touches A-half wm and
  // B-half (wn>>1); within-half row offsets are compile-time per fragment
  const int b_inhalf = (wn & 1) * 64;  // col base within the B half

  g8_f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // fragment read helpers (compile-time q/ks/set indices at call sites)
#define G8_AREAD(dst, buf, mfrag, ks)                                       \
  dst = ((const lds_vec*)(G8_ASLOT(buf, wm) + ((mfrag) * 16 + fl) * 64 +    \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v
#define G8_BREAD(dst, buf, nfrag, ks)                                       \
  dst = ((const lds_vec*)(G8_BSLOT(buf, (wn >> 1)) +                        \
                          (b_inhalf + (nfrag) * 16 + fl) * 64 +             \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v

  bf16x8_v a_set0[2][2], a_set1[2][2];  // [mi][ks] — two quad sets
  bf16x8_v bA[4][2], bB[4][2];          // [nfrag][ks] — even/odd tile B

  // per-lane staging byte offsets (tile-invariant; see g8_stage_half)
  int stg_off[2];
  {
    const int lane_ = threadIdx.x & 63;
    const int wave_ = threadIdx.x >> 6;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      int ci = (s * 8 + wave_) * 64 + lane_;
      int row = ci >> 3;
      int sc = (ci & 7) ^ (row & 7);
      stg_off[s] = (int)((row * K + sc * 8) * 2);
    }
  }
  // uniform base pointers: A halves at rows m0 / m0+128, B at n0 / n0+128
  const char* Ah0 = (const char*)(A + m0 * K);
  const char* Ah1 = (const char*)(A + (m0 + 128) * K);
  const char* Bh0 = (const char*)(B + n0 * K);
  const char* Bh1 = (const char*)(B + (n0 + 128) * K);
#define G8_KOFF(T) ((long long)(T) * (G8_BK * 2))

  // ---- prologue: 7 half-tiles, then land tile 0 -------------------------
  g8_stage_half(G8_ASLOT(0, 0), Ah0, stg_off[0], stg_off[1]);
  g8_stage_half(G8_ASLOT(0, 1), Ah1, stg_off[0], stg_off[1]);
  g8_stage_half(G8_BSLOT(0, 0), Bh0, stg_off[0], stg_off[1]);
  g8_stage_half(G8_BSLOT(0, 1), Bh1, stg_off[0], stg_off[1]);
  if (nt > 1) {
    g8_stage_half(G8_BSLOT(1, 0), Bh0 + G8_KOFF(1), stg_off[0], stg_off[1]);
    g8_stage_half(G8_BSLOT(1, 1), Bh1 + G8_KOFF(1), stg_off[0], stg_off[1]);
    g8_stage_half(G8_ASLOT(1, 0), Ah0 + G8_KOFF(1), stg_off[0], stg_off[1]);
  }
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");  // tile 0 landed
  __builtin_amdgcn_s_barrier();
  // boundary-style reads for tile 0: all B(0) + A(0) quad 0 -> set 0
#pragma unroll
  for (int nf = 0; nf < 4; ++nf) {
    G8_BREAD(bA[nf][0], 0, nf, 0);
    G8_BREAD(bA[nf][1], 0, nf, 1);
  }
  G8_AREAD(a_set0[0][0], 0, 0, 0);
  G8_AREAD(a_set0[0][1], 0, 0, 1);
  G8_AREAD(a_set0[1][0], 0, 1, 0);
  G8_AREAD(a_set0[1][1], 0, 1, 1);

  // one phase: issue -> reads -> MFMA.  T: tile index (runtime), TPAR:
  // compile-time tile parity (buffers/B-set), Q: compile-time quadrant.
  // a_rd / a_mm: the A-register sets to read-into / mfma-from.
#define G8_PHASE(T, TPAR, Q, BCUR, BNXT, A_RD, A_MM)                        \
  do {                                                                      \
    /* vmcnt(4): newest 2 half-tiles (4 wave-glds) stay in flight; the    \
       3rd-newest is A1(T+1) issued at (T,0), which THIS phase's boundary \
       reads consume — it must be landed (vmcnt(6) would not pin it) */   \
    if (Q == 3) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");            \
    __builtin_amdgcn_s_barrier();                                           \
    /* issue schedule */                                                    \
    if (Q == 0) {                                                           \
      if ((T) + 1 < nt)                                                     \
        g8_stage_half(G8_ASLOT(((TPAR) ^ 1), 1), Ah1 + G8_KOFF((T) + 1),    \
                      stg_off[0], stg_off[1]);                              \
    } else if (Q == 1) {                                                    \
      if ((T) + 2 < nt)                                                     \
        g8_stage_half(G8_BSLOT(TPAR, 0), Bh0 + G8_KOFF((T) + 2),            \
                      stg_off[0], stg_off[1]);                              \
    } else if (Q == 2) {                                                    \
      if ((T) + 2 < nt)                                                     \
        g8_stage_half(G8_BSLOT(TPAR, 1), Bh1 + G8_KOFF((T) + 2),            \
                      stg_off[0], stg_off[1]);                              \
    } else {                                                                \
      if ((T) + 2 < nt)                                                     \
        g8_stage_half(G8_ASLOT(TPAR, 0), Ah0 + G8_KOFF((T) + 2),            \
                      stg_off[0], stg_off[1]);                              \
    }                                                                       \
    /* reads for the NEXT phase */                                          \
    if (Q < 3) {                                                            \
      G8_AREAD(A_RD[0][0], TPAR, 2 * (Q + 1), 0);                           \
      G8_AREAD(A_RD[0][1], TPAR, 2 * (Q + 1), 1);                           \
      G8_AREAD(A_RD[1][0], TPAR, 2 * (Q + 1) + 1, 0);                       \
      G8_AREAD(A_RD[1][1], TPAR, 2 * (Q + 1) + 1, 1);                       \
    } else if ((T) + 1 < nt) { /* boundary: B(T+1) + A(T+1) quad 0 */       \
      _Pragma("unroll") for (int nf = 0; nf < 4; ++nf) {                    \
        G8_BREAD(BNXT[nf][0], ((TPAR) ^ 1), nf, 0);                         \
        G8_BREAD(BNXT[nf][1], ((TPAR) ^ 1), nf, 1);                         \
      }                                                                     \
      G8_AREAD(A_RD[0][0], ((TPAR) ^ 1), 0, 0);                             \
      G8_AREAD(A_RD[0][1], ((TPAR) ^ 1), 0, 1);                             \
      G8_AREAD(A_RD[1][0], ((TPAR) ^ 1), 1, 0);                             \
      G8_AREAD(A_RD[1][1], ((TPAR) ^ 1), 1, 1);                             \
    }                                                                       \
    /* 16 MFMAs: quadrant Q (m-frags 2Q, 2Q+1) x 4 n-frags x 2 ks */        \
    __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int mi = 0; mi < 2; ++mi)                        \
        _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                    \
        _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                    \
            acc[2 * (Q) + mi][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16( \
                A_MM[mi][ks], BCUR[nf][ks], acc[2 * (Q) + mi][nf], 0, 0, 0); \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

  // main loop: 2 K-tiles (8 phases) per iteration; nt is even (K%128==0)
  for (long long t = 0; t < nt; t += 2) {
    G8_PHASE(t, 0, 0, bA, bB, a_set1, a_set0);
    G8_PHASE(t, 0, 1, bA, bB, a_set0, a_set1);
    G8_PHASE(t, 0, 2, bA, bB, a_set1, a_set0);
    G8_PHASE(t, 0, 3, bA, bB, a_set0, a_set1);
    G8_PHASE(t + 1, 1, 0, bB, bA, a_set1, a_set0);
    G8_PHASE(t + 1, 1, 1, bB, bA, a_set0, a_set1);
    G8_PHASE(t + 1, 1, 2, bB, bA, a_set1, a_set0);
    G8_PHASE(t + 1, 1, 3, bB, bA, a_set0, a_set1);
  }
#undef G8_PHASE
#undef G8_AREAD
#undef G8_BREAD
#undef G8_ASLOT
#undef G8_BSLOT

  g8_epilogue<EPI, HAS_BIAS, OUT_FP32, EMIT_MASK>(
      acc, bias, mask, mask_out, C, M, N, m0, n0, wm, wn, fl, kg);
}

// ---- launcher (called from gemm.hip's dispatch) ---------------------------
void launch_gemm8(int epi, bool has_bias, bool out_fp32, bool emit_mask,
                  const void* ap, const void* bp, const float* bias,
                  const unsigned char* mask, unsigned char* mask_out,
                  void* cp, long long M, long long N, long long K) {
  dim3 grid((unsigned)((N + G8_BN - 1) / G8_BN),
            (unsigned)((M + G8_BM - 1) / G8_BM));
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)ap;
  const bf16_t* b = (const bf16_t*)bp;
#define L8(EPI_, HB_, OF_, EM_)                                             \
  hipLaunchKernelGGL((gemm_nt_8phase_kernel<EPI_, HB_, OF_, EM_>), grid,    \
                     dim3(G8_THREADS), 0, stream, a, b, bias, mask,         \
                     mask_out, cp, M, N, K)
  if (epi == G8_EPI_BIAS_RELU && emit_mask) {
    if (has_bias) L8(G8_EPI_BIAS_RELU, true, false, true);
    else          L8(G8_EPI_BIAS_RELU, false, false, true);
  } else if (epi == G8_EPI_BIAS_RELU) {
    if (has_bias) { if (out_fp32) L8(G8_EPI_BIAS_RELU, true, true, false);
                    else          L8(G8_EPI_BIAS_RELU, true, false, false); }
    else          { if (out_fp32) L8(G8_EPI_BIAS_RELU, false, true, false);
                    else          L8(G8_EPI_BIAS_RELU, false, false, false); }
  } else if (epi == G8_EPI_MASK) {
    if (out_fp32) L8(G8_EPI_MASK, false, true, false);
    else          L8(G8_EPI_MASK, false, false, false);
  } else {
    if (out_fp32) L8(G8_EPI_NONE, false, true, false);
    else          L8(G8_EPI_NONE, false, false, false);
  }
#undef L8
}
