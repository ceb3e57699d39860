// 256x256-tile 8-phase software-pipelined NT MFMA GEMM (gfx950).
//
// The deep-pipeline variant of gemm.hip's 128^2 kernel (which sits at
// the documented plain-HIP 2-phase ceiling, ~1000 TF): 16 waves per
// 1024-thread block (4x4), per-wave C = 64x64 = 4x4 fragments; one
// K-tile (BK=64) is computed as 4 "quadrant" phases of 8 MFMAs each,
// with the operand traffic software-pipelined at HALF-TILE (16 KiB)
// granularity:
//
//   phase (u,q):  [q==0: s_waitcnt vmcnt(4), or vmcnt(0) on the last
//                  tile]                        <- once per K-tile
//                 s_barrier                     <- ONE barrier per phase
//                 issue glds (schedule below; one wave-level glds
//                   instruction per half-tile at 1024 threads)
//                 ds_read this quadrant's A fragments (2x b128) feeding
//                   the SAME phase's MFMAs [q==0: + ALL of tile u's B
//                   fragments (8); q==2: + quadrant 3 pre-read]
//                 s_setprio(1); 8 x mfma_f32_16x16x32_bf16; s_setprio(0)
//
// Issue schedule:
//   (u,1): B0(u+2)   (u,2): B1(u+2)   (u,3): A0(u+2)+A1(u+2)
// Landing proof: at (u,0) the four newest outstanding glds are tile
// u+1's {B0,B1,A0,A1}; the 5th-newest is tile u's A1 -> vmcnt(4) proves
// tile u fully landed while keeping the pipeline 4 deep (never drains;
// the final tile, with no newer issues, drains with vmcnt(0) once).
// Slot-reuse safety: every glds overwrites a slot whose last ds_read
// happened in an EARLIER phase, separated by a barrier — B(u) slots are
// read only at (u,0) and re-staged at (u,1)/(u,2); A(u) slots are last
// read at (u,2) (quadrant 3 pre-reads there) and re-staged at (u,3).
//
// LDS: per operand 2 buffers x 2 halves x [128][64] shorts = 64 KiB;
// A+B = 128 KiB -> 1 block/CU, 16 waves (4/SIMD, so the per-lane VGPR
// cap is 128).  Measured allocation: 124-126 VGPR, ZERO spills (acc 64 +
// B 32 + A 2x8 + addressing) — unlike the earlier 8-wave/128-row-per-wave
// variant which hit the 256 cap with spills.
// Same 16-byte-chunk XOR swizzle as gemm.hip, carried on the glds SOURCE
// address and the fragment read.  The K-loop is unrolled two tiles per
// iteration so every register-set index is compile-time (rule 20).
//
// Requirements: M % 256 == 0, N % 256 == 0, K % 128 == 0 (dispatched for
// those shapes only; others use the 128^2 kernel).  DEFAULT path for
// qualifying shapes — A/B-measured 1249-1292 TF vs the 128^2 kernel's
// 1011-1084 (same box, profiles/r01_gemm_variant_study.md);
// BODYWORK_GEMM_8PHASE=0 opts out.
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "bf16_utils.h"

#define G8_BM 256
#define G8_BN 256
#define G8_BK 64
#define G8_THREADS 1024
#define G8_HT (128 * 64)  // shorts per half-tile slot

typedef __attribute__((ext_vector_type(4))) float g8_f32x4;

#define G8_EPI_NONE 0
#define G8_EPI_BIAS_RELU 1
#define G8_EPI_MASK 2
// fused scoring head: y[row] += sum_col relu(acc + bias[col]) * w3[col]
// (the [M,N] activation never reaches HBM — mirrors gemm_mx8.hip's
// MX_EPI_RELU_DOT).  In this mode the `mask` kernel argument carries
// w3 (const float*) and C is y (fp32 [M], caller-zeroed).
#define G8_EPI_RELU_DOT 3

__device__ __forceinline__ void g8_glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

// stage one [128][64]-bf16 half-tile: 1024 16-B chunks, 1024 threads ->
// ONE wave-level glds per wave.  Lane-linear LDS; XOR swizzle on the
// source.  The per-lane source BYTE offset (row*K + swizzled column
// chunk) is tile-invariant — computed once at kernel entry (32-bit), so
// the hot loop's staging is {uniform base + int offset} with no 64-bit
// per-lane math (this kernel is dispatched only for M,N % 256 == 0, so
// no row clamping is needed).
__device__ __forceinline__ void g8_stage_half(short* __restrict__ slot,
                                              const char* __restrict__ base,
                                              int off0) {
  const int wave = threadIdx.x >> 6;
  g8_glds16(base + off0, (char*)slot + wave * 1024);
}

template <int EPI, bool HAS_BIAS, bool OUT_FP32, bool EMIT_MASK>
__device__ __forceinline__ void g8_epilogue(
    g8_f32x4 (&acc)[4][4], const float* __restrict__ bias,
    const unsigned char* __restrict__ mask,
    unsigned char* __restrict__ mask_out, void* __restrict__ C, long long M,
    long long N, long long m0, long long n0, int wm, int wn, int fl, int kg) {
  if (EPI == G8_EPI_RELU_DOT) {
    const float* w3 = (const float*)mask;
    float w3v[4], b2v[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long long col = n0 + wn * 64 + j * 16 + fl;
      w3v[j] = w3[col];
      b2v[j] = HAS_BIAS ? bias[col] : 0.0f;
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float part = 0.0f;
#pragma unroll
        for (int j = 0; j < 4; ++j)
          part = fmaf(fmaxf(acc[i][j][r] + b2v[j], 0.0f), w3v[j], part);
#pragma unroll
        for (int m = 1; m < 16; m <<= 1)
          part += __shfl_xor(part, m, 64);  // reduce across fl (lane&15)
        if (fl == 0) {
          long long row = m0 + wm * 64 + i * 16 + kg * 4 + r;
          atomicAdd((float*)C + row, part);
        }
      }
    }
    return;
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    unsigned long long mrow[4];
    if (EPI == G8_EPI_MASK) {
      const long long stripe = n0 + wn * 64;
      const bool full = stripe + 64 <= N;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long row = m0 + wm * 64 + i * 16 + kg * 4 + r;
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
        long long row = m0 + wm * 64 + i * 16 + kg * 4 + r;
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
  const int wm = wave >> 2;  // 0..3 (64 C-rows per wave)
  const int wn = wave & 3;   // 0..3 (64 C-cols per wave)
  const int fl = lane & 15;
  const int kg = lane >> 4;
  const int swz = fl & 7;
  const long long nt = K / G8_BK;

  // slot base (in shorts): op*4HT + buf*2HT + half*HT
#define G8_ASLOT(buf, half) (lds + ((buf) * 2 + (half)) * G8_HT)
#define G8_BSLOT(buf, half) (lds + 4 * G8_HT + ((buf) * 2 + (half)) * G8_HT)

  // wave-local read bases: this wave only touches A-half (wm>>1) and
  // B-half (wn>>1); within-half row offsets are compile-time
  const int a_inhalf = (wm & 1) * 64;
  const int b_inhalf = (wn & 1) * 64;

  g8_f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

#define G8_AREAD(dst, buf, mfrag, ks)                                       \
  dst = ((const lds_vec*)(G8_ASLOT(buf, (wm >> 1)) +                        \
                          (a_inhalf + (mfrag) * 16 + fl) * 64 +             \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v
#define G8_BREAD(dst, buf, nfrag, ks)                                       \
  dst = ((const lds_vec*)(G8_BSLOT(buf, (wn >> 1)) +                        \
                          (b_inhalf + (nfrag) * 16 + fl) * 64 +             \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v

  bf16x8_v a_q[2];     // current quadrant's A fragments [ks]
  bf16x8_v a_q3[2];    // quadrant 3, pre-read one phase early (slot-free proof)
  bf16x8_v b_t[4][2];  // current tile's B fragments [nfrag][ks]

  // per-lane staging byte offset (tile-invariant; see g8_stage_half)
  int stg_off;
  {
    int ci = (int)threadIdx.x;
    int row = ci >> 3;
    int sc = (ci & 7) ^ (row & 7);
    stg_off = (int)((row * K + sc * 8) * 2);
  }
  const char* Ah0 = (const char*)(A + m0 * K);
  const char* Ah1 = (const char*)(A + (m0 + 128) * K);
  const char* Bh0 = (const char*)(B + n0 * K);
  const char* Bh1 = (const char*)(B + (n0 + 128) * K);
#define G8_KOFF(T) ((long long)(T) * (G8_BK * 2))

  // ---- prologue: A(0), B(0), B(1) (6 half-tiles; A(1) is issued by the
  // loop at phase (0,A)).  At the first wait `vmcnt(2)` keeps the newest
  // two (tile 1's B halves) in flight and proves tile 0 landed.
  g8_stage_half(G8_ASLOT(0, 0), Ah0, stg_off);
  g8_stage_half(G8_ASLOT(0, 1), Ah1, stg_off);
  g8_stage_half(G8_BSLOT(0, 0), Bh0, stg_off);
  g8_stage_half(G8_BSLOT(0, 1), Bh1, stg_off);
  if (nt > 1) {
    g8_stage_half(G8_BSLOT(1, 0), Bh0 + G8_KOFF(1), stg_off);
    g8_stage_half(G8_BSLOT(1, 1), Bh1 + G8_KOFF(1), stg_off);
  }

  // One phase = barrier + (issue) + LDS reads + 8 MFMAs.  Reads feed the
  // SAME phase's MFMAs (lgkm waits are in-phase), except quadrant 3 which
  // is pre-read at phase 2 so that phase 3's A-slot glds issues target a
  // slot with no reads outstanding.  Issue/wait schedule per tile u:
  //   (u,0): s_waitcnt vmcnt(4); read B(u) x8 + A q0; MFMA q0
  //   (u,1): issue B0(u+2);      read A q1;           MFMA q1
  //   (u,2): issue B1(u+2);      read A q2 AND q3;    MFMA q2
  //   (u,3): issue A0,A1(u+2);                        MFMA q3
  // Landing proof: at (u,0) the newest 4 outstanding glds are tile u+1's
  // {B0,B1,A0,A1}; the 5th-newest is A1(u) -> vmcnt(4) proves all of tile
  // u landed.  Slot-free proof: every glds targets a slot whose last
  // ds_read happened in an EARLIER phase (barrier-separated): B(u) slots
  // read only at (u,0), re-staged at (u,1)/(u,2); A(u) slots last read at
  // (u,2) (q3 pre-read), re-staged at (u,3).
// Two 32-MFMA phases per K-tile (v4): half the barriers of the
// 4-phase schedule.  Phase A: wait+barrier, issue A(T+1), first-touch
// reads of B(T) x8 + A q0,q1, MFMA q0+q1 (16).  Phase B: barrier,
// issue B(T+2), read A q2,q3, MFMA q2+q3 (16).
// Landing: at wait(T,A) the newest 2 outstanding glds are B(T+1)'s
// halves (issued at (T-1,B)); the 3rd/4th-newest are A(T)'s halves
// (issued at (T-1,A)) -> vmcnt(2) proves tile T landed; the last tile
// drains with vmcnt(0).  Slot safety is completion-airtight: A(T+1)
// overwrites A(T-1), whose q2/q3 reads were CONSUMED by MFMAs before
// any wave passed barrier(T,A); B(T+2) overwrites B(T), whose reads
// were consumed by MFMA q0/q1 before barrier(T,B).
#define G8_PHASE_A(T, TPAR)                                                 \
  do {                                                                      \
    if ((T) + 1 < nt)                                                       \
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");                      \
    else                                                                    \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_barrier();                                           \
    if ((T) + 1 < nt) {                                                     \
      g8_stage_half(G8_ASLOT(1 - (TPAR), 0), Ah0 + G8_KOFF((T) + 1),        \
                    stg_off);                                               \
      g8_stage_half(G8_ASLOT(1 - (TPAR), 1), Ah1 + G8_KOFF((T) + 1),        \
                    stg_off);                                               \
    }                                                                       \
    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf) {                      \
      G8_BREAD(b_t[nf][0], TPAR, nf, 0);                                    \
      G8_BREAD(b_t[nf][1], TPAR, nf, 1);                                    \
    }                                                                       \
    G8_AREAD(a_q[0], TPAR, 0, 0);                                           \
    G8_AREAD(a_q[1], TPAR, 0, 1);                                           \
    G8_AREAD(a_q3[0], TPAR, 1, 0);                                          \
    G8_AREAD(a_q3[1], TPAR, 1, 1);                                          \
    __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                        \
        _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                    \
            acc[0][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(           \
                a_q[ks], b_t[nf][ks], acc[0][nf], 0, 0, 0);                 \
    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                        \
        _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                    \
            acc[1][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(           \
                a_q3[ks], b_t[nf][ks], acc[1][nf], 0, 0, 0);                \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

#define G8_PHASE_B(T, TPAR)                                                 \
  do {                                                                      \
    __builtin_amdgcn_s_barrier();                                           \
    if ((T) + 2 < nt) {                                                     \
      g8_stage_half(G8_BSLOT(TPAR, 0), Bh0 + G8_KOFF((T) + 2), stg_off);    \
      g8_stage_half(G8_BSLOT(TPAR, 1), Bh1 + G8_KOFF((T) + 2), stg_off);    \
    }                                                                       \
    G8_AREAD(a_q[0], TPAR, 2, 0);                                           \
    G8_AREAD(a_q[1], TPAR, 2, 1);                                           \
    G8_AREAD(a_q3[0], TPAR, 3, 0);                                          \
    G8_AREAD(a_q3[1], TPAR, 3, 1);                                          \
    __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                        \
        _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                    \
            acc[2][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(           \
                a_q[ks], b_t[nf][ks], acc[2][nf], 0, 0, 0);                 \
    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                        \
        _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                    \
            acc[3][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(           \
                a_q3[ks], b_t[nf][ks], acc[3][nf], 0, 0, 0);                \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

  // two tiles per iteration: buffer indices stay compile-time (rule 20)
  for (long long t = 0; t < nt; t += 2) {
    G8_PHASE_A(t, 0);
    G8_PHASE_B(t, 0);
    if (t + 1 < nt) {
      G8_PHASE_A(t + 1, 1);
      G8_PHASE_B(t + 1, 1);
    }
  }
#undef G8_PHASE_A
#undef G8_PHASE_B
#undef G8_AREAD
#undef G8_BREAD
#undef G8_ASLOT
#undef G8_BSLOT

  g8_epilogue<EPI, HAS_BIAS, OUT_FP32, EMIT_MASK>(
      acc, bias, mask, mask_out, C, M, N, m0, n0, wm, wn, fl, kg);
}

// ---- single-phase-per-tile variant (env BODYWORK_GEMM_1P) -----------------
//
// The bf16 port of gemm_mx8.hip's single-phase schedule: ONE barrier
// per K-tile (the 2-phase schedule above pays two), A double-buffered,
// B TRIPLE-buffered into the full 160 KB LDS.  Slot indices only
// affect LDS addresses, so the slots swap/rotate as plain pointer
// variables and the loop body stays one phase (a compile-time slot
// unroll of this schedule spilled ~400 VGPRs on the fp8 twin —
// profiles/r02_mx8_1p.md).  Landing proof: per phase the issue order
// is [A(T+1), B(T+2)], so at wait(T) the 2 newest outstanding glds are
// B(T+1)'s halves — vmcnt(2) proves A(T) and everything older landed;
// the last tile waits vmcnt(0).  Slot safety: both overwrites target
// buffers read by the PREVIOUS phase, barrier-separated.
template <int EPI, bool HAS_BIAS, bool OUT_FP32, bool EMIT_MASK = false>
__launch_bounds__(G8_THREADS)
__global__ void gemm_nt_1p_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    const float* __restrict__ bias, const unsigned char* __restrict__ mask,
    unsigned char* __restrict__ mask_out, void* __restrict__ C, long long M,
    long long N, long long K) {
  __shared__ short lds[10 * G8_HT];  // A 2 slots + B 3 slots = 160 KB
  const long long m0 = (long long)blockIdx.y * G8_BM;
  const long long n0 = (long long)blockIdx.x * G8_BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int fl = lane & 15;
  const int kg = lane >> 4;
  const int swz = fl & 7;
  const int nt = (int)(K / G8_BK);
  const int a_inhalf = (wm & 1) * 64;
  const int b_inhalf = (wn & 1) * 64;

  g8_f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  short* aw_cur = lds;                 // A slot 0 (2 halves)
  short* aw_nxt = lds + 2 * G8_HT;     // A slot 1
  short* bw0 = lds + 4 * G8_HT;        // B(T)
  short* bw1 = lds + 6 * G8_HT;        // B(T+1)
  short* bw2 = lds + 8 * G8_HT;        // staging target for B(T+2)

#define G1_READ(dst, rowbase, rowoff, ks)                                   \
  dst = ((const lds_vec*)((rowbase) + (rowoff) * 64 +                       \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v

  bf16x8_v a_q[2][2];  // current mfrag pair [mf&1][ks]
  bf16x8_v b_t[4][2];

  int stg_off;
  {
    int ci = (int)threadIdx.x;
    int row = ci >> 3;
    int sc = (ci & 7) ^ (row & 7);
    stg_off = (int)((row * K + sc * 8) * 2);
  }
  const char* Ah0 = (const char*)(A + m0 * K);
  const char* Ah1 = (const char*)(A + (m0 + 128) * K);
  const char* Bh0 = (const char*)(B + n0 * K);
  const char* Bh1 = (const char*)(B + (n0 + 128) * K);

  // prologue: A(0), B(0), B(1)
  g8_stage_half(aw_cur, Ah0, stg_off);
  g8_stage_half(aw_cur + G8_HT, Ah1, stg_off);
  g8_stage_half(bw0, Bh0, stg_off);
  g8_stage_half(bw0 + G8_HT, Bh1, stg_off);
  if (nt > 1) {
    g8_stage_half(bw1, Bh0 + G8_KOFF(1), stg_off);
    g8_stage_half(bw1 + G8_HT, Bh1 + G8_KOFF(1), stg_off);
  }

#define G1_MFMA(acc_i, apair)                                               \
  _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                          \
      _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                      \
          acc[acc_i][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(         \
              apair[ks], b_t[nf][ks], acc[acc_i][nf], 0, 0, 0)

  for (int t = 0; t < nt; ++t) {
    if (t + 1 < nt)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 1 < nt) {
      g8_stage_half(aw_nxt, Ah0 + G8_KOFF(t + 1), stg_off);
      g8_stage_half(aw_nxt + G8_HT, Ah1 + G8_KOFF(t + 1), stg_off);
    }
    if (t + 2 < nt) {
      g8_stage_half(bw2, Bh0 + G8_KOFF(t + 2), stg_off);
      g8_stage_half(bw2 + G8_HT, Bh1 + G8_KOFF(t + 2), stg_off);
    }
    {
      const short* br = bw0 + (wn >> 1) * G8_HT;
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        G1_READ(b_t[nf][0], br, b_inhalf + nf * 16 + fl, 0);
        G1_READ(b_t[nf][1], br, b_inhalf + nf * 16 + fl, 1);
      }
    }
    {
      const short* ar = aw_cur + (wm >> 1) * G8_HT;
      G1_READ(a_q[0][0], ar, a_inhalf + 0 * 16 + fl, 0);
      G1_READ(a_q[0][1], ar, a_inhalf + 0 * 16 + fl, 1);
      G1_READ(a_q[1][0], ar, a_inhalf + 1 * 16 + fl, 0);
      G1_READ(a_q[1][1], ar, a_inhalf + 1 * 16 + fl, 1);
      __builtin_amdgcn_s_setprio(1);
      G1_MFMA(0, a_q[0]);
      G1_MFMA(1, a_q[1]);
      __builtin_amdgcn_s_setprio(0);
      G1_READ(a_q[0][0], ar, a_inhalf + 2 * 16 + fl, 0);
      G1_READ(a_q[0][1], ar, a_inhalf + 2 * 16 + fl, 1);
      G1_READ(a_q[1][0], ar, a_inhalf + 3 * 16 + fl, 0);
      G1_READ(a_q[1][1], ar, a_inhalf + 3 * 16 + fl, 1);
      __builtin_amdgcn_s_setprio(1);
      G1_MFMA(2, a_q[0]);
      G1_MFMA(3, a_q[1]);
      __builtin_amdgcn_s_setprio(0);
    }
    short* tmp = aw_cur; aw_cur = aw_nxt; aw_nxt = tmp;
    short* b0 = bw0; bw0 = bw1; bw1 = bw2; bw2 = b0;
  }
#undef G1_MFMA
#undef G1_READ

  g8_epilogue<EPI, HAS_BIAS, OUT_FP32, EMIT_MASK>(
      acc, bias, mask, mask_out, C, M, N, m0, n0, wm, wn, fl, kg);
}

// ---- 8-wave / 512-thread variant (env BODYWORK_GEMM_WAVES=8) --------------
//
// Same 256^2 tile and LDS layout, but 8 waves in a 2(M)x4(N) grid with
// 128x64 output per wave: acc[8][4] = 128 VGPR at the 2-waves/SIMD /
// 256-VGPR budget (the guide's verified geometry).  B is read once per
// tile (8 b128), A in 2-mfrag pairs per phase (4 b128); peak register
// liveness ~= acc 128 + B 32 + A 2x16 + addressing ~ 220.  Half-tile
// staging needs TWO wave-glds at 512 threads (chunks tid and tid+512).
//
// Phase schedule per tile u (one barrier per 16-MFMA phase):
//   (u,0): wait vmcnt(4) [vmcnt(0) on last tile]; barrier;
//          issue A0,A1(u+1) [4 glds]; read B(u) x8 + A mf01; MFMA mf01
//   (u,1): read A mf23 (pre-barrier); barrier; issue B0(u+2); MFMA mf23
//   (u,2): read A mf45+mf67 (pre-barrier); barrier; issue B1(u+2); MFMA mf45
//   (u,3): barrier; MFMA mf67 (pre-read registers; no reads, no issues)
// Landing: at wait(u,0) the 4 newest outstanding glds are B(u+1)'s
// (2 per half x 2 halves, issued at (u-1,1)/(u-1,2)); the next-newest
// are A(u)'s 4 (issued at (u-1,0)) -> vmcnt(4) proves tile u landed.
// Slot safety mirrors the 16-wave kernel: first-touch reads post-
// barrier; A(u+1) (issued post-barrier(u,0)) overwrites A(u-1), whose
// mf45/mf67 reads were consumed by MFMAs before barrier(u,0);
// B(u+2) overwrites B(u), read at (u,0) and consumed before (u,1)'s
// barrier -> every overwrite is completion-separated by a barrier.

#define G8W_THREADS 512

__device__ __forceinline__ void g8w_stage_half(short* __restrict__ slot,
                                               const char* __restrict__ base,
                                               int off0, int off1) {
  const int wave = threadIdx.x >> 6;
  g8_glds16(base + off0, (char*)slot + wave * 1024);
  g8_glds16(base + off1, (char*)slot + 8192 + wave * 1024);
}

template <int EPI, bool HAS_BIAS, bool OUT_FP32, bool EMIT_MASK>
__device__ __forceinline__ void g8w_epilogue(
    g8_f32x4 (&acc)[8][4], const float* __restrict__ bias,
    const unsigned char* __restrict__ mask,
    unsigned char* __restrict__ mask_out, void* __restrict__ C, long long M,
    long long N, long long m0, long long n0, int wm2, int wn4, int fl,
    int kg) {
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    unsigned long long mrow[4];
    if (EPI == G8_EPI_MASK) {
      const long long stripe = n0 + wn4 * 64;
      const bool full = stripe + 64 <= N;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long row = m0 + wm2 * 128 + i * 16 + kg * 4 + r;
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
      long long col = n0 + wn4 * 64 + j * 16 + fl;
      bool col_ok = col < N;
      float bval =
          (EPI == G8_EPI_BIAS_RELU && HAS_BIAS && col_ok) ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long row = m0 + wm2 * 128 + i * 16 + kg * 4 + r;
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
__launch_bounds__(G8W_THREADS)
__global__ void gemm_nt_8phase_w8_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    const float* __restrict__ bias, const unsigned char* __restrict__ mask,
    unsigned char* __restrict__ mask_out, void* __restrict__ C, long long M,
    long long N, long long K) {
  __shared__ short lds[8 * G8_HT];
  const long long m0 = (long long)blockIdx.y * G8_BM;
  const long long n0 = (long long)blockIdx.x * G8_BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm2 = wave >> 2;  // 0..1 (128 C-rows per wave)
  const int wn4 = wave & 3;   // 0..3 (64 C-cols per wave)
  const int fl = lane & 15;
  const int kg = lane >> 4;
  const int swz = fl & 7;
  const long long nt = K / G8_BK;

#define G8W_ASLOT(buf) (lds + ((buf) * 2 + wm2) * G8_HT)
#define G8W_ASLOTH(buf, half) (lds + ((buf) * 2 + (half)) * G8_HT)
#define G8W_BSLOT(buf) (lds + 4 * G8_HT + ((buf) * 2 + (wn4 >> 1)) * G8_HT)
#define G8W_BSLOTH(buf, half) (lds + 4 * G8_HT + ((buf) * 2 + (half)) * G8_HT)
  const int b_inhalf = (wn4 & 1) * 64;

  g8_f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // wave reads its OWN A half (wm2): in-half row = mfrag*16 + fl
#define G8W_AREAD(dst, buf, mfrag, ks)                                      \
  dst = ((const lds_vec*)(G8W_ASLOT(buf) + ((mfrag) * 16 + fl) * 64 +       \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v
#define G8W_BREAD(dst, buf, nfrag, ks)                                      \
  dst = ((const lds_vec*)(G8W_BSLOT(buf) +                                  \
                          (b_inhalf + (nfrag) * 16 + fl) * 64 +             \
                          (((ks) * 4 + kg) ^ swz) * 8))                     \
            ->v

  bf16x8_v a_lo[4];    // current pair's A fragments [mfrag&1][ks]
  bf16x8_v a_hi[4];    // pre-read pair (mf67) for phase 3
  bf16x8_v b_t[4][2];  // the tile's B fragments [nfrag][ks]

  // two tile-invariant per-lane staging byte offsets (chunks tid, tid+512)
  int stg0, stg1;
  {
    int c0 = (int)threadIdx.x, c1 = c0 + 512;
    int r0 = c0 >> 3, r1 = c1 >> 3;
    stg0 = (int)((r0 * K + ((c0 & 7) ^ (r0 & 7)) * 8) * 2);
    stg1 = (int)((r1 * K + ((c1 & 7) ^ (r1 & 7)) * 8) * 2);
  }
  const char* Ah0 = (const char*)(A + m0 * K);
  const char* Ah1 = (const char*)(A + (m0 + 128) * K);
  const char* Bh0 = (const char*)(B + n0 * K);
  const char* Bh1 = (const char*)(B + (n0 + 128) * K);

  // prologue: A(0), B(0), B(1) — A(1) is issued by the loop at (0,0)
  g8w_stage_half(G8W_ASLOTH(0, 0), Ah0, stg0, stg1);
  g8w_stage_half(G8W_ASLOTH(0, 1), Ah1, stg0, stg1);
  g8w_stage_half(G8W_BSLOTH(0, 0), Bh0, stg0, stg1);
  g8w_stage_half(G8W_BSLOTH(0, 1), Bh1, stg0, stg1);
  if (nt > 1) {
    g8w_stage_half(G8W_BSLOTH(1, 0), Bh0 + G8_KOFF(1), stg0, stg1);
    g8w_stage_half(G8W_BSLOTH(1, 1), Bh1 + G8_KOFF(1), stg0, stg1);
  }

#define G8W_MFMA_PAIR(mf0, ks_arr_lo)                                       \
  _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                          \
      _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                      \
          acc[(mf0)][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(         \
              ks_arr_lo[ks], b_t[nf][ks], acc[(mf0)][nf], 0, 0, 0)

#define G8W_PHASE(T, TPAR, Q)                                               \
  do {                                                                      \
    if (Q == 1) {                                                           \
      G8W_AREAD(a_lo[0], TPAR, 2, 0);                                       \
      G8W_AREAD(a_lo[1], TPAR, 2, 1);                                       \
      G8W_AREAD(a_lo[2], TPAR, 3, 0);                                       \
      G8W_AREAD(a_lo[3], TPAR, 3, 1);                                       \
    } else if (Q == 2) {                                                    \
      G8W_AREAD(a_lo[0], TPAR, 4, 0);                                       \
      G8W_AREAD(a_lo[1], TPAR, 4, 1);                                       \
      G8W_AREAD(a_lo[2], TPAR, 5, 0);                                       \
      G8W_AREAD(a_lo[3], TPAR, 5, 1);                                       \
      G8W_AREAD(a_hi[0], TPAR, 6, 0);                                       \
      G8W_AREAD(a_hi[1], TPAR, 6, 1);                                       \
      G8W_AREAD(a_hi[2], TPAR, 7, 0);                                       \
      G8W_AREAD(a_hi[3], TPAR, 7, 1);                                       \
    }                                                                       \
    if (Q == 0) {                                                           \
      if ((T) + 1 < nt)                                                     \
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                    \
      else                                                                  \
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                    \
    }                                                                       \
    asm volatile("" ::: "memory");                                          \
    __builtin_amdgcn_s_barrier();                                           \
    if (Q == 0) {                                                           \
      if ((T) + 1 < nt) {                                                   \
        g8w_stage_half(G8W_ASLOTH(1 - (TPAR), 0), Ah0 + G8_KOFF((T) + 1),   \
                       stg0, stg1);                                         \
        g8w_stage_half(G8W_ASLOTH(1 - (TPAR), 1), Ah1 + G8_KOFF((T) + 1),   \
                       stg0, stg1);                                         \
      }                                                                     \
      _Pragma("unroll") for (int nf = 0; nf < 4; ++nf) {                    \
        G8W_BREAD(b_t[nf][0], TPAR, nf, 0);                                 \
        G8W_BREAD(b_t[nf][1], TPAR, nf, 1);                                 \
      }                                                                     \
      G8W_AREAD(a_lo[0], TPAR, 0, 0);                                       \
      G8W_AREAD(a_lo[1], TPAR, 0, 1);                                       \
      G8W_AREAD(a_lo[2], TPAR, 1, 0);                                       \
      G8W_AREAD(a_lo[3], TPAR, 1, 1);                                       \
    } else if (Q == 1) {                                                    \
      if ((T) + 2 < nt)                                                     \
        g8w_stage_half(G8W_BSLOTH(TPAR, 0), Bh0 + G8_KOFF((T) + 2), stg0,   \
                       stg1);                                               \
    } else if (Q == 2) {                                                    \
      if ((T) + 2 < nt)                                                     \
        g8w_stage_half(G8W_BSLOTH(TPAR, 1), Bh1 + G8_KOFF((T) + 2), stg0,   \
                       stg1);                                               \
    }                                                                       \
    __builtin_amdgcn_s_setprio(1);                                          \
    if (Q == 0) {                                                           \
      G8W_MFMA_PAIR(0, (a_lo + 0));                                         \
      G8W_MFMA_PAIR(1, (a_lo + 2));                                         \
    } else if (Q == 1) {                                                    \
      G8W_MFMA_PAIR(2, (a_lo + 0));                                         \
      G8W_MFMA_PAIR(3, (a_lo + 2));                                         \
    } else if (Q == 2) {                                                    \
      G8W_MFMA_PAIR(4, (a_lo + 0));                                         \
      G8W_MFMA_PAIR(5, (a_lo + 2));                                         \
    } else {                                                                \
      G8W_MFMA_PAIR(6, (a_hi + 0));                                         \
      G8W_MFMA_PAIR(7, (a_hi + 2));                                         \
    }                                                                       \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

  for (long long t = 0; t < nt; t += 2) {
    G8W_PHASE(t, 0, 0);
    G8W_PHASE(t, 0, 1);
    G8W_PHASE(t, 0, 2);
    G8W_PHASE(t, 0, 3);
    if (t + 1 < nt) {
      G8W_PHASE(t + 1, 1, 0);
      G8W_PHASE(t + 1, 1, 1);
      G8W_PHASE(t + 1, 1, 2);
      G8W_PHASE(t + 1, 1, 3);
    }
  }
#undef G8W_PHASE
#undef G8W_MFMA_PAIR
#undef G8W_AREAD
#undef G8W_BREAD
#undef G8W_ASLOT
#undef G8W_ASLOTH
#undef G8W_BSLOT
#undef G8W_BSLOTH

  g8w_epilogue<EPI, HAS_BIAS, OUT_FP32, EMIT_MASK>(
      acc, bias, mask, mask_out, C, M, N, m0, n0, wm2, wn4, fl, kg);
}

// y[M] = sum_col relu(x.w^T + b2) * w3 — the bf16 MLP scoring forward's
// h2 GEMM and rowdot head in ONE kernel (16-wave variant only; the
// [M,N] activation tensor never reaches HBM).  Caller adds b3.
at::Tensor gemm8_relu_dot_bf16_hip(const at::Tensor& x, const at::Tensor& w,
                                   const at::Tensor& b2,
                                   const at::Tensor& w3) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 2 && w.dim() == 2 &&
                  x.size(1) == w.size(1),
              "gemm8_relu_dot: [M,K]x[N,K] cuda");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  long long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M % G8_BM == 0 && N % G8_BN == 0 && K % (2 * G8_BK) == 0,
              "gemm8_relu_dot requires M%256==0, N%256==0, K%128==0");
  TORCH_CHECK(b2.is_cuda() && b2.scalar_type() == at::kFloat &&
                  b2.numel() == N, "b2: fp32 [N]");
  TORCH_CHECK(w3.is_cuda() && w3.scalar_type() == at::kFloat &&
                  w3.numel() == N, "w3: fp32 [N]");
  auto y = at::zeros({M}, x.options().dtype(at::kFloat));
  dim3 grid((unsigned)(N / G8_BN), (unsigned)(M / G8_BM));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(
      (gemm_nt_8phase_kernel<G8_EPI_RELU_DOT, true, true, false>), grid,
      dim3(G8_THREADS), 0, stream, (const bf16_t*)x.data_ptr(),
      (const bf16_t*)w.data_ptr(), b2.data_ptr<float>(),
      (const unsigned char*)w3.data_ptr<float>(), nullptr, y.data_ptr(),
      M, N, K);
  return y;
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
  static int w8 = [] {  // A/B: 8-wave/512-thread variant (2 waves/SIMD)
    const char* e = getenv("BODYWORK_GEMM_WAVES");
    return (e && e[0] == '8') ? 1 : 0;
  }();
  static int onep = [] {  // single-phase variant (see gemm_nt_1p_kernel)
    const char* e = getenv("BODYWORK_GEMM_1P");
    return e ? atoi(e) : 0;
  }();
#define L8(EPI_, HB_, OF_, EM_)                                             \
  do {                                                                      \
    if (onep)                                                               \
      hipLaunchKernelGGL((gemm_nt_1p_kernel<EPI_, HB_, OF_, EM_>), grid,    \
                         dim3(G8_THREADS), 0, stream, a, b, bias,           \
                         mask, mask_out, cp, M, N, K);                      \
    else if (w8)                                                            \
      hipLaunchKernelGGL((gemm_nt_8phase_w8_kernel<EPI_, HB_, OF_, EM_>),   \
                         grid, dim3(G8W_THREADS), 0, stream, a, b, bias,    \
                         mask, mask_out, cp, M, N, K);                      \
    else                                                                    \
      hipLaunchKernelGGL((gemm_nt_8phase_kernel<EPI_, HB_, OF_, EM_>),      \
                         grid, dim3(G8_THREADS), 0, stream, a, b, bias,     \
                         mask, mask_out, cp, M, N, K);                      \
  } while (0)
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
