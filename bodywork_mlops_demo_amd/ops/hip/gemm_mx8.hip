// MX-FP8 (e4m3) 256x256-tile NT GEMM for gfx950 — the 2x-rate
// low-precision path.  TWO schedules live here:
//   - gemm_mx8_nt_1p_kernel (DEFAULT): single-phase-per-tile, 1 barrier
//     per K-tile, B triple-buffered into the full 160 KB LDS — PMC
//     shows −20% wave cycles vs the 2-phase schedule at identical MFMA
//     work, measured +6-10% on every shape (profiles/r02_mx8_1p.md);
//   - gemm_mx8_nt_kernel: the original 2-phase port of gemm8.hip's
//     8-phase structure (BODYWORK_MX_1P=0), kept as the reference
//     schedule and for the supertile/XCD remap experiments.
//
// Why this shape: the non-scaled fp8 MFMAs (mfma_f32_16x16x32_fp8_fp8)
// run at the BF16 rate on CDNA4 — the ONLY 2x-rate fp8 instruction is
// the block-scaled __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4
// (K=128, per-32-element E8M0 scales, HW-fused dequant).  This kernel
// feeds it PER-TENSOR power-of-two scales: every lane passes the same
// E8M0 byte (the tensor's shared exponent), so the hardware dequant does
// the rescale for free and there is ZERO scale memory traffic in the
// K-loop — which lets the kernel keep gemm8.hip's proven 16-wave 8-phase
// structure completely unchanged in its memory system:
//
//   BK = 128 fp8 bytes/row  ==  gemm8's BK=64 bf16 bytes/row (128 B)
//   -> identical half-tile size (16 KiB), identical glds schedule,
//      identical vmcnt(2)/vmcnt(0) landing proofs, identical XOR swizzle
//      (8 x 16-B chunks per row), identical barrier structure.
//
// What changes: one K-tile is ONE K=128 MFMA per output fragment (vs
// four K=32 bf16 MFMAs), each lane's A/B fragment is 32 consecutive k
// bytes = chunks {2*kg, 2*kg+1} (two ds_read_b128), and the MFMA count
// per phase halves while each MFMA is 4x the work -> the schedule gets
// MORE compute per staged byte, i.e. deeper latency hiding than bf16.
//
// Numerics: A and B are e4m3 with per-tensor shared exponents ea, eb
// (value = 2^e * stored).  Scale operands sa = ea+127, sb = eb+127
// (E8M0); C = 2^(ea+eb) * (Aq . Bq) accumulated in fp32 by the MFMA.
// Power-of-two scaling keeps dequantisation exact.
//
// The reference computes its scoring forward in fp64 sklearn
// (stage_2_serve_model.py:78); this kernel is the opt-in low-precision
// serving/training path with MAPE parity gates in the tests.
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include "bf16_utils.h"

#define MX_BM 256
#define MX_BN 256
#define MX_BK 128           // fp8 elements per K-tile (128 bytes/row)
#define MX_THREADS 1024
#define MX_HTB (128 * 128)  // bytes per half-tile slot

typedef __attribute__((ext_vector_type(8))) int mx_i32x8;
typedef __attribute__((ext_vector_type(4))) int mx_i32x4;
typedef __attribute__((ext_vector_type(4))) float mx_f32x4;

#define MX_EPI_NONE 0
#define MX_EPI_BIAS_RELU 1
#define MX_EPI_RELU_DOT 2  // y[row] += sum_col relu(acc+bias)*w3[col]

__device__ __forceinline__ void mx_glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

// stage one [128 rows][128 B] half-tile: 1024 16-B chunks, 1024 threads
// -> ONE wave-level glds per wave (identical to gemm8's g8_stage_half).
__device__ __forceinline__ void mx_stage_half(char* __restrict__ slot,
                                              const char* __restrict__ base,
                                              int off0) {
  const int wave = threadIdx.x >> 6;
  mx_glds16(base + off0, slot + wave * 1024);
}

template <int EPI, bool HAS_BIAS, bool OUT_FP32>
__device__ __forceinline__ void mx_epilogue(
    mx_f32x4 (&acc)[4][4], const float* __restrict__ bias,
    const float* __restrict__ w3, void* __restrict__ C, long long M,
    long long N, long long m0, long long n0, int wm, int wn, int fl,
    int kg) {
  if (EPI == MX_EPI_RELU_DOT) {
    // fused scoring head: y[row] += sum_col relu(acc + b2[col]) *
    // w3[col] — the h2 activation tensor is never materialised (saves
    // its [M,N] bf16 write + the rowdot re-read, ~16 GB of HBM per 1M
    // rows at N=4096).  Per (i,r) row: 16 lanes (fl) x 4 j-frags hold
    // the wave's 64-column strip; butterfly-reduce across fl, then the
    // fl==0 lane of each kg row atomically adds its wave's partial
    // (4 wn-waves per block and N/256 blocks accumulate per row; the
    // caller zero-fills y and adds the b3 offset).
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
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long long col = n0 + wn * 64 + j * 16 + fl;
      bool col_ok = col < N;
      float bval =
          (EPI == MX_EPI_BIAS_RELU && HAS_BIAS && col_ok) ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long long row = m0 + wm * 64 + i * 16 + kg * 4 + r;
        float v = acc[i][j][r];
        if (EPI == MX_EPI_BIAS_RELU) {
          v += bval;
          v = fmaxf(v, 0.0f);
        }
        if (row < M && col_ok) {
          if (OUT_FP32)
            ((float*)C)[row * N + col] = v;
          else
            ((bf16_t*)C)[row * N + col] = f32_to_bf16(v);
        }
      }
    }
  }
}

template <int EPI, bool HAS_BIAS, bool OUT_FP32>
__launch_bounds__(MX_THREADS)
__global__ void gemm_mx8_nt_kernel(
    const unsigned char* __restrict__ A,  // [M,K] e4m3, row-major
    const unsigned char* __restrict__ B,  // [N,K] e4m3, row-major
    const float* __restrict__ bias, const float* __restrict__ w3,
    void* __restrict__ C, long long M,
    long long N, long long K, int sa, int sb,
    int super,      // supertile height in y-blocks (sa/sb: E8M0 bytes)
    int xcd_aware) {
  __shared__ char lds[8 * MX_HTB];  // ONE __shared__ object (guide trap (a))
  // Supertile blockIdx remap (super > 1): consecutive dispatch ids walk
  // an (x-cols x super-rows) PATCH instead of a full grid-x row, so the
  // ~256 concurrently-resident workgroups (1 block/CU) touch
  // super x 1 MB of A + (concurrent/super) x 1 MB of B instead of
  // re-streaming all of B per grid-y row.  At M=65536, N=K=4096 the
  // x-major order reads ~8 GB of tiles per launch (= 7.5 TB/s at the
  // measured rate — HBM-saturated); a square patch cuts that ~4-8x.
  int bx = blockIdx.x, by = blockIdx.y;
  if (super > 1) {
    int linear = blockIdx.x + blockIdx.y * gridDim.x;
    // XCD-aware transpose (env BODYWORK_MX_XCD=0 disables): dispatch
    // round-robins workgroups across the 8 XCDs (block i -> XCD i%8),
    // each with its OWN L2 — a naive patch spreads over 8 disjoint
    // L2s.  Mapping i -> (i%8)*(G/8) + i/8 gives each XCD a CONTIGUOUS
    // run of the supertile order, i.e. a compact sub-patch resident in
    // ITS L2 (guide: +10-12% when HBM-bound).
    const int G = (int)(gridDim.x * gridDim.y);
    if (xcd_aware && G % 8 == 0)
      linear = (linear % 8) * (G / 8) + (linear / 8);
    int band_sz = super * gridDim.x;
    int band = linear / band_sz;
    int in_band = linear % band_sz;
    int gy = min(super, (int)gridDim.y - band * super);
    bx = in_band / gy;
    by = band * super + in_band % gy;
  }
  const long long m0 = (long long)by * MX_BM;
  const long long n0 = (long long)bx * MX_BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 2;  // 0..3 (64 C-rows per wave)
  const int wn = wave & 3;   // 0..3 (64 C-cols per wave)
  const int fl = lane & 15;
  const int kg = lane >> 4;
  // XOR swizzle at 32-BYTE granularity — the fp8 fragment size, so a
  // fragment is two ADJACENT b128 reads (chunk32 c of row r lives at
  // (c ^ (r & 3)) * 32).  Measured equivalent to the 16-B-chunk
  // stride-2 variant: SQ_LDS_BANK_CONFLICT reads exactly 4 events per
  // v_mfma_scale under BOTH layouts (6.711e8 = 4x the MFMA count,
  // unchanged by the swizzle change) — an artifact of the scaled-MFMA
  // instruction, not real ds_read serialisation
  // (profiles/r02_mx8_pmc.md).
  const int swz = fl & 3;
  const long long nt = K / MX_BK;

#define MX_ASLOT(buf, half) (lds + ((buf) * 2 + (half)) * MX_HTB)
#define MX_BSLOT(buf, half) (lds + 4 * MX_HTB + ((buf) * 2 + (half)) * MX_HTB)

  const int a_inhalf = (wm & 1) * 64;
  const int b_inhalf = (wn & 1) * 64;

  mx_f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // one fragment = 32 consecutive k bytes = LDS chunks {2kg, 2kg+1}
  // (XOR-swizzled), read as two b128s straight into the halves of the
  // i32x8 MFMA operand (a per-element insertelement assembly costs ~5
  // v_mov per fragment — 304 v_movs per unrolled loop, measured in the
  // .s — so the loads are written through a union to land in the
  // operand's own register quadruples)
#define MX_READ8(dst, rowbase, rowoff)                                      \
  do {                                                                      \
    const char* _rb = (rowbase) + (rowoff) * 128 + ((kg ^ swz) * 32);       \
    ((mx_i32x4*)&(dst))[0] = *(const mx_i32x4*)(_rb);                       \
    ((mx_i32x4*)&(dst))[1] = *(const mx_i32x4*)(_rb + 16);                  \
  } while (0)

#define MX_AREAD(dst, buf, mfrag)                                           \
  MX_READ8(dst, MX_ASLOT(buf, (wm >> 1)), a_inhalf + (mfrag) * 16 + fl)
#define MX_BREAD(dst, buf, nfrag)                                           \
  MX_READ8(dst, MX_BSLOT(buf, (wn >> 1)), b_inhalf + (nfrag) * 16 + fl)

  mx_i32x8 a_q[2];     // this phase's two A fragments (mfrag pair)
  mx_i32x8 b_t[4];     // current tile's B fragments [nfrag]

  // per-lane staging byte offset (tile-invariant): thread ci stages the
  // 16-B half `ci&1` of 32-B chunk `(ci&7)>>1` of row `ci>>3`; the
  // 32-B chunk lands at (chunk ^ (row & 3)) * 32 in LDS, halves stay
  // adjacent, so the glds SOURCE address carries the swizzle exactly as
  // in the bf16 kernel
  int stg_off;
  {
    int ci = (int)threadIdx.x;
    int row = ci >> 3;
    int c32 = ((ci & 7) >> 1) ^ (row & 3);
    stg_off = (int)(row * K + (c32 * 32 + (ci & 1) * 16));
  }
  const char* Ah0 = (const char*)(A + m0 * K);
  const char* Ah1 = (const char*)(A + (m0 + 128) * K);
  const char* Bh0 = (const char*)(B + n0 * K);
  const char* Bh1 = (const char*)(B + (n0 + 128) * K);
#define MX_KOFF(T) ((long long)(T) * MX_BK)

  // prologue: A(0), B(0), B(1) — A(1) is issued by the loop at (0,A)
  mx_stage_half(MX_ASLOT(0, 0), Ah0, stg_off);
  mx_stage_half(MX_ASLOT(0, 1), Ah1, stg_off);
  mx_stage_half(MX_BSLOT(0, 0), Bh0, stg_off);
  mx_stage_half(MX_BSLOT(0, 1), Bh1, stg_off);
  if (nt > 1) {
    mx_stage_half(MX_BSLOT(1, 0), Bh0 + MX_KOFF(1), stg_off);
    mx_stage_half(MX_BSLOT(1, 1), Bh1 + MX_KOFF(1), stg_off);
  }

  // Two phases per K-tile, one barrier each — gemm8's v4 schedule with
  // identical glds issue order, so the same landing proof holds: at
  // wait(T,A) the newest 2 outstanding glds are B(T+1)'s halves, the
  // 3rd/4th-newest are A(T)'s -> vmcnt(2) proves tile T landed; the
  // last tile drains with vmcnt(0).  Slot safety: A(T+1) overwrites
  // A(T-1) whose mfrag2/3 reads were consumed before barrier(T,A);
  // B(T+2) overwrites B(T), consumed before barrier(T,B).
#define MX_MFMA(acc_i, afrag)                                               \
  _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                          \
      acc[acc_i][nf] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(    \
          afrag, b_t[nf], acc[acc_i][nf], 0, 0, 0, sa, 0, sb)

#define MX_PHASE_A(T, TPAR)                                                 \
  do {                                                                      \
    if ((T) + 1 < nt)                                                       \
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");                      \
    else                                                                    \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_barrier();                                           \
    if ((T) + 1 < nt) {                                                     \
      mx_stage_half(MX_ASLOT(1 - (TPAR), 0), Ah0 + MX_KOFF((T) + 1),        \
                    stg_off);                                               \
      mx_stage_half(MX_ASLOT(1 - (TPAR), 1), Ah1 + MX_KOFF((T) + 1),        \
                    stg_off);                                               \
    }                                                                       \
    _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                        \
        MX_BREAD(b_t[nf], TPAR, nf);                                        \
    MX_AREAD(a_q[0], TPAR, 0);                                              \
    MX_AREAD(a_q[1], TPAR, 1);                                              \
    __builtin_amdgcn_s_setprio(1);                                          \
    MX_MFMA(0, a_q[0]);                                                     \
    MX_MFMA(1, a_q[1]);                                                     \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

#define MX_PHASE_B(T, TPAR)                                                 \
  do {                                                                      \
    __builtin_amdgcn_s_barrier();                                           \
    if ((T) + 2 < nt) {                                                     \
      mx_stage_half(MX_BSLOT(TPAR, 0), Bh0 + MX_KOFF((T) + 2), stg_off);    \
      mx_stage_half(MX_BSLOT(TPAR, 1), Bh1 + MX_KOFF((T) + 2), stg_off);    \
    }                                                                       \
    MX_AREAD(a_q[0], TPAR, 2);                                              \
    MX_AREAD(a_q[1], TPAR, 3);                                              \
    __builtin_amdgcn_s_setprio(1);                                          \
    MX_MFMA(2, a_q[0]);                                                     \
    MX_MFMA(3, a_q[1]);                                                     \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

  // two tiles per iteration: buffer indices stay compile-time
  for (long long t = 0; t < nt; t += 2) {
    MX_PHASE_A(t, 0);
    MX_PHASE_B(t, 0);
    if (t + 1 < nt) {
      MX_PHASE_A(t + 1, 1);
      MX_PHASE_B(t + 1, 1);
    }
  }
#undef MX_PHASE_A
#undef MX_PHASE_B
#undef MX_MFMA
#undef MX_AREAD
#undef MX_BREAD
#undef MX_READ8
#undef MX_ASLOT
#undef MX_BSLOT

  mx_epilogue<EPI, HAS_BIAS, OUT_FP32>(acc, bias, w3, C, M, N, m0, n0, wm,
                                       wn, fl, kg);
}

// ---- per-tensor e4m3 quantisation ----------------------------------------
//
// x (bf16 or fp32) -> e4m3 bytes, value = 2^e * stored.  e is chosen
// host-side so |x|max / 2^e <= 448 (no saturation).  The HW fast-path
// packed convert (cvt_pk_fp8_f32) rounds to nearest-even.
template <typename T>
__global__ void quantize_e4m3_kernel(const T* __restrict__ x,
                                     unsigned char* __restrict__ out,
                                     long long n, float inv_scale) {
  long long i = (long long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i * 2 + 1 < n; i += stride) {
    float v0, v1;
    if constexpr (sizeof(T) == 2) {
      v0 = bf16_to_f32(((const bf16_t*)x)[i * 2]);
      v1 = bf16_to_f32(((const bf16_t*)x)[i * 2 + 1]);
    } else {
      v0 = ((const float*)x)[i * 2];
      v1 = ((const float*)x)[i * 2 + 1];
    }
    // packed RNE convert; lower 16 bits hold the two e4m3 bytes
    int packed = __builtin_amdgcn_cvt_pk_fp8_f32(v0 * inv_scale,
                                                 v1 * inv_scale, 0, false);
    *(unsigned short*)(out + i * 2) = (unsigned short)(packed & 0xFFFF);
  }
  // odd tail element (single-element convert via the same packed op)
  if (i * 2 < n && n % 2 == 1 && i * 2 == n - 1) {
    float v = sizeof(T) == 2 ? bf16_to_f32(((const bf16_t*)x)[n - 1])
                             : ((const float*)x)[n - 1];
    int packed =
        __builtin_amdgcn_cvt_pk_fp8_f32(v * inv_scale, 0.0f, 0, false);
    out[n - 1] = (unsigned char)(packed & 0xFF);
  }
}

// ---- single-phase-per-tile variant (env BODYWORK_MX_1P=1) -----------------
//
// ONE barrier per K-tile instead of two: A stays double-buffered, B
// goes TRIPLE-buffered (LDS = 2x32 + 3x32 = 160 KB, the full CU), so
// every overwrite stays barrier-separated with half the barrier count
// per MFMA.  Schedule per tile T (slots: A T%2, B T%3; 6-phase unroll
// so every slot index is compile-time):
//
//   wait vmcnt(2) [vmcnt(0) on the last tile]; barrier
//   issue A(T+1) -> slot (T+1)%2   [overwrites A(T-1), read last phase]
//   issue B(T+2) -> slot (T+2)%3   [overwrites B(T-1), read last phase]
//   read B(T) x4 frags + A mfrag 0,1; 8 MFMAs; read A mfrag 2,3; 8 MFMAs
//
// Landing proof: per phase the issue order is [A(T+1), B(T+2)], so at
// wait(T) the 2 newest outstanding glds are B(T+1)'s halves (issued at
// phase T-1 after A(T)) — vmcnt(2) proves A(T) and everything older
// (incl. B(T), issued at phase T-2) landed.  Near the end the guarded
// issues only SHRINK the outstanding set, and the final tile waits
// vmcnt(0).  Slot safety: both overwrites target buffers whose reads
// were consumed by the PREVIOUS phase's MFMAs, before this phase's
// barrier.
#define P1_ASLOT(buf, half) (lds + ((buf) * 2 + (half)) * MX_HTB)
#define P1_BSLOT(buf, half) (lds + (4 + (buf) * 2 + (half)) * MX_HTB)

template <int EPI, bool HAS_BIAS, bool OUT_FP32>
__launch_bounds__(MX_THREADS)
__global__ void gemm_mx8_nt_1p_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ B,
    const float* __restrict__ bias, const float* __restrict__ w3,
    void* __restrict__ C, long long M, long long N, long long K, int sa,
    int sb) {
  __shared__ char lds[10 * MX_HTB];  // A 2 slots + B 3 slots, 160 KB
  const long long m0 = (long long)blockIdx.y * MX_BM;
  const long long n0 = (long long)blockIdx.x * MX_BN;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int fl = lane & 15;
  const int kg = lane >> 4;
  const int swz = fl & 3;
  const int nt = (int)(K / MX_BK);
  const int a_inhalf = (wm & 1) * 64;
  const int b_inhalf = (wn & 1) * 64;

  mx_f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // slot indices only affect LDS ADDRESSES (never register-array
  // indices), so the A slots swap and the B slots rotate as plain
  // pointer variables — the loop body is ONE phase, which keeps
  // register pressure low (a 6-tile compile-time unroll of this
  // schedule spilled ~400 VGPRs)
  char* aw_cur = P1_ASLOT(0, 0);
  char* aw_nxt = P1_ASLOT(1, 0);
  char* bw0 = P1_BSLOT(0, 0);  // holds B(T)
  char* bw1 = P1_BSLOT(1, 0);  // holds B(T+1) (in flight or staged)
  char* bw2 = P1_BSLOT(2, 0);  // staging target for B(T+2)
  // this wave reads its half directly: precompute read bases
#define P1_AR_BASE(slotp) ((slotp) + (wm >> 1) * MX_HTB)
#define P1_BR_BASE(slotp) ((slotp) + (wn >> 1) * MX_HTB)

#define P1_READ8(dst, rowbase, rowoff)                                      \
  do {                                                                      \
    const char* _rb = (rowbase) + (rowoff) * 128 + ((kg ^ swz) * 32);       \
    ((mx_i32x4*)&(dst))[0] = *(const mx_i32x4*)(_rb);                       \
    ((mx_i32x4*)&(dst))[1] = *(const mx_i32x4*)(_rb + 16);                  \
  } while (0)

  mx_i32x8 a_q[2];
  mx_i32x8 b_t[4];

  int stg_off;
  {
    int ci = (int)threadIdx.x;
    int row = ci >> 3;
    int c32 = ((ci & 7) >> 1) ^ (row & 3);
    stg_off = (int)(row * K + (c32 * 32 + (ci & 1) * 16));
  }
  const char* Ah0 = (const char*)(A + m0 * K);
  const char* Ah1 = (const char*)(A + (m0 + 128) * K);
  const char* Bh0 = (const char*)(B + n0 * K);
  const char* Bh1 = (const char*)(B + (n0 + 128) * K);

  // prologue: A(0), B(0), B(1)
  mx_stage_half(aw_cur, Ah0, stg_off);
  mx_stage_half(aw_cur + MX_HTB, Ah1, stg_off);
  mx_stage_half(bw0, Bh0, stg_off);
  mx_stage_half(bw0 + MX_HTB, Bh1, stg_off);
  if (nt > 1) {
    mx_stage_half(bw1, Bh0 + MX_KOFF(1), stg_off);
    mx_stage_half(bw1 + MX_HTB, Bh1 + MX_KOFF(1), stg_off);
  }

#define P1_MFMA(acc_i, afrag)                                               \
  _Pragma("unroll") for (int nf = 0; nf < 4; ++nf)                          \
      acc[acc_i][nf] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(    \
          afrag, b_t[nf], acc[acc_i][nf], 0, 0, 0, sa, 0, sb)

  for (int t = 0; t < nt; ++t) {
    if (t + 1 < nt)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 1 < nt) {
      mx_stage_half(aw_nxt, Ah0 + MX_KOFF(t + 1), stg_off);
      mx_stage_half(aw_nxt + MX_HTB, Ah1 + MX_KOFF(t + 1), stg_off);
    }
    if (t + 2 < nt) {
      mx_stage_half(bw2, Bh0 + MX_KOFF(t + 2), stg_off);
      mx_stage_half(bw2 + MX_HTB, Bh1 + MX_KOFF(t + 2), stg_off);
    }
    {
      const char* br = P1_BR_BASE(bw0);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        P1_READ8(b_t[nf], br, b_inhalf + nf * 16 + fl);
    }
    {
      const char* ar = P1_AR_BASE(aw_cur);
      P1_READ8(a_q[0], ar, a_inhalf + 0 * 16 + fl);
      P1_READ8(a_q[1], ar, a_inhalf + 1 * 16 + fl);
      __builtin_amdgcn_s_setprio(1);
      P1_MFMA(0, a_q[0]);
      P1_MFMA(1, a_q[1]);
      __builtin_amdgcn_s_setprio(0);
      P1_READ8(a_q[0], ar, a_inhalf + 2 * 16 + fl);
      P1_READ8(a_q[1], ar, a_inhalf + 3 * 16 + fl);
      __builtin_amdgcn_s_setprio(1);
      P1_MFMA(2, a_q[0]);
      P1_MFMA(3, a_q[1]);
      __builtin_amdgcn_s_setprio(0);
    }
    // rotate slots: A swaps, B rotates forward
    char* tmp = aw_cur; aw_cur = aw_nxt; aw_nxt = tmp;
    char* b0 = bw0; bw0 = bw1; bw1 = bw2; bw2 = b0;
  }
#undef P1_MFMA
#undef P1_READ8
#undef P1_AR_BASE
#undef P1_BR_BASE

  mx_epilogue<EPI, HAS_BIAS, OUT_FP32>(acc, bias, w3, C, M, N, m0, n0, wm,
                                       wn, fl, kg);
}

// ---- fused rank-1 expand -> e4m3 ------------------------------------------
//
// The MLP fp8 scoring forward's layer 1: h1 = relu(x*w + b) emitted
// DIRECTLY as e4m3 bytes (value = 2^e * stored).  Unfused, the path
// writes h1 as bf16 (2 B/elem), re-reads it (2 B) and writes the fp8
// copy (1 B) — 5 bytes of HBM per element; this kernel writes 1.
template <bool HAS_BIAS>
__global__ void expand1d_e4m3_kernel(const float* __restrict__ x,
                                     const bf16_t* __restrict__ w,
                                     const bf16_t* __restrict__ b,
                                     unsigned char* __restrict__ out,
                                     long long n, int h8 /* H/8 */,
                                     float inv_scale) {
  const long long total = (long long)n * h8;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const long long row = idx / h8;
    const int c8 = (int)(idx % h8);
    float xv = x[row];
    float wv[8], bv[8];
    bf16x8_to_f32(load_bf16x8(w + c8 * 8), wv);
    if (HAS_BIAS) bf16x8_to_f32(load_bf16x8(b + c8 * 8), bv);
    unsigned long long packed = 0;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      float a0 = HAS_BIAS ? fmaf(xv, wv[2 * p], bv[2 * p]) : xv * wv[2 * p];
      float a1 = HAS_BIAS ? fmaf(xv, wv[2 * p + 1], bv[2 * p + 1])
                          : xv * wv[2 * p + 1];
      a0 = fmaxf(a0, 0.0f) * inv_scale;  // relu fused
      a1 = fmaxf(a1, 0.0f) * inv_scale;
      unsigned int pk = (unsigned int)
          __builtin_amdgcn_cvt_pk_fp8_f32(a0, a1, 0, false);
      packed |= (unsigned long long)(pk & 0xFFFFu) << (16 * p);
    }
    *(unsigned long long*)(out + row * (long long)h8 * 8 + c8 * 8) = packed;
  }
}

// ---- host wrappers --------------------------------------------------------

// supertile height (y-blocks) for the blockIdx remap; latched once.
// DEFAULT 1 (identity): the A/B matrix across supertile heights and
// the XCD transpose measured within +-2-4% of the natural dispatch
// order on every shape (profiles/r02_mx8_pmc.md) — L2 + dispatch
// overlap already cover the tile re-streaming, so the least machinery
// stays in play.  Envs kept for future A/B on other shapes.
static int mx_supertile() {
  static int v = [] {
    const char* e = getenv("BODYWORK_MX_SUPER");
    return e ? atoi(e) : 1;
  }();
  return v;
}

static int mx_xcd_aware() {
  static int v = [] {
    const char* e = getenv("BODYWORK_MX_XCD");
    return e ? atoi(e) : 0;
  }();
  return v;
}

// single-phase-per-tile variant (1 barrier/K-tile, B triple-buffered,
// 160 KB LDS).  DEFAULT since its hardware A/B won on every shape
// (+6-10%: 2042/2412/2186 TF vs 1920/2194/1998 same box) with the full
// test file green normally AND under AMD_SERIALIZE_KERNEL/COPY,
// bitwise-repeatable over 20 replays, exact at odd/clamp shapes
// (profiles/r02_mx8_1p.md).  BODYWORK_MX_1P=0 reverts to the 2-phase
// kernel.
static int mx_one_phase() {
  static int v = [] {
    const char* e = getenv("BODYWORK_MX_1P");
    return e ? atoi(e) : 1;
  }();
  return v;
}

at::Tensor expand1d_e4m3_hip(const at::Tensor& x, const at::Tensor& w,
                             const c10::optional<at::Tensor>& b,
                             int64_t e) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kFloat);
  TORCH_CHECK(w.scalar_type() == at::kBFloat16);
  const long long n = x.numel();
  const long long H = w.numel();
  TORCH_CHECK(H % 8 == 0, "expand1d_e4m3: H must be a multiple of 8");
  auto out = at::empty({n, H}, w.options().dtype(at::kByte));
  const bool has_bias = b.has_value();
  auto stream = at::cuda::getCurrentCUDAStream();
  long long total = n * (H / 8);
  int grid = (int)std::min<long long>((total + 255) / 256, 2048);
  float inv_scale = ldexpf(1.0f, (int)-e);
  const bf16_t* wp = (const bf16_t*)w.data_ptr();
  const bf16_t* bp = has_bias ? (const bf16_t*)b->data_ptr() : nullptr;
  if (has_bias)
    hipLaunchKernelGGL((expand1d_e4m3_kernel<true>), dim3(grid), dim3(256),
                       0, stream, x.data_ptr<float>(), wp, bp,
                       out.data_ptr<unsigned char>(), n, (int)(H / 8),
                       inv_scale);
  else
    hipLaunchKernelGGL((expand1d_e4m3_kernel<false>), dim3(grid), dim3(256),
                       0, stream, x.data_ptr<float>(), wp, bp,
                       out.data_ptr<unsigned char>(), n, (int)(H / 8),
                       inv_scale);
  return out;
}

at::Tensor quantize_e4m3_hip(const at::Tensor& x, int64_t e) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "quantize_e4m3: cuda contig");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 ||
                  torch::kFloat == x.scalar_type(),
              "quantize_e4m3: bf16 or fp32 input");
  auto out = torch::empty_like(x, x.options().dtype(torch::kUInt8));
  long long n = x.numel();
  if (n == 0) return out;
  float inv_scale = ldexpf(1.0f, (int)-e);
  auto stream = at::cuda::getCurrentCUDAStream();
  long long pairs = (n + 1) / 2;
  int threads = 256;
  int blocks = (int)std::min<long long>((pairs + threads - 1) / threads,
                                        8192);
  if (x.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL(quantize_e4m3_kernel<short>, dim3(blocks),
                       dim3(threads), 0, stream,
                       (const short*)x.data_ptr(),
                       out.data_ptr<unsigned char>(), n, inv_scale);
  else
    hipLaunchKernelGGL(quantize_e4m3_kernel<float>, dim3(blocks),
                       dim3(threads), 0, stream, x.data_ptr<float>(),
                       out.data_ptr<unsigned char>(), n, inv_scale);
  return out;
}

// C[M,N] = 2^(ea+eb) * (A[M,K]e4m3 . B[N,K]e4m3^T), optional fused
// bias+relu epilogue, fp32 or bf16 output.
at::Tensor gemm_mx8_nt_hip(const at::Tensor& a8, int64_t ea,
                           const at::Tensor& b8, int64_t eb,
                           const c10::optional<at::Tensor>& bias, bool relu,
                           bool out_fp32) {
  TORCH_CHECK(a8.is_cuda() && b8.is_cuda(), "gemm_mx8_nt: cuda tensors");
  TORCH_CHECK(a8.scalar_type() == torch::kUInt8 &&
                  b8.scalar_type() == torch::kUInt8,
              "gemm_mx8_nt: e4m3 byte tensors");
  TORCH_CHECK(a8.dim() == 2 && b8.dim() == 2 && a8.size(1) == b8.size(1),
              "gemm_mx8_nt: [M,K]x[N,K]");
  TORCH_CHECK(a8.is_contiguous() && b8.is_contiguous(), "contiguous");
  long long M = a8.size(0), K = a8.size(1), N = b8.size(0);
  TORCH_CHECK(M % MX_BM == 0 && N % MX_BN == 0 && K % MX_BK == 0,
              "gemm_mx8_nt requires M%256==0, N%256==0, K%128==0 (got ", M,
              "x", N, "x", K, ")");
  int sa = (int)std::min<long long>(std::max<long long>(ea + 127, 0), 254);
  int sb = (int)std::min<long long>(std::max<long long>(eb + 127, 0), 254);
  TORCH_CHECK(sa == ea + 127 && sb == eb + 127,
              "per-tensor exponent out of E8M0 range");
  auto opts = a8.options().dtype(out_fp32 ? torch::kFloat : torch::kBFloat16);
  auto C = torch::empty({M, N}, opts);
  const float* bias_p = nullptr;
  bool has_bias = false;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->is_cuda() && bias->scalar_type() == torch::kFloat &&
                    bias->numel() == N,
                "bias: fp32 [N] cuda");
    bias_p = bias->data_ptr<float>();
    has_bias = true;
  }
  dim3 grid((unsigned)(N / MX_BN), (unsigned)(M / MX_BM));
  auto stream = at::cuda::getCurrentCUDAStream();
  const unsigned char* ap = a8.data_ptr<unsigned char>();
  const unsigned char* bp = b8.data_ptr<unsigned char>();
#define LMX(EPI_, HB_, OF_)                                                 \
  do {                                                                      \
    if (mx_one_phase())                                                     \
      hipLaunchKernelGGL((gemm_mx8_nt_1p_kernel<EPI_, HB_, OF_>), grid,     \
                         dim3(MX_THREADS), 0, stream, ap, bp, bias_p,       \
                         nullptr, C.data_ptr(), M, N, K, sa, sb);           \
    else                                                                    \
      hipLaunchKernelGGL((gemm_mx8_nt_kernel<EPI_, HB_, OF_>), grid,        \
                         dim3(MX_THREADS), 0, stream, ap, bp, bias_p,       \
                         nullptr, C.data_ptr(), M, N, K, sa, sb,            \
                         mx_supertile(), mx_xcd_aware());                   \
  } while (0)
  if (relu) {
    if (has_bias) { if (out_fp32) LMX(MX_EPI_BIAS_RELU, true, true);
                    else          LMX(MX_EPI_BIAS_RELU, true, false); }
    else          { if (out_fp32) LMX(MX_EPI_BIAS_RELU, false, true);
                    else          LMX(MX_EPI_BIAS_RELU, false, false); }
  } else {
    TORCH_CHECK(!has_bias, "bias requires relu epilogue for now");
    if (out_fp32) LMX(MX_EPI_NONE, false, true);
    else          LMX(MX_EPI_NONE, false, false);
  }
#undef LMX
  return C;
}

// y[M] = sum_col relu(a8.b8^T dequant + b2) * w3  — the MLP scoring
// forward's h2 GEMM and rowdot head in ONE kernel: the [M,N] activation
// tensor is never written to HBM.  Caller adds the scalar b3.
at::Tensor gemm_mx8_relu_dot_hip(const at::Tensor& a8, int64_t ea,
                                 const at::Tensor& b8, int64_t eb,
                                 const at::Tensor& b2,
                                 const at::Tensor& w3) {
  TORCH_CHECK(a8.is_cuda() && b8.is_cuda() && a8.dim() == 2 &&
                  b8.dim() == 2 && a8.size(1) == b8.size(1),
              "gemm_mx8_relu_dot: [M,K]x[N,K] cuda");
  TORCH_CHECK(a8.scalar_type() == torch::kUInt8 &&
              b8.scalar_type() == torch::kUInt8);
  TORCH_CHECK(a8.is_contiguous() && b8.is_contiguous());
  long long M = a8.size(0), K = a8.size(1), N = b8.size(0);
  TORCH_CHECK(M % MX_BM == 0 && N % MX_BN == 0 && K % MX_BK == 0,
              "gemm_mx8_relu_dot requires M%256==0, N%256==0, K%128==0");
  TORCH_CHECK(b2.is_cuda() && b2.scalar_type() == torch::kFloat &&
                  b2.numel() == N, "b2: fp32 [N]");
  TORCH_CHECK(w3.is_cuda() && w3.scalar_type() == torch::kFloat &&
                  w3.numel() == N, "w3: fp32 [N]");
  int sa = (int)(ea + 127), sb = (int)(eb + 127);
  TORCH_CHECK(0 <= sa && sa <= 254 && 0 <= sb && sb <= 254,
              "exponent out of E8M0 range");
  auto y = torch::zeros({M}, a8.options().dtype(torch::kFloat));
  dim3 grid((unsigned)(N / MX_BN), (unsigned)(M / MX_BM));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (mx_one_phase())
    hipLaunchKernelGGL((gemm_mx8_nt_1p_kernel<MX_EPI_RELU_DOT, true, true>),
                       grid, dim3(MX_THREADS), 0, stream,
                       a8.data_ptr<unsigned char>(),
                       b8.data_ptr<unsigned char>(), b2.data_ptr<float>(),
                       w3.data_ptr<float>(), y.data_ptr(), M, N, K, sa, sb);
  else
    hipLaunchKernelGGL((gemm_mx8_nt_kernel<MX_EPI_RELU_DOT, true, true>),
                       grid, dim3(MX_THREADS), 0, stream,
                       a8.data_ptr<unsigned char>(),
                       b8.data_ptr<unsigned char>(), b2.data_ptr<float>(),
                       w3.data_ptr<float>(), y.data_ptr(), M, N, K, sa, sb,
                       mx_supertile(), mx_xcd_aware());
  return y;
}
