// Flash attention forward + backward for gfx950 / CDNA4.
//
// Replaces the reference's paddle `flash_attention` consumption
// (ppfleetx hybrid_model.py:284-301) with hand-written MFMA kernels.
//
// Forward: 8-wave workgroup, 256-row Q tile (RB=2 16-row groups/wave),
// 64-row KV tiles double-buffered in LDS with split staging (issue
// global->reg loads BEFORE the tile's compute, ds_write after — guide
// T14): HBM latency for tile t+1 hides under tile t's MFMAs, one
// __syncthreads per tile. QK^T runs with SWAPPED operands (mfma(K, Q) ->
// C[kv][q]) so each lane's 16 scores belong to ONE q row: the online
// softmax state is a per-lane scalar, row reductions are 2 shuffles, and
// P lands in the per-wave LDS image with two b32 stores per 16-col tile.
// bh rides blockIdx.x so the linear dispatch deals every CU one block of
// each causal tile position (load balance). hardware v_exp (__expf);
// mfma_f32_16x16x32_bf16 tiles; K row-major +16B-padded (conflict-free
// b128 reads), V transposed at stage time.
//
// Backward: split into dKV kernel (parallel over kv tiles) and dQ kernel
// (parallel over q tiles) so neither needs atomics; both recompute
// P from (q, k, lse) — the standard flash backward decomposition.
//
// All tensors are STRIDED: (batch_stride, head_stride, row_stride) in
// elements with the head_dim axis contiguous. This lets the kernels read
// the fused-QKV linear output [B, S, h, 3, D] and write attention output
// [B, S, h*D] directly — no split/transpose/cat copies on either side.
// lse/delta are [B, H, S] fp32 contiguous. D in {64, 128}; any S.
// Both causal and bidirectional (ViT) attention.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf8;
typedef __attribute__((ext_vector_type(4))) float f4;
typedef __attribute__((ext_vector_type(8))) short short8v;

#define MFMA_BF16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

constexpr int TILE = 64;       // kv-tile rows (and bwd q-tile rows)
constexpr int NWAVES = 4;      // waves per block (backward kernels)
constexpr int BLOCKT = NWAVES * WAVE;
constexpr int PAD = 8;         // +16 B LDS row pad: alignment + bank spread

// forward geometry
constexpr int FW_WAVES = 8;            // 512 threads
constexpr int FW_BLOCKT = FW_WAVES * WAVE;
constexpr int QTILE = FW_WAVES * 16;   // 128 q rows per block

struct Strided {
  const __hip_bfloat16* p;
  long bs, hs, rs;  // batch/head/row strides (elements)
  DEV_INLINE const __hip_bfloat16* at(int b, int h) const {
    return p + (long)b * bs + (long)h * hs;
  }
};
struct StridedMut {
  __hip_bfloat16* p;
  long bs, hs, rs;
  DEV_INLINE __hip_bfloat16* at(int b, int h) const {
    return p + (long)b * bs + (long)h * hs;
  }
};

// ---------------------------------------------------------------------------
// LDS staging (synchronous forms, used by the backward kernels).
// Row-major [TILE][D+PAD] and transposed [D][TILE+PAD].
// Rows beyond `nvalid` zero-filled. `g` points at row 0 of the tile.
// ---------------------------------------------------------------------------
template <int D>
DEV_INLINE void stage_rowmajor(const __hip_bfloat16* __restrict__ g, long rs,
                               int nvalid, unsigned short* lds) {
  constexpr int RS = D + PAD;
  constexpr int NV = TILE * D / 8;
  for (int v = threadIdx.x; v < NV; v += BLOCKT) {
    int row = v / (D / 8);
    int col = (v % (D / 8)) * 8;
    short4v lo = {0, 0, 0, 0}, hi = {0, 0, 0, 0};
    if (row < nvalid) {
      const unsigned short* src = (const unsigned short*)g + (long)row * rs + col;
      lo = *reinterpret_cast<const short4v*>(src);
      hi = *reinterpret_cast<const short4v*>(src + 4);
    }
    *reinterpret_cast<short4v*>(lds + row * RS + col) = lo;
    *reinterpret_cast<short4v*>(lds + row * RS + col + 4) = hi;
  }
}

// Stage both images in one global read.
template <int D>
DEV_INLINE void stage_both(const __hip_bfloat16* __restrict__ g, long rs,
                           int nvalid, unsigned short* lds_rm,
                           unsigned short* lds_tr) {
  constexpr int RSM = D + PAD;
  constexpr int RST = TILE + PAD;
  constexpr int NV = TILE * D / 8;
  for (int v = threadIdx.x; v < NV; v += BLOCKT) {
    int row = v / (D / 8);
    int col = (v % (D / 8)) * 8;
    short4v lo = {0, 0, 0, 0}, hi = {0, 0, 0, 0};
    if (row < nvalid) {
      const unsigned short* src = (const unsigned short*)g + (long)row * rs + col;
      lo = *reinterpret_cast<const short4v*>(src);
      hi = *reinterpret_cast<const short4v*>(src + 4);
    }
    *reinterpret_cast<short4v*>(lds_rm + row * RSM + col) = lo;
    *reinterpret_cast<short4v*>(lds_rm + row * RSM + col + 4) = hi;
    unsigned short tmp[8];
    *reinterpret_cast<short4v*>(tmp) = lo;
    *reinterpret_cast<short4v*>(tmp + 4) = hi;
#pragma unroll
    for (int j = 0; j < 8; ++j) lds_tr[(col + j) * RST + row] = tmp[j];
  }
}

// Load this wave's 16 rows into MFMA A-fragments.
// A[row][k]: row = lane&15, k = (lane>>4)*8 + e (+ kc*32).
template <int D>
DEV_INLINE void load_a_frags(const __hip_bfloat16* __restrict__ g, long rs,
                             int row0, int nvalid, int lane, bf8* frags) {
  int row = row0 + (lane & 15);
  int d0 = (lane >> 4) * 8;
#pragma unroll
  for (int kc = 0; kc < D / 32; ++kc) {
    bf8 f = {};
    if (row < nvalid) {
      const unsigned short* src =
          (const unsigned short*)g + (long)row * rs + kc * 32 + d0;
      union { short4v v[2]; bf8 b; } u;
      u.v[0] = *reinterpret_cast<const short4v*>(src);
      u.v[1] = *reinterpret_cast<const short4v*>(src + 4);
      f = u.b;
    }
    frags[kc] = f;
  }
}

// B-fragment from a row-major LDS tile: B[k][col], col selects LDS row.
DEV_INLINE bf8 read_b_frag(const unsigned short* lds, int row_stride, int col0,
                           int k0, int lane) {
  const unsigned short* p =
      lds + (col0 + (lane & 15)) * row_stride + k0 + (lane >> 4) * 8;
  union { short4v v[2]; bf8 b; } u;
  u.v[0] = *reinterpret_cast<const short4v*>(p);
  u.v[1] = *reinterpret_cast<const short4v*>(p + 4);
  return u.b;
}

DEV_INLINE bf8 read_a_frag_lds(const unsigned short* lds, int row_stride,
                               int k0, int lane) {
  const unsigned short* p =
      lds + (lane & 15) * row_stride + k0 + (lane >> 4) * 8;
  union { short4v v[2]; bf8 b; } u;
  u.v[0] = *reinterpret_cast<const short4v*>(p);
  u.v[1] = *reinterpret_cast<const short4v*>(p + 4);
  return u.b;
}

// ===========================================================================
// Forward v2: 8 waves, 128-row Q tile, double-buffered KV, split staging.
// ===========================================================================
template <int D, bool CAUSAL, bool DROP>
__global__ __launch_bounds__(FW_BLOCKT) void attn_fwd_kernel(
    Strided q, Strided k, Strided v, StridedMut o, float* __restrict__ lse_out,
    int H, int S, float scale, int kv_total, int n_home,
    unsigned drop_thresh, float drop_inv_keep, unsigned seed0,
    unsigned seed1) {
  constexpr int KRS = D + PAD;          // K row-major row stride
  // V^T row stride: 68 shorts (136 B) keeps b64 writes/reads 8B-aligned
  // while avoiding the 0-mod-128B d-stride that made the transpose writes
  // 16-way bank conflicted (measured SQ_LDS_BANK_CONFLICT 15% of cycles)
  constexpr int VRS = TILE + 4;
  constexpr int NDT = D / 16;
  constexpr int KSZ = TILE * KRS;       // one K buffer (shorts)
  constexpr int VSZ = D * VRS;          // one V^T buffer

  // single __shared__ object (guide §5 trap 4a). P never touches LDS:
  // the swapped-QK^T score registers are redistributed into PV A-fragments
  // with permlane16/32 swaps (guide T12) — smem is 69.6 KB at D=128, so
  // TWO blocks fit per CU (4 waves/SIMD) instead of one
  __shared__ unsigned short smem[2 * KSZ + 2 * VSZ];
  unsigned short* k_lds = smem;                  // [2][KSZ]
  unsigned short* vt_lds = smem + 2 * KSZ;       // [2][VSZ]

  // bh on blockIdx.x: the linear dispatch then gives every CU one block
  // of each qt, balancing the causal tile-count imbalance when
  // B*H >= 256. For SMALL bh (pipeline-stage micro shapes), n_home < 0
  // marks PAIRED dispatch: gridDim.y is halved and each block runs BOTH
  // qt = blockIdx.y and its complement (-n_home-1) - blockIdx.y, whose
  // causal tile counts sum to a constant — equal work per block without
  // needing >= 256 blocks per tile position.
  const int b = blockIdx.x / H, hh = blockIdx.x % H;
  const long bh = blockIdx.x;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;

  const __hip_bfloat16* qp = q.at(b, hh);
  const __hip_bfloat16* kp = k.at(b, hh);
  const __hip_bfloat16* vp = v.at(b, hh);

  auto run_qtile = [&](const int qt) {

  // RB 16-row fragments per wave (32 q rows/wave, 256/block): every K/V
  // B-fragment LDS read feeds RB MFMAs, and barriers amortize over 2x rows
  constexpr int RB = 2;
  const int qrow0 = qt * (QTILE * RB) + wid * (16 * RB);
  bf8 qfrag[RB][D / 32];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
    load_a_frags<D>(qp, q.rs, qrow0 + rb * 16, S, lane, qfrag[rb]);

  f4 oacc[RB][NDT];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
#pragma unroll
    for (int i = 0; i < NDT; ++i) oacc[rb][i] = f4{0.f, 0.f, 0.f, 0.f};
  // per-lane ONE-row softmax state (swapped-QK^T layout: row = ccol)
  float m_r[RB][1], l_r[RB][1];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb) { m_r[rb][0] = -INFINITY; l_r[rb][0] = 0.f; }

  const int ccol = lane & 15;
  const int crow4 = (lane >> 4) * 4;

  const int kv_tiles = CAUSAL
      ? min(kv_total, (qt * QTILE * RB + QTILE * RB - 1) / TILE + 1)
      : kv_total;

  // ---- staging thread maps ----
  // K (row-major image): D/16 threads per row, 16 shorts each
  constexpr int TPR = D / 16;
  const int st_row = threadIdx.x / TPR;          // > TILE-1 threads idle (D=64)
  const int st_col = (threadIdx.x % TPR) * 16;
  const bool st_on = st_row < TILE;
  // V (transposed image): each thread owns a 4x4 block, transposes it in
  // registers and writes 4 b64 rows of the [d][kv] image — no b16 scatter
  constexpr int VCB = D / 4;                      // 4-col blocks per row
  const int vb_row = (threadIdx.x / VCB) * 4;
  const int vb_col = (threadIdx.x % VCB) * 4;
  const bool vb_on = vb_row < TILE;
  short8v kreg[2];
  short4v vblk[4];

  auto load_kv = [&](int kv0) {
#pragma unroll
    for (int h2 = 0; h2 < 2; ++h2) kreg[h2] = short8v{};
#pragma unroll
    for (int r = 0; r < 4; ++r) vblk[r] = short4v{};
    const int krow = kv0 + st_row;
    if (st_on && krow < S) {
      const unsigned short* ks = (const unsigned short*)kp +
                                 (long)krow * k.rs + st_col;
      kreg[0] = *reinterpret_cast<const short8v*>(ks);
      kreg[1] = *reinterpret_cast<const short8v*>(ks + 8);
    }
    if (vb_on) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int vrow = kv0 + vb_row + r;
        if (vrow < S)
          vblk[r] = *reinterpret_cast<const short4v*>(
              (const unsigned short*)vp + (long)vrow * v.rs + vb_col);
      }
    }
  };
  auto write_kv = [&](unsigned short* kd, unsigned short* vd) {
    if (st_on) {
      *reinterpret_cast<short8v*>(kd + st_row * KRS + st_col) = kreg[0];
      *reinterpret_cast<short8v*>(kd + st_row * KRS + st_col + 8) = kreg[1];
    }
    if (vb_on) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        short4v t = {vblk[0][c], vblk[1][c], vblk[2][c], vblk[3][c]};
        *reinterpret_cast<short4v*>(vd + (vb_col + c) * VRS + vb_row) = t;
      }
    }
  };

  // prologue: load tile 0 and write buffer 0
  load_kv(0);
  write_kv(k_lds, vt_lds);
  __syncthreads();
  // T5 static form: the younger dispatch half loses VALU arbitration to
  // the older half; one wave-uniform s_setprio(1) before the loop levels
  // it (guide T5 static form; readfirstlane keeps it scalar-branch).
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= FW_BLOCKT / 2)
    __builtin_amdgcn_s_setprio(1);

  for (int kt = 0; kt < kv_tiles; ++kt) {
    const int kv0 = kt * TILE;
    const int cur = kt & 1;
    const unsigned short* kb_lds = k_lds + cur * KSZ;
    const unsigned short* vb_lds = vt_lds + cur * VSZ;

    // ---- issue next tile's global loads (land during this tile's MFMAs) --
    const bool have_next = (kt + 1) < kv_tiles;
    if (have_next) load_kv((kt + 1) * TILE);

    // waves whose rows are entirely above this kv tile skip compute
    // (they still stage and hit the barrier)
    const bool active = !CAUSAL ||
        (kv0 <= min(qrow0 + RB * 16 - 1, S - 1));
    if (active) {
      // ---- swapped QK^T (guide T12 direction): mfma(K, Q) gives
      // C[kv][q] so each lane's 16 values are 16 kv scores of ONE q row
      // (row = ccol) — softmax state collapses to scalars and the
      // cross-lane reduction is 2 shuffles instead of a 4-step chain
      // per row. A- and B-fragments share a register layout, so the
      // operand swap is just argument order. ----
      f4 s[RB][4];
      bf8 pfrag[RB][2];  // PV A-fragments, assembled in registers (T12)
      bf8 kbuf[3];
      kbuf[0] = read_b_frag(kb_lds, KRS, 0, 0, lane);
      kbuf[1] = read_b_frag(kb_lds, KRS, 0, 32, lane);
#pragma unroll
      for (int i = 0; i < 4 * (D / 32); ++i) {
        const int ct = i / (D / 32), kc = i % (D / 32);
        if (i + 2 < 4 * (D / 32)) {
          const int j = i + 2;
          kbuf[j % 3] = read_b_frag(kb_lds, KRS, (j / (D / 32)) * 16,
                                    (j % (D / 32)) * 32, lane);
        }
        const bf8 kb = kbuf[i % 3];
#pragma unroll
        for (int rb = 0; rb < RB; ++rb) {
          f4 acc = (kc == 0) ? f4{0.f, 0.f, 0.f, 0.f} : s[rb][ct];
          s[rb][ct] = MFMA_BF16(kb, qfrag[rb][kc], acc);
        }
      }

      // ---- mask + scale + per-lane online softmax ----
#pragma unroll
      for (int rb = 0; rb < RB; ++rb) {
        const int my_qrow = qrow0 + rb * 16 + ccol;  // this lane's q row
        float pmax = -INFINITY;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int kcol = kv0 + ct * 16 + crow4 + r;
            float val = s[rb][ct][r] * scale;
            if ((CAUSAL && kcol > my_qrow) || kcol >= S) val = -INFINITY;
            s[rb][ct][r] = val;
            pmax = fmaxf(pmax, val);
          }
        }
        // row max across the 4 lane groups holding this q row
        pmax = fmaxf(pmax, __shfl_xor(pmax, 16, WAVE));
        pmax = fmaxf(pmax, __shfl_xor(pmax, 32, WAVE));

        // defer-max (guide T13): skip the O-wide rescale while the tile
        // max grew by <= THR — P is then bounded by e^THR instead of 1,
        // which the fp32 l/O accumulators absorb (P is bf16-packed for
        // PV: ~0.4% relative there; max-abs O error ~3x the THR=0 case
        // per the guide's numbers). Decision precedes exponentiation
        // (textbook order), so no pending-tile hazard.
        constexpr float RESCALE_THR = 8.0f;
        bool grew = pmax > m_r[rb][0] + RESCALE_THR;
        if (__builtin_amdgcn_ballot_w64(grew)) {
          float nm = fmaxf(m_r[rb][0], pmax);
          float alpha = (m_r[rb][0] == -INFINITY) ? 0.f
              : __expf(m_r[rb][0] - nm);
          m_r[rb][0] = (m_r[rb][0] == -INFINITY && pmax == -INFINITY)
              ? -INFINITY : nm;
          l_r[rb][0] *= alpha;
          // redistribute alpha from softmax rows (row=ccol) to the O
          // accumulator rows (row=crow4+r): row x's alpha lives in lane x
          float a4[4];
#pragma unroll
          for (int r = 0; r < 4; ++r)
            a4[r] = __shfl(alpha, crow4 + r, WAVE);
#pragma unroll
          for (int i = 0; i < NDT; ++i) {
#pragma unroll
            for (int r = 0; r < 4; ++r) oacc[rb][i][r] *= a4[r];
          }
        }

        float me = (m_r[rb][0] == -INFINITY) ? 0.f : m_r[rb][0];
        float psum = 0.f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float pv = __expf(s[rb][ct][r] - me);
            s[rb][ct][r] = pv;
            psum += pv;
          }
        }
        psum += __shfl_xor(psum, 16, WAVE);
        psum += __shfl_xor(psum, 32, WAVE);
        l_r[rb][0] += psum;

        if constexpr (DROP) {
          // attention dropout (reference hybrid_model.py:328 RNG-tracker
          // dropout): Philox keyed on (q row, kv chunk, bh) -> the SAME
          // mask regenerates in the backward kernels; the softmax
          // normalizer l uses the UN-dropped P (dropout acts on the
          // normalized weights), so only the PV operand is masked
#pragma unroll
          for (int ct = 0; ct < 4; ++ct) {
            unsigned kvc = (unsigned)((kv0 + ct * 16 + crow4) >> 2);
            philox4 r4 = philox4x32_10((unsigned)my_qrow, kvc,
                                       (unsigned)bh, seed1, seed0, seed1);
            const unsigned rr[4] = {r4.x, r4.y, r4.z, r4.w};
#pragma unroll
            for (int r = 0; r < 4; ++r)
              s[rb][ct][r] = (rr[r] >= drop_thresh)
                  ? s[rb][ct][r] * drop_inv_keep : 0.f;
          }
        }

        // ---- in-register P hand-off (guide T12): this lane holds ONE
        // q row's 16 P values as 8 packed u32 chunks C[ct][h]
        // (kv = 4g + 16·ct + {2h, 2h+1}, g = lane>>4). The PV A-fragment
        // needs kv = 8g+e (+32·kc) on this lane instead — a fixed
        // redistribution among the 4 lanes sharing this q row (l, l±16,
        // l±32). Phase A (permlane16_swap on a duplicated reg) leaves
        // e16 = the even-16-group's chunk and o16 = the odd group's on
        // every lane of each 16-pair; phase B (permlane16/32_swap of
        // e16[c_even] against e16[c_odd]) lands exactly the fragment's
        // low/high chunks split by even/odd 16-group, so ONE cndmask per
        // u32 finishes the job. Replaces 8 ds_write_b32 + 4 ds_read_b64
        // per row group AND frees the 36.9 KB P image -> 2 blocks/CU.
        unsigned int own[4][2];
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          own[ct][0] = cvt_pk_bf16(s[rb][ct][0], s[rb][ct][1]);
          own[ct][1] = cvt_pk_bf16(s[rb][ct][2], s[rb][ct][3]);
        }
        unsigned int e16[4][2], o16[4][2];
#pragma unroll
        for (int ct = 0; ct < 4; ++ct)
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            auto pr = __builtin_amdgcn_permlane16_swap(own[ct][h],
                                                       own[ct][h],
                                                       false, false);
            e16[ct][h] = pr[0];  // even 16-group's chunk (both lanes)
            o16[ct][h] = pr[1];  // odd 16-group's chunk
          }
        const bool even16 = ((lane >> 4) & 1) == 0;
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          const int ce = 2 * kc, co = 2 * kc + 1;
          unsigned int fr[4];
#pragma unroll
          for (int h = 0; h < 2; ++h) {
            auto pe = __builtin_amdgcn_permlane32_swap(e16[ce][h],
                                                       e16[co][h],
                                                       false, false);
            auto po = __builtin_amdgcn_permlane32_swap(o16[ce][h],
                                                       o16[co][h],
                                                       false, false);
            // pe[0]: lower lanes (0,ce) | upper (0,co); pe[1]: lower
            // (2,ce) | upper (2,co) -> even 16-group takes pe[0]
            fr[h] = even16 ? pe[0] : pe[1];
            fr[2 + h] = even16 ? po[0] : po[1];
          }
          union { unsigned int u[4]; bf8 b; } pu;
#pragma unroll
          for (int j = 0; j < 4; ++j) pu.u[j] = fr[j];
          pfrag[rb][kc] = pu.b;
        }
      }

      // ---- PV: both row groups share each V B-fragment read, 2-deep
      // prefetch over the (kc,dt) stream ----
      const bf8(&pa)[RB][2] = pfrag;
      bf8 vbuf[3];
      vbuf[0] = read_b_frag(vb_lds, VRS, 0, 0, lane);
      vbuf[1] = read_b_frag(vb_lds, VRS, 16, 0, lane);
#pragma unroll
      for (int i = 0; i < 2 * NDT; ++i) {
        const int kc = i / NDT, dt = i % NDT;
        if (i + 2 < 2 * NDT) {
          const int j = i + 2;
          vbuf[j % 3] = read_b_frag(vb_lds, VRS, (j % NDT) * 16,
                                    (j / NDT) * 32, lane);
        }
        const bf8 vb = vbuf[i % 3];
#pragma unroll
        for (int rb = 0; rb < RB; ++rb)
          oacc[rb][dt] = MFMA_BF16(pa[rb][kc], vb, oacc[rb][dt]);
      }
    }

    // ---- write next tile into the other buffer, one barrier per tile ----
    if (have_next)
      write_kv(k_lds + (cur ^ 1) * KSZ, vt_lds + (cur ^ 1) * VSZ);
    __syncthreads();
  }

  // ---- epilogue (strided o): softmax state lives on row=ccol lanes;
  // redistribute 1/l to the O-accumulator rows (crow4+r) via shfl ----
  __hip_bfloat16* op = o.at(b, hh);
#pragma unroll
  for (int rb = 0; rb < RB; ++rb) {
    float inv_own = (l_r[rb][0] > 0.f) ? 1.f / l_r[rb][0] : 0.f;
    float inv4[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) inv4[r] = __shfl(inv_own, crow4 + r, WAVE);
    if ((lane >> 4) == 0 && lse_out) {
      int qrow = qrow0 + rb * 16 + ccol;
      if (qrow < S)
        lse_out[bh * (long)S + qrow] =
            (l_r[rb][0] > 0.f) ? m_r[rb][0] + logf(l_r[rb][0]) : -INFINITY;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qrow = qrow0 + rb * 16 + crow4 + r;
      if (qrow >= S) continue;
      unsigned short* orow = (unsigned short*)op + (long)qrow * o.rs;
#pragma unroll
      for (int dt = 0; dt < NDT; dt += 2) {
        unsigned int u = cvt_pk_bf16(oacc[rb][dt][r] * inv4[r],
                                     oacc[rb][dt + 1][r] * inv4[r]);
        orow[dt * 16 + ccol] = (unsigned short)u;
        orow[(dt + 1) * 16 + ccol] = (unsigned short)(u >> 16);
      }
    }
  }
  };  // run_qtile

  const int n_home_abs = n_home < 0 ? -n_home : n_home;
  run_qtile(blockIdx.y);
  if (n_home < 0) {
    const int qt2 = (n_home_abs - 1) - (int)blockIdx.y;
    if (qt2 != (int)blockIdx.y && qt2 >= 0) {
      __syncthreads();
      run_qtile(qt2);
    }
  }
}

// ===========================================================================
// Backward: delta = rowsum(dO * O), strided inputs
// ===========================================================================
__global__ void attn_bwd_delta_kernel(Strided dout, Strided o,
                                      float* __restrict__ delta, long rows,
                                      int H, int S, int D) {
  // one row per 16-lane group (4 b64 loads per row at D=128): wider
  // loads + 4 rows in flight per wave instead of one latency-bound row
  constexpr int GROUPS = BLOCKT / 16;
  long row = (long)blockIdx.x * GROUPS + threadIdx.x / 16;
  if (row >= rows) return;
  int s = (int)(row % S);
  long bh = row / S;
  int hh = (int)(bh % H);
  int b = (int)(bh / H);
  int lane16 = threadIdx.x % 16;
  const unsigned short* dp =
      (const unsigned short*)dout.at(b, hh) + (long)s * dout.rs;
  const unsigned short* op = (const unsigned short*)o.at(b, hh) + (long)s * o.rs;
  float acc = 0.f;
  for (int i = lane16 * 4; i < D; i += 16 * 4) {
    short4v dv = *reinterpret_cast<const short4v*>(dp + i);
    short4v ov = *reinterpret_cast<const short4v*>(op + i);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      acc += bf_raw2f(((unsigned short*)&dv)[j]) *
             bf_raw2f(((unsigned short*)&ov)[j]);
  }
  acc = group16_reduce_sum(acc);
  if (lane16 == 0) delta[bh * (long)S + s] = acc;
}

// ===========================================================================
// Backward dK/dV v3 (transposed score space S^T = K Q^T).
// 8 waves, 128 kv rows per block (16/wave); 64-row q tiles streamed with
// issue-early register staging (guide T14): tile t+1's global loads are
// in flight during tile t's MFMAs; the four q/dO LDS images are
// DOUBLE-BUFFERED (158.7 KB of the CU's 160 KB LDS) so each tile costs
// one barrier instead of two.
// ===========================================================================
template <int D, bool CAUSAL, bool DROP>
__global__ __launch_bounds__(FW_BLOCKT) void attn_bwd_dkv_kernel(
    Strided q, Strided k, Strided v, Strided dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    StridedMut dk, StridedMut dv, int H, int S, float scale, int q_tiles,
    int n_home, unsigned drop_thresh, float drop_inv_keep, unsigned seed0,
    unsigned seed1) {
  constexpr int RS = D + PAD;
  constexpr int TS = TILE + PAD;       // P image stride (b128-aligned)
  constexpr int TRS = TILE + 4;        // transposed q/dO stride (see VRS)
  constexpr int NDT = D / 16;
  constexpr int QSZ = TILE * RS;       // q / dO row-major buffer
  constexpr int TSZ = D * TRS;         // q^T / dO^T buffer

  __shared__ unsigned short q_lds[2 * QSZ];
  __shared__ unsigned short qt_lds[2 * TSZ];
  __shared__ unsigned short do_lds[2 * QSZ];
  __shared__ unsigned short dot_lds[2 * TSZ];
  __shared__ unsigned short p_lds[FW_WAVES * 16 * TS];
  __shared__ float lsed_lds[2][2 * TILE];

  const int b = blockIdx.x / H, hh = blockIdx.x % H;
  const long bh = blockIdx.x;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;

  const __hip_bfloat16* qp = q.at(b, hh);
  const __hip_bfloat16* kp = k.at(b, hh);
  const __hip_bfloat16* vp = v.at(b, hh);
  const __hip_bfloat16* dop = dout.at(b, hh);

  auto run_kvtile = [&](const int kt) {
  const int kvrow0 = kt * QTILE + wid * 16;
  bf8 kfrag[D / 32], vfrag[D / 32];
  load_a_frags<D>(kp, k.rs, kvrow0, S, lane, kfrag);
  load_a_frags<D>(vp, v.rs, kvrow0, S, lane, vfrag);

  f4 dvacc[NDT], dkacc[NDT];
#pragma unroll
  for (int i = 0; i < NDT; ++i) {
    dvacc[i] = f4{0.f, 0.f, 0.f, 0.f};
    dkacc[i] = f4{0.f, 0.f, 0.f, 0.f};
  }

  const int ccol = lane & 15;
  const int crow4 = (lane >> 4) * 4;
  const int my_kvrow = kvrow0 + crow4;
  unsigned short* myp = p_lds + wid * 16 * TS;

  // staging maps: row-major via D/16 threads per q row; transposed images
  // via per-thread 4x4 register transpose (b64 writes — see fwd VRS note)
  constexpr int TPR = D / 16;
  const int st_row = threadIdx.x / TPR;
  const int st_col = (threadIdx.x % TPR) * 16;
  const bool st_on = st_row < TILE;
  constexpr int VCB = D / 4;
  const int vb_row = (threadIdx.x / VCB) * 4;
  const int vb_col = (threadIdx.x % VCB) * 4;
  const bool vb_on = vb_row < TILE;
  short8v qreg[2], dreg[2];
  short4v qblk[4], dblk[4];
  float lse_s = 0.f, dlt_s = 0.f;

  const int qt_first = CAUSAL ? (kt * QTILE) / TILE : 0;

  auto issue_loads = [&](int q0) {
#pragma unroll
    for (int h2 = 0; h2 < 2; ++h2) { qreg[h2] = short8v{}; dreg[h2] = short8v{}; }
#pragma unroll
    for (int r = 0; r < 4; ++r) { qblk[r] = short4v{}; dblk[r] = short4v{}; }
    const int row = q0 + st_row;
    if (st_on && row < S) {
      const unsigned short* qs = (const unsigned short*)qp + (long)row * q.rs + st_col;
      const unsigned short* ds = (const unsigned short*)dop + (long)row * dout.rs + st_col;
      qreg[0] = *reinterpret_cast<const short8v*>(qs);
      qreg[1] = *reinterpret_cast<const short8v*>(qs + 8);
      dreg[0] = *reinterpret_cast<const short8v*>(ds);
      dreg[1] = *reinterpret_cast<const short8v*>(ds + 8);
    }
    if (vb_on) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int vrow = q0 + vb_row + r;
        if (vrow < S) {
          qblk[r] = *reinterpret_cast<const short4v*>(
              (const unsigned short*)qp + (long)vrow * q.rs + vb_col);
          dblk[r] = *reinterpret_cast<const short4v*>(
              (const unsigned short*)dop + (long)vrow * dout.rs + vb_col);
        }
      }
    }
    const int li = threadIdx.x;
    if (li < TILE) {
      int qi = q0 + li;
      lse_s = (qi < S) ? lse[bh * (long)S + qi] : 0.f;
      dlt_s = (qi < S) ? delta[bh * (long)S + qi] : 0.f;
    }
  };
  auto write_tiles = [&](int buf) {
    unsigned short* qd = q_lds + buf * QSZ;
    unsigned short* dd = do_lds + buf * QSZ;
    unsigned short* qtd = qt_lds + buf * TSZ;
    unsigned short* dtd = dot_lds + buf * TSZ;
    if (st_on) {
      *reinterpret_cast<short8v*>(qd + st_row * RS + st_col) = qreg[0];
      *reinterpret_cast<short8v*>(qd + st_row * RS + st_col + 8) = qreg[1];
      *reinterpret_cast<short8v*>(dd + st_row * RS + st_col) = dreg[0];
      *reinterpret_cast<short8v*>(dd + st_row * RS + st_col + 8) = dreg[1];
    }
    if (vb_on) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        short4v tq = {qblk[0][c], qblk[1][c], qblk[2][c], qblk[3][c]};
        short4v td = {dblk[0][c], dblk[1][c], dblk[2][c], dblk[3][c]};
        *reinterpret_cast<short4v*>(qtd + (vb_col + c) * TRS + vb_row) = tq;
        *reinterpret_cast<short4v*>(dtd + (vb_col + c) * TRS + vb_row) = td;
      }
    }
    if (threadIdx.x < TILE) {
      lsed_lds[buf][threadIdx.x] = lse_s;
      lsed_lds[buf][TILE + threadIdx.x] = dlt_s;
    }
  };

  issue_loads(qt_first * TILE);
  write_tiles(0);
  __syncthreads();

  for (int qt = qt_first; qt < q_tiles; ++qt) {
    const int q0 = qt * TILE;
    const int cur = (qt - qt_first) & 1;
    const unsigned short* qb_lds = q_lds + cur * QSZ;
    const unsigned short* db_lds = do_lds + cur * QSZ;
    const unsigned short* qtb_lds = qt_lds + cur * TSZ;
    const unsigned short* dtb_lds = dot_lds + cur * TSZ;
    const float* lsed_b = lsed_lds[cur];
    const bool have_next = (qt + 1) < q_tiles;
    if (have_next) issue_loads(q0 + TILE);

    // waves whose kv rows are entirely below this q tile have no overlap
    const bool active = !CAUSAL || (q0 + TILE - 1 >= kvrow0);
    f4 pt[4], dpt[4];
    if (active) {
      // ---- S^T = K Q^T; P^T = exp(scale*S^T - lse[q]) ----
      unsigned dmask[4];  // per-ct keep bits (r=0..3) under dropout
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        f4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kc = 0; kc < D / 32; ++kc) {
          bf8 qb = read_b_frag(qb_lds, RS, ct * 16, kc * 32, lane);
          acc = MFMA_BF16(kfrag[kc], qb, acc);
        }
        int qcol = q0 + ct * 16 + ccol;
        float l = lsed_b[ct * 16 + ccol];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int kvr = my_kvrow + r;
          float p = 0.f;
          if ((!CAUSAL || qcol >= kvr) && qcol < S && kvr < S)
            p = __expf(acc[r] * scale - l);
          pt[ct][r] = p;
        }
        if constexpr (DROP) {
          // regenerate the fwd mask: counter (q row, kv chunk, bh)
          philox4 r4 = philox4x32_10((unsigned)qcol,
                                     (unsigned)(my_kvrow >> 2),
                                     (unsigned)bh, seed1, seed0, seed1);
          const unsigned rr[4] = {r4.x, r4.y, r4.z, r4.w};
          unsigned m = 0;
#pragma unroll
          for (int r = 0; r < 4; ++r)
            if (rr[r] >= drop_thresh) m |= (1u << r);
          dmask[ct] = m;
        }
      }

      // ---- stage P^T (dropout-masked: dV sees the dropped weights);
      //      dV += P^T dO ----
#pragma unroll
      for (int ct = 0; ct < 4; ++ct)
#pragma unroll
        for (int r = 0; r < 4; r += 2) {
          float p0 = pt[ct][r], p1 = pt[ct][r + 1];
          if constexpr (DROP) {
            p0 = (dmask[ct] >> r) & 1 ? p0 * drop_inv_keep : 0.f;
            p1 = (dmask[ct] >> (r + 1)) & 1 ? p1 * drop_inv_keep : 0.f;
          }
          unsigned int u = cvt_pk_bf16(p0, p1);
          unsigned short* base = myp + (crow4 + r) * TS + ct * 16 + ccol;
          base[0] = (unsigned short)u;
          base[TS] = (unsigned short)(u >> 16);
        }
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf8 pa = read_a_frag_lds(myp, TS, kc * 32, lane);
#pragma unroll
        for (int dt = 0; dt < NDT; ++dt) {
          bf8 db = read_b_frag(dtb_lds, TRS, dt * 16, kc * 32, lane);
          dvacc[dt] = MFMA_BF16(pa, db, dvacc[dt]);
        }
      }

      // ---- dP^T = V dO^T ----
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        f4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kc = 0; kc < D / 32; ++kc) {
          bf8 db = read_b_frag(db_lds, RS, ct * 16, kc * 32, lane);
          acc = MFMA_BF16(vfrag[kc], db, acc);
        }
        dpt[ct] = acc;
      }

      // ---- dS^T -> LDS; dK += dS^T Q ----
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        float dlt = lsed_b[TILE + ct * 16 + ccol];
        float ds[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float dp = dpt[ct][r];
          if constexpr (DROP)
            dp = (dmask[ct] >> r) & 1 ? dp * drop_inv_keep : 0.f;
          ds[r] = pt[ct][r] * (dp - dlt) * scale;
        }
#pragma unroll
        for (int r = 0; r < 4; r += 2) {
          unsigned int u = cvt_pk_bf16(ds[r], ds[r + 1]);
          unsigned short* base = myp + (crow4 + r) * TS + ct * 16 + ccol;
          base[0] = (unsigned short)u;
          base[TS] = (unsigned short)(u >> 16);
        }
      }
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf8 dsa = read_a_frag_lds(myp, TS, kc * 32, lane);
#pragma unroll
        for (int dt = 0; dt < NDT; ++dt) {
          bf8 qb = read_b_frag(qtb_lds, TRS, dt * 16, kc * 32, lane);
          dkacc[dt] = MFMA_BF16(dsa, qb, dkacc[dt]);
        }
      }
    }

    // write the next tile into the spare buffer; one barrier per tile
    if (have_next) write_tiles(cur ^ 1);
    __syncthreads();
  }

  __hip_bfloat16* dkp = dk.at(b, hh);
  __hip_bfloat16* dvp = dv.at(b, hh);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int kvr = my_kvrow + r;
    if (kvr >= S) continue;
    unsigned short* krow = (unsigned short*)dkp + (long)kvr * dk.rs;
    unsigned short* vrow = (unsigned short*)dvp + (long)kvr * dv.rs;
#pragma unroll
    for (int dt = 0; dt < NDT; dt += 2) {
      unsigned int uk = cvt_pk_bf16(dkacc[dt][r], dkacc[dt + 1][r]);
      unsigned int uv = cvt_pk_bf16(dvacc[dt][r], dvacc[dt + 1][r]);
      krow[dt * 16 + ccol] = (unsigned short)uk;
      krow[(dt + 1) * 16 + ccol] = (unsigned short)(uk >> 16);
      vrow[dt * 16 + ccol] = (unsigned short)uv;
      vrow[(dt + 1) * 16 + ccol] = (unsigned short)(uv >> 16);
    }
  }
  };  // run_kvtile

  // paired dispatch for small bh: block kt=0 streams ALL q tiles while
  // the last kt streams few — (kt, n-1-kt) pairs even the totals
  const int n_home_abs = n_home < 0 ? -n_home : n_home;
  run_kvtile(blockIdx.y);
  if (n_home < 0) {
    const int kt2 = (n_home_abs - 1) - (int)blockIdx.y;
    if (kt2 != (int)blockIdx.y && kt2 >= 0) {
      __syncthreads();
      run_kvtile(kt2);
    }
  }
}

// ===========================================================================
// Backward dQ v3: 8 waves, 256 q rows per block (RB=2 16-row groups per
// wave — every K/V/K^T B-fragment LDS read feeds 2 MFMAs, like fwd);
// 64-row kv tiles double-buffered in LDS with issue-early staging, one
// barrier per tile.
// ===========================================================================
template <int D, bool CAUSAL, bool DROP>
__global__ __launch_bounds__(FW_BLOCKT) void attn_bwd_dq_kernel(
    Strided q, Strided k, Strided v, Strided dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    StridedMut dq, int H, int S, float scale, int kv_total, int n_home,
    unsigned drop_thresh, float drop_inv_keep, unsigned seed0,
    unsigned seed1) {
  constexpr int RS = D + PAD;
  constexpr int TS = TILE + PAD;   // P image stride
  constexpr int TRS = TILE + 4;    // K^T stride (b64 images, see fwd VRS)
  constexpr int NDT = D / 16;
  constexpr int KSZ = TILE * RS;   // K row-major buffer
  constexpr int TSZ = D * TRS;     // K^T buffer
  constexpr int VSZ = TILE * RS;   // V row-major buffer
  constexpr int RB = 2;

  __shared__ unsigned short smem[2 * (KSZ + TSZ + VSZ) + FW_WAVES * 32 * TS];
  unsigned short* k_lds = smem;
  unsigned short* kt_lds = smem + 2 * KSZ;
  unsigned short* v_lds = smem + 2 * (KSZ + TSZ);
  unsigned short* p_lds = smem + 2 * (KSZ + TSZ + VSZ);

  const int b = blockIdx.x / H, hh = blockIdx.x % H;
  const long bh = blockIdx.x;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;

  const __hip_bfloat16* qp = q.at(b, hh);
  const __hip_bfloat16* kp = k.at(b, hh);
  const __hip_bfloat16* vp = v.at(b, hh);
  const __hip_bfloat16* dop = dout.at(b, hh);

  auto run_qtile = [&](const int qt) {
  const int qrow0 = qt * (QTILE * RB) + wid * (16 * RB);
  bf8 qfrag[RB][D / 32], dofrag[RB][D / 32];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb) {
    load_a_frags<D>(qp, q.rs, qrow0 + rb * 16, S, lane, qfrag[rb]);
    load_a_frags<D>(dop, dout.rs, qrow0 + rb * 16, S, lane, dofrag[rb]);
  }

  f4 dqacc[RB][NDT];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
#pragma unroll
    for (int i = 0; i < NDT; ++i) dqacc[rb][i] = f4{0.f, 0.f, 0.f, 0.f};

  const int ccol = lane & 15;
  const int crow4 = (lane >> 4) * 4;
  unsigned short* myp = p_lds + wid * 32 * TS;

  float lse_r[RB][4], dlt_r[RB][4];
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qr = qrow0 + rb * 16 + crow4 + r;
      lse_r[rb][r] = (qr < S) ? lse[bh * (long)S + qr] : 0.f;
      dlt_r[rb][r] = (qr < S) ? delta[bh * (long)S + qr] : 0.f;
    }

  const int kv_tiles = CAUSAL
      ? min(kv_total, (qt * QTILE * RB + QTILE * RB - 1) / TILE + 1)
      : kv_total;
  const int wave_last_row = min(qrow0 + RB * 16 - 1, S - 1);

  // staging maps: D/16 threads per kv row (row-major K/V); per-thread
  // 4x4 register transpose for the K^T image (b64 writes, see fwd)
  constexpr int TPR = D / 16;
  const int st_row = threadIdx.x / TPR;
  const int st_col = (threadIdx.x % TPR) * 16;
  const bool st_on = st_row < TILE;
  constexpr int VCB = D / 4;
  const int vb_row = (threadIdx.x / VCB) * 4;
  const int vb_col = (threadIdx.x % VCB) * 4;
  const bool vb_on = vb_row < TILE;
  short8v kreg[2], vreg[2];
  short4v kblk[4];

  auto issue_loads = [&](int kv0) {
#pragma unroll
    for (int h2 = 0; h2 < 2; ++h2) { kreg[h2] = short8v{}; vreg[h2] = short8v{}; }
#pragma unroll
    for (int r = 0; r < 4; ++r) kblk[r] = short4v{};
    const int row = kv0 + st_row;
    if (st_on && row < S) {
      const unsigned short* ks = (const unsigned short*)kp + (long)row * k.rs + st_col;
      const unsigned short* vs = (const unsigned short*)vp + (long)row * v.rs + st_col;
      kreg[0] = *reinterpret_cast<const short8v*>(ks);
      kreg[1] = *reinterpret_cast<const short8v*>(ks + 8);
      vreg[0] = *reinterpret_cast<const short8v*>(vs);
      vreg[1] = *reinterpret_cast<const short8v*>(vs + 8);
    }
    if (vb_on) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int vrow = kv0 + vb_row + r;
        if (vrow < S)
          kblk[r] = *reinterpret_cast<const short4v*>(
              (const unsigned short*)kp + (long)vrow * k.rs + vb_col);
      }
    }
  };
  auto write_tiles = [&](int buf) {
    unsigned short* kd = k_lds + buf * KSZ;
    unsigned short* ktd = kt_lds + buf * TSZ;
    unsigned short* vd = v_lds + buf * VSZ;
    if (st_on) {
      *reinterpret_cast<short8v*>(kd + st_row * RS + st_col) = kreg[0];
      *reinterpret_cast<short8v*>(kd + st_row * RS + st_col + 8) = kreg[1];
      *reinterpret_cast<short8v*>(vd + st_row * RS + st_col) = vreg[0];
      *reinterpret_cast<short8v*>(vd + st_row * RS + st_col + 8) = vreg[1];
    }
    if (vb_on) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        short4v t = {kblk[0][c], kblk[1][c], kblk[2][c], kblk[3][c]};
        *reinterpret_cast<short4v*>(ktd + (vb_col + c) * TRS + vb_row) = t;
      }
    }
  };

  issue_loads(0);
  write_tiles(0);
  __syncthreads();

  for (int ktl = 0; ktl < kv_tiles; ++ktl) {
    const int kv0 = ktl * TILE;
    const int cur = ktl & 1;
    const unsigned short* kb_lds = k_lds + cur * KSZ;
    const unsigned short* ktb_lds = kt_lds + cur * TSZ;
    const unsigned short* vb_lds = v_lds + cur * VSZ;

    const bool have_next = (ktl + 1) < kv_tiles;
    if (have_next) issue_loads(kv0 + TILE);

    const bool active = !CAUSAL || (kv0 <= wave_last_row);
    if (active) {
      // ---- per ct: S = Q K^T and dP = dO V^T for BOTH row groups
      // (each kb/vb read feeds 2 MFMAs), then dS -> LDS ----
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        f4 sacc[RB], dpacc[RB];
#pragma unroll
        for (int rb = 0; rb < RB; ++rb) {
          sacc[rb] = f4{0.f, 0.f, 0.f, 0.f};
          dpacc[rb] = f4{0.f, 0.f, 0.f, 0.f};
        }
#pragma unroll
        for (int kc = 0; kc < D / 32; ++kc) {
          bf8 kb = read_b_frag(kb_lds, RS, ct * 16, kc * 32, lane);
#pragma unroll
          for (int rb = 0; rb < RB; ++rb)
            sacc[rb] = MFMA_BF16(qfrag[rb][kc], kb, sacc[rb]);
        }
#pragma unroll
        for (int kc = 0; kc < D / 32; ++kc) {
          bf8 vb = read_b_frag(vb_lds, RS, ct * 16, kc * 32, lane);
#pragma unroll
          for (int rb = 0; rb < RB; ++rb)
            dpacc[rb] = MFMA_BF16(dofrag[rb][kc], vb, dpacc[rb]);
        }
        int kcol = kv0 + ct * 16 + ccol;
#pragma unroll
        for (int rb = 0; rb < RB; ++rb) {
          const int row_base = qrow0 + rb * 16 + crow4;
          float ds[4];
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float pv = 0.f;
            if ((!CAUSAL || kcol <= row_base + r) && kcol < S &&
                row_base + r < S)
              pv = __expf(sacc[rb][r] * scale - lse_r[rb][r]);
            float dp = dpacc[rb][r];
            if constexpr (DROP) {
              // q varies with r here, so each value needs its own
              // counter; only element kcol&3 of the 4 outputs is used
              philox4 r4 = philox4x32_10((unsigned)(row_base + r),
                                         (unsigned)(kcol >> 2),
                                         (unsigned)bh, seed1, seed0, seed1);
              const unsigned rr[4] = {r4.x, r4.y, r4.z, r4.w};
              dp = (rr[kcol & 3] >= drop_thresh) ? dp * drop_inv_keep : 0.f;
            }
            ds[r] = pv * (dp - dlt_r[rb][r]) * scale;
          }
#pragma unroll
          for (int r = 0; r < 4; r += 2) {
            unsigned int u = cvt_pk_bf16(ds[r], ds[r + 1]);
            unsigned short* base =
                myp + (rb * 16 + crow4 + r) * TS + ct * 16 + ccol;
            base[0] = (unsigned short)u;
            base[TS] = (unsigned short)(u >> 16);
          }
        }
      }

      // ---- dQ += dS K (each K^T read feeds both row groups) ----
      bf8 dsa[RB][2];
#pragma unroll
      for (int rb = 0; rb < RB; ++rb)
#pragma unroll
        for (int kc = 0; kc < 2; ++kc)
          dsa[rb][kc] = read_a_frag_lds(myp + rb * 16 * TS, TS, kc * 32,
                                        lane);
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
        for (int dt = 0; dt < NDT; ++dt) {
          bf8 kb = read_b_frag(ktb_lds, TRS, dt * 16, kc * 32, lane);
#pragma unroll
          for (int rb = 0; rb < RB; ++rb)
            dqacc[rb][dt] = MFMA_BF16(dsa[rb][kc], kb, dqacc[rb][dt]);
        }
      }
    }

    if (have_next) write_tiles(cur ^ 1);
    __syncthreads();
  }

  __hip_bfloat16* dqp = dq.at(b, hh);
#pragma unroll
  for (int rb = 0; rb < RB; ++rb)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qr = qrow0 + rb * 16 + crow4 + r;
      if (qr >= S) continue;
      unsigned short* qrow_p = (unsigned short*)dqp + (long)qr * dq.rs;
#pragma unroll
      for (int dt = 0; dt < NDT; dt += 2) {
        unsigned int u = cvt_pk_bf16(dqacc[rb][dt][r],
                                     dqacc[rb][dt + 1][r]);
        qrow_p[dt * 16 + ccol] = (unsigned short)u;
        qrow_p[(dt + 1) * 16 + ccol] = (unsigned short)(u >> 16);
      }
    }
  };  // run_qtile

  const int n_home_abs = n_home < 0 ? -n_home : n_home;
  run_qtile(blockIdx.y);
  if (n_home < 0) {
    const int qt2 = (n_home_abs - 1) - (int)blockIdx.y;
    if (qt2 != (int)blockIdx.y && qt2 >= 0) {
      __syncthreads();
      run_qtile(qt2);
    }
  }
}

Strided strided_of(const torch::Tensor& t, int b_dim, int h_dim, int s_dim) {
  Strided s;
  s.p = (const __hip_bfloat16*)t.data_ptr();
  s.bs = t.stride(b_dim);
  s.hs = t.stride(h_dim);
  s.rs = t.stride(s_dim);
  return s;
}
StridedMut strided_mut_of(torch::Tensor& t, int b_dim, int h_dim, int s_dim) {
  StridedMut s;
  s.p = (__hip_bfloat16*)t.data_ptr();
  s.bs = t.stride(b_dim);
  s.hs = t.stride(h_dim);
  s.rs = t.stride(s_dim);
  return s;
}

void check_attn_tensor(const torch::Tensor& t, int d_dim, int D) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(t.size(d_dim) == D && t.stride(d_dim) == 1,
              "head_dim axis must be contiguous");
}

void launch_fwd(Strided q, Strided k, Strided v, StridedMut o, float* lse,
                int B, int H, int S, int D, float scale, bool causal,
                float p_drop = 0.f, unsigned long long seed = 0) {
  int q_blocks = (S + QTILE * 2 - 1) / (QTILE * 2);  // RB=2 home tiles
  int kv_total = (S + TILE - 1) / TILE;
  // small-bh causal balance: pair (qt, n-1-qt) per block so per-block
  // work is constant without needing B*H >= 256 (ROADMAP r1 weak #6)
  bool pair = causal && B * H < 256 && q_blocks > 1;
  int grid_y = pair ? (q_blocks + 1) / 2 : q_blocks;
  int n_home = pair ? -q_blocks : q_blocks;
  dim3 grid(B * H, grid_y);
  auto stream = at::hip::getCurrentHIPStream();
  unsigned thresh = (unsigned)(p_drop * 4294967296.0);
  float invk = 1.f / (1.f - p_drop);
  unsigned s0 = (unsigned)seed, s1 = (unsigned)(seed >> 32);
  bool drop = p_drop > 0.f;
#define LAUNCH_FWD(DD, CC, PP)                                              \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, CC, PP>), grid, dim3(FW_BLOCKT),  \
                     0, stream, q, k, v, o, lse, H, S, scale, kv_total,     \
                     n_home, thresh, invk, s0, s1)
#define PICK_FWD(DD, CC) do { if (drop) LAUNCH_FWD(DD, CC, true); \
                              else LAUNCH_FWD(DD, CC, false); } while (0)
  if (D == 128) { if (causal) PICK_FWD(128, true); else PICK_FWD(128, false); }
  else          { if (causal) PICK_FWD(64, true);  else PICK_FWD(64, false); }
#undef PICK_FWD
#undef LAUNCH_FWD
}

void launch_bwd(Strided q, Strided k, Strided v, Strided dout, Strided o,
                StridedMut dq, StridedMut dk, StridedMut dv, float* lse,
                float* delta, int B, int H, int S, int D, float scale,
                bool causal, float p_drop = 0.f,
                unsigned long long seed = 0) {
  auto stream = at::hip::getCurrentHIPStream();
  long rows = (long)B * H * S;
  hipLaunchKernelGGL(attn_bwd_delta_kernel,
                     dim3((rows + (BLOCKT / 16) - 1) / (BLOCKT / 16)),
                     dim3(BLOCKT), 0, stream, dout, o, delta, rows, H, S, D);
  int tiles64 = (S + TILE - 1) / TILE;       // inner streamed tiles
  int blocks128 = (S + QTILE - 1) / QTILE;   // dkv per-block home tile
  int blocks256 = (S + QTILE * 2 - 1) / (QTILE * 2);  // dq RB=2 home tiles
  bool pair = causal && B * H < 256;
  int nh_dkv = (pair && blocks128 > 1) ? -blocks128 : blocks128;
  int nh_dq = (pair && blocks256 > 1) ? -blocks256 : blocks256;
  dim3 grid(B * H, nh_dkv < 0 ? (blocks128 + 1) / 2 : blocks128);
  dim3 grid_dq(B * H, nh_dq < 0 ? (blocks256 + 1) / 2 : blocks256);
  unsigned thresh = (unsigned)(p_drop * 4294967296.0);
  float invk = 1.f / (1.f - p_drop);
  unsigned s0 = (unsigned)seed, s1 = (unsigned)(seed >> 32);
  bool drop = p_drop > 0.f;
#define LAUNCH_BWD(DD, CC, PP)                                               \
  do {                                                                       \
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<DD, CC, PP>), grid,              \
                       dim3(FW_BLOCKT), 0, stream, q, k, v, dout, lse,       \
                       delta, dk, dv, H, S, scale, tiles64, nh_dkv, thresh,  \
                       invk, s0, s1);                                        \
    hipLaunchKernelGGL((attn_bwd_dq_kernel<DD, CC, PP>), grid_dq,            \
                       dim3(FW_BLOCKT), 0, stream, q, k, v, dout, lse,       \
                       delta, dq, H, S, scale, tiles64, nh_dq, thresh,       \
                       invk, s0, s1);                                        \
  } while (0)
#define PICK_BWD(DD, CC) do { if (drop) LAUNCH_BWD(DD, CC, true); \
                              else LAUNCH_BWD(DD, CC, false); } while (0)
  if (D == 128) { if (causal) PICK_BWD(128, true); else PICK_BWD(128, false); }
  else          { if (causal) PICK_BWD(64, true);  else PICK_BWD(64, false); }
#undef PICK_BWD
#undef LAUNCH_BWD
}

}  // namespace

// ===========================================================================
// Host bindings
// ===========================================================================

// q,k,v: [B, H, S, D] contiguous (layout-compat path)
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    double scale, double p_drop,
                                    long seed) {
  TORCH_CHECK(q.dim() == 4 && q.is_contiguous());
  int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(k.size(2) == S, "q and k seq length must match");
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  check_attn_tensor(q, 3, D);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));
  launch_fwd(strided_of(q, 0, 1, 2), strided_of(k, 0, 1, 2),
             strided_of(v, 0, 1, 2), strided_mut_of(o, 0, 1, 2),
             lse.data_ptr<float>(), B, H, S, D, (float)scale, causal,
             (float)p_drop, (unsigned long long)seed);
  return {o, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal, double scale,
                                    double p_drop, long seed) {
  int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));
  auto dc = dout.contiguous();
  launch_bwd(strided_of(q, 0, 1, 2), strided_of(k, 0, 1, 2),
             strided_of(v, 0, 1, 2), strided_of(dc, 0, 1, 2),
             strided_of(o, 0, 1, 2), strided_mut_of(dq, 0, 1, 2),
             strided_mut_of(dk, 0, 1, 2), strided_mut_of(dv, 0, 1, 2),
             lse.data_ptr<float>(), delta.data_ptr<float>(), B, H, S, D,
             (float)scale, causal, (float)p_drop,
             (unsigned long long)seed);
  return {dq, dk, dv};
}

// Packed path: qkv [B, S, Hh, 3, D] (the fused-QKV linear output viewed);
// returns o [B, S, Hh*D] + lse [B, Hh, S]. Zero layout copies.
std::vector<torch::Tensor> attn_fwd_packed(torch::Tensor qkv, long num_heads,
                                           double scale, double p_drop,
                                           long seed) {
  TORCH_CHECK(qkv.dim() == 5 && qkv.is_contiguous());
  int B = qkv.size(0), S = qkv.size(1), H = qkv.size(2), D = qkv.size(4);
  TORCH_CHECK(qkv.size(3) == 3 && H == num_heads);
  TORCH_CHECK(D == 64 || D == 128);
  check_attn_tensor(qkv, 4, D);
  auto o = torch::empty({B, S, (long)H * D}, qkv.options());
  auto lse = torch::empty({B, H, S}, qkv.options().dtype(torch::kFloat));
  Strided q{(const __hip_bfloat16*)qkv.data_ptr(), qkv.stride(0),
            qkv.stride(2), qkv.stride(1)};
  Strided k = q; k.p += D;
  Strided v = q; v.p += 2 * D;
  StridedMut om{(__hip_bfloat16*)o.data_ptr(), (long)S * H * D, (long)D,
                (long)H * D};
  launch_fwd(q, k, v, om, lse.data_ptr<float>(), B, H, S, D, (float)scale,
             /*causal=*/true, (float)p_drop, (unsigned long long)seed);
  return {o, lse};
}

// dout [B, S, Hh*D]; returns dqkv [B, S, Hh, 3, D]
torch::Tensor attn_bwd_packed(torch::Tensor dout, torch::Tensor qkv,
                              torch::Tensor o, torch::Tensor lse,
                              long num_heads, double scale, double p_drop,
                              long seed) {
  int B = qkv.size(0), S = qkv.size(1), H = qkv.size(2), D = qkv.size(4);
  auto dqkv = torch::empty_like(qkv);
  auto delta = torch::empty({B, H, S}, qkv.options().dtype(torch::kFloat));
  auto dc = dout.contiguous();
  Strided q{(const __hip_bfloat16*)qkv.data_ptr(), qkv.stride(0),
            qkv.stride(2), qkv.stride(1)};
  Strided k = q; k.p += D;
  Strided v = q; v.p += 2 * D;
  Strided dos{(const __hip_bfloat16*)dc.data_ptr(), (long)S * H * D, (long)D,
              (long)H * D};
  Strided os{(const __hip_bfloat16*)o.data_ptr(), (long)S * H * D, (long)D,
             (long)H * D};
  StridedMut dqm{(__hip_bfloat16*)dqkv.data_ptr(), dqkv.stride(0),
                 dqkv.stride(2), dqkv.stride(1)};
  StridedMut dkm = dqm; dkm.p += D;
  StridedMut dvm = dqm; dvm.p += 2 * D;
  launch_bwd(q, k, v, dos, os, dqm, dkm, dvm, lse.data_ptr<float>(),
             delta.data_ptr<float>(), B, H, S, D, (float)scale,
             /*causal=*/true, (float)p_drop, (unsigned long long)seed);
  return dqkv;
}
