// spark_sklearn_amd native kernels -- MI355X (gfx950, CDNA4) only.
//
// K1 (batched linear fit) + K4 (scoring) of SURVEY.md section 2.5: the
// worker-side compute the reference borrows from sklearn's Cython/BLAS
//  re-designed for CDNA4.  B candidate-columns (candidates x
// OVR classes) are fitted SIMULTANEOUSLY: the per-epoch work is two
// MFMA-shaped GEMMs over the resident bf16 matrix
//
//     Z = X[idx] . W^T          (fwd_d_kernel: fused elementwise d = dL/dz,
//                                per-column loss/db LDS reductions,
//                                transposed D write)
//     G = D^T . X[idx]          (bwd_g_kernel: split-K over row segments,
//                                atomic fp32 tile adds)
//
// followed by two tiny elementwise kernels (col_finalize_kernel: per-column
// bias/momentum/convergence state, fully on-device so the epoch loop never
// syncs the host; update_w_kernel: masked momentum update + bf16 mirror +
// G zeroing).  fp32 accumulation everywhere; bf16 is the storage/MFMA
// input dtype.
//
// Layout decisions (why everything is candidate-major):
//   X      [n][Fp]    bf16, row-major, Fp = roundup(F, 64) zero-padded
//   W,V,G  [Bp][Fp]   fp32, candidate-major -> update kernel fully
//                     coalesced, and the bf16 mirror Wt [Bp][Fp] stages
//                     into LDS with contiguous 16B reads as the MFMA
//                     B-operand (k-contiguous per column)
//   D^T    [Bp][NTp]  bf16 -- the fwd epilogue's C-fragment holds 4
//                     CONSECUTIVE rows per (lane, reg-quad) (C/D layout:
//                     col = lane&15, row = (lane>>4)*4 + reg), so the
//                     transposed store is a single 8-byte write; and the
//                     bwd kernel then reads D^T k-contiguously as its
//                     A-operand.  Zero-padded rows make split-K segments
//                     self-masking.
//
// MFMA fragment layout: C/D mapping is HW-verified in the programming
// guide; the A/B operand k-mapping (contiguous-8 vs two 4-blocks at k and
// k+16) is probed AT RUNTIME on first use (probe_kernel, all 4 A x B
// combinations) against a host reference -- see Common-mistake #3 /
// Guideline G9 (transpose-detecting checks with asymmetric inputs).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define WAVE 64

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));   \
  } while (0)

// ---------------------------------------------------------------------------
// fragment k-mapping (template param P = AMAP*2 + BMAP)
//   map 0: k = (lane>>4)*8 + i            (contiguous 8)
//   map 1: k = (lane>>4)*4 + (i&3) + (i>>2)*16   (two 4-blocks, +16 apart)
// ---------------------------------------------------------------------------
template <int MAP>
__device__ __forceinline__ int frag_k(int lane, int i) {
  if (MAP == 0) return ((lane >> 4) << 3) + i;
  return ((lane >> 4) << 2) + (i & 3) + ((i >> 2) << 4);
}

// Read an 8-element bf16 fragment from an LDS row `base` (element pointer to
// k=0 of that row).  map 0 -> one 16B read; map 1 -> two 8B reads.
template <int MAP>
__device__ __forceinline__ bf16x8 lds_frag(const __bf16* base, int lane) {
  bf16x8 out;
  if (MAP == 0) {
    const bf16x8* p =
        reinterpret_cast<const bf16x8*>(base + ((lane >> 4) << 3));
    out = *p;
  } else {
    const bf16x4* p0 =
        reinterpret_cast<const bf16x4*>(base + ((lane >> 4) << 2));
    const bf16x4* p1 =
        reinterpret_cast<const bf16x4*>(base + 16 + ((lane >> 4) << 2));
    bf16x4 a = *p0, b = *p1;
    out[0] = a[0]; out[1] = a[1]; out[2] = a[2]; out[3] = a[3];
    out[4] = b[0]; out[5] = b[1]; out[6] = b[2]; out[7] = b[3];
  }
  return out;
}

// ---------------------------------------------------------------------------
// layout probe: D = A(16x32) . B(32x16), one wave, LDS-staged like the real
// kernels.  Host test compares all 4 (AMAP, BMAP) combos vs reference.
// ---------------------------------------------------------------------------
template <int AMAP, int BMAP>
__global__ void probe_kernel(const __bf16* __restrict__ A,   // [16][32]
                             const __bf16* __restrict__ B,   // [16][32] = B^T (col-major B: B[k][c] stored as Bt[c][k])
                             float* __restrict__ D) {        // [16][16]
  __shared__ __bf16 sA[16][32];
  __shared__ __bf16 sB[16][32];
  int lane = threadIdx.x & 63;
  for (int e = lane; e < 16 * 32; e += WAVE) {
    sA[e / 32][e % 32] = A[e];
    sB[e / 32][e % 32] = B[e];
  }
  __syncthreads();
  bf16x8 a, b;
  int row = lane & 15;
  for (int i = 0; i < 8; ++i) {
    a[i] = sA[row][frag_k<AMAP>(lane, i)];
    b[i] = sB[row][frag_k<BMAP>(lane, i)];   // row here = output col
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) {
    int orow = ((lane >> 4) << 2) + r;
    int ocol = lane & 15;
    D[orow * 16 + ocol] = acc[r];
  }
}

// ---------------------------------------------------------------------------
// mean squared row norm (auto-lr curvature bound)
// ---------------------------------------------------------------------------
__global__ void msq_kernel(const __bf16* __restrict__ X, long n, int Fp,
                           double* __restrict__ out) {
  long row = blockIdx.x * blockDim.x + threadIdx.x;
  double local = 0.0;
  for (; row < n; row += (long)gridDim.x * blockDim.x) {
    const __bf16* xr = X + row * Fp;
    float s = 0.f;
    for (int f = 0; f < Fp; f += 8) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + f);
      for (int j = 0; j < 8; ++j) {
        float x = (float)v[j];
        s += x * x;
      }
    }
    local += (double)s;
  }
  // block-level reduce in LDS, one double atomic per block
  __shared__ double red[256];
  red[threadIdx.x] = local;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(out, red[0]);
}

// ---------------------------------------------------------------------------
// loss elementwise: d = dL/dz and per-element loss
// ---------------------------------------------------------------------------
#define LOSS_LOGISTIC 0
#define LOSS_HINGE 1
#define LOSS_SQUARED 2

__device__ __forceinline__ void loss_grad(int lid, float z, float yv,
                                          int tgt, float* d, float* ls) {
  const float eps = 1e-12f;
  if (lid == LOSS_LOGISTIC) {
    float t = (yv == (float)tgt) ? 1.f : 0.f;
    float p = 1.f / (1.f + __expf(-z));
    *d = p - t;
    *ls = -(t * __logf(p + eps) + (1.f - t) * __logf(1.f - p + eps));
  } else if (lid == LOSS_HINGE) {
    float s = (yv == (float)tgt) ? 1.f : -1.f;
    float m = s * z;
    *d = (m < 1.f) ? -s : 0.f;
    *ls = fmaxf(0.f, 1.f - m);
  } else {
    float t = (tgt < 0) ? yv : ((yv == (float)tgt) ? 1.f : -1.f);
    float r = z - t;
    *d = r;
    *ls = 0.5f * r * r;
  }
}

// ---------------------------------------------------------------------------
// fwd_d_kernel: Z = X[idx] . Wt^T + b, fused d/loss epilogue.
// grid (NTp/64, Bp/64), block 256 (4 waves as 2x2 of 32x32 wave-tiles).
// ---------------------------------------------------------------------------
template <int P>
__global__ __launch_bounds__(256)
void fwd_d_kernel(const __bf16* __restrict__ X,       // [n][Fp]
                  const float* __restrict__ y,        // [n]
                  const int* __restrict__ idx,        // [NTp]
                  const __bf16* __restrict__ Wt,      // [Bp][Fp]
                  const float* __restrict__ bias,     // [Bp]
                  const int* __restrict__ loss_id,    // [Bp]
                  const int* __restrict__ target,     // [Bp]
                  __bf16* __restrict__ Dt,            // [Bp][NTp]
                  float* __restrict__ loss_sum,       // [Bp]
                  float* __restrict__ db_sum,         // [Bp]
                  int nt, int NTp, int Fp) {
  constexpr int AMAP = P >> 1, BMAP = P & 1;
  __shared__ __bf16 sX[64][72];
  __shared__ __bf16 sW[64][72];
  __shared__ float sLoss[64];
  __shared__ float sDb[64];
  __shared__ int sIdx[64];
  __shared__ float sY[64];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave & 1, wn = wave >> 1;
  const int rt = blockIdx.x, ct = blockIdx.y;

  if (tid < 64) {
    int r = rt * 64 + tid;
    int gi = idx[r < NTp ? r : NTp - 1];
    sIdx[tid] = gi;
    sY[tid] = y[gi];
    sLoss[tid] = 0.f;
    sDb[tid] = 0.f;
  }
  __syncthreads();

  f32x4 acc[2][2];
  for (int i = 0; i < 2; ++i)
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int srow = tid >> 2;          // staging: 64 rows x 4 segs of 16
  const int sseg = tid & 3;
  for (int kc = 0; kc < Fp; kc += 64) {
    // stage X rows (gathered) and W rows (candidate-major): 16B vectors
    {
      const uint4* gx = reinterpret_cast<const uint4*>(
          X + (long)sIdx[srow] * Fp + kc + sseg * 16);
      uint4* lx = reinterpret_cast<uint4*>(&sX[srow][sseg * 16]);
      lx[0] = gx[0];
      lx[1] = gx[1];
      const uint4* gw = reinterpret_cast<const uint4*>(
          Wt + (long)(ct * 64 + srow) * Fp + kc + sseg * 16);
      uint4* lw = reinterpret_cast<uint4*>(&sW[srow][sseg * 16]);
      lw[0] = gw[0];
      lw[1] = gw[1];
    }
    __syncthreads();
    for (int ks = 0; ks < 64; ks += 32) {
      bf16x8 a0 = lds_frag<AMAP>(&sX[wm * 32 + (lane & 15)][ks], lane);
      bf16x8 a1 = lds_frag<AMAP>(&sX[wm * 32 + 16 + (lane & 15)][ks], lane);
      bf16x8 b0 = lds_frag<BMAP>(&sW[wn * 32 + (lane & 15)][ks], lane);
      bf16x8 b1 = lds_frag<BMAP>(&sW[wn * 32 + 16 + (lane & 15)][ks], lane);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: z -> d, loss; transposed D write (4 consecutive rows / 8B)
  for (int fn = 0; fn < 2; ++fn) {
    int col_in = wn * 32 + fn * 16 + (lane & 15);
    int c = ct * 64 + col_in;
    float bia = bias[c];
    int lid = loss_id[c];
    int tgt = target[c];
    float lsum = 0.f, dsum = 0.f;
    for (int fm = 0; fm < 2; ++fm) {
      int row_base = wm * 32 + fm * 16 + ((lane >> 4) << 2);
      bf16x4 dpack;
      for (int r = 0; r < 4; ++r) {
        int row_in = row_base + r;
        int gr = rt * 64 + row_in;
        float z = acc[fm][fn][r] + bia;
        float d, ls;
        loss_grad(lid, z, sY[row_in], tgt, &d, &ls);
        if (gr >= nt) { d = 0.f; ls = 0.f; }
        dpack[r] = (__bf16)d;
        lsum += ls;
        dsum += d;
      }
      bf16x4* dst = reinterpret_cast<bf16x4*>(
          Dt + (long)c * NTp + rt * 64 + row_base);
      *dst = dpack;
    }
    atomicAdd(&sLoss[col_in], lsum);
    atomicAdd(&sDb[col_in], dsum);
  }
  __syncthreads();
  if (tid < 64) {
    atomicAdd(&loss_sum[ct * 64 + tid], sLoss[tid]);
    atomicAdd(&db_sum[ct * 64 + tid], sDb[tid]);
  }
}

// ---------------------------------------------------------------------------
// bwd_g_kernel: G^T[c][f] += sum_r D^T[c][r] * X[idx[r]][f], split-K over
// row segments.  grid (Bp/64, Fp/64, nseg), block 256.
// ---------------------------------------------------------------------------
template <int P>
__global__ __launch_bounds__(256)
void bwd_g_kernel(const __bf16* __restrict__ X,     // [n][Fp]
                  const int* __restrict__ idx,      // [NTp]
                  const __bf16* __restrict__ Dt,    // [Bp][NTp]
                  float* __restrict__ Gt,           // [Bp][Fp]
                  int NTp, int Fp, int seg_rows) {
  constexpr int AMAP = P >> 1, BMAP = P & 1;
  __shared__ __bf16 sD[64][72];    // candidate x row-chunk
  __shared__ __bf16 sXt[64][72];   // feature x row-chunk (transposed)
  __shared__ int sRid[64];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave & 1, wn = wave >> 1;
  const int ct = blockIdx.x, ft = blockIdx.y;
  const int r0 = blockIdx.z * seg_rows;
  const int r1 = min(r0 + seg_rows, NTp);

  f32x4 acc[2][2];
  for (int i = 0; i < 2; ++i)
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int srow = tid >> 2;
  const int sseg = tid & 3;
  for (int rc = r0; rc < r1; rc += 64) {
    if (tid < 64) sRid[tid] = idx[rc + tid];
    __syncthreads();
    // stage D^T tile: candidate-major, contiguous in r
    {
      const uint4* gd = reinterpret_cast<const uint4*>(
          Dt + (long)(ct * 64 + srow) * NTp + rc + sseg * 16);
      uint4* ld = reinterpret_cast<uint4*>(&sD[srow][sseg * 16]);
      ld[0] = gd[0];
      ld[1] = gd[1];
    }
    // stage X transposed: read 8 contiguous f per thread, scatter rows
    {
      int r = tid & 63;
      int fs = tid >> 6;                 // 4 threads-groups x 2 iters = 8 segs
      for (int it = 0; it < 2; ++it) {
        int fseg = fs + it * 4;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            X + (long)sRid[r] * Fp + ft * 64 + fseg * 8);
        for (int j = 0; j < 8; ++j) sXt[fseg * 8 + j][r] = v[j];
      }
    }
    __syncthreads();
    for (int ks = 0; ks < 64; ks += 32) {
      bf16x8 a0 = lds_frag<AMAP>(&sD[wm * 32 + (lane & 15)][ks], lane);
      bf16x8 a1 = lds_frag<AMAP>(&sD[wm * 32 + 16 + (lane & 15)][ks], lane);
      bf16x8 b0 = lds_frag<BMAP>(&sXt[wn * 32 + (lane & 15)][ks], lane);
      bf16x8 b1 = lds_frag<BMAP>(&sXt[wn * 32 + 16 + (lane & 15)][ks], lane);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  for (int fm = 0; fm < 2; ++fm) {
    for (int fn = 0; fn < 2; ++fn) {
      int f_in = wn * 32 + fn * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r) {
        int c_in = wm * 32 + fm * 16 + ((lane >> 4) << 2) + r;
        atomicAdd(&Gt[(long)(ct * 64 + c_in) * Fp + ft * 64 + f_in],
                  acc[fm][fn][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// col_finalize_kernel: per-column bias update + convergence, on-device.
// ---------------------------------------------------------------------------
__global__ void col_finalize_kernel(
    float* __restrict__ loss_sum, float* __restrict__ db_sum,
    float* __restrict__ bias, float* __restrict__ Vb,
    float* __restrict__ prev_loss, float* __restrict__ cur_loss,
    int* __restrict__ active, int* __restrict__ epoch_mask,
    int* __restrict__ n_iter,
    const int* __restrict__ max_iter, const float* __restrict__ lr,
    const float* __restrict__ mom, const float* __restrict__ tol,
    const unsigned char* __restrict__ fit_ic,
    float inv_n, int epoch, int Bp, int* __restrict__ n_active) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= Bp) return;
  float cur = loss_sum[c] * inv_n;
  int ea = active[c] && (epoch < max_iter[c]);
  epoch_mask[c] = ea;
  if (ea) {
    n_iter[c] = epoch + 1;
    cur_loss[c] = cur;
    float gb = db_sum[c] * inv_n * (float)fit_ic[c];
    float vb = mom[c] * Vb[c] - lr[c] * gb;
    Vb[c] = vb;
    bias[c] += vb;
    if (fabsf(prev_loss[c] - cur) <= tol[c] * fmaxf(cur, 1.f)) active[c] = 0;
    prev_loss[c] = cur;
  }
  loss_sum[c] = 0.f;
  db_sum[c] = 0.f;
  if (active[c] && (epoch + 1) < max_iter[c]) atomicAdd(n_active, 1);
}

// ---------------------------------------------------------------------------
// update_w_kernel: masked momentum update, bf16 mirror, G zeroing.
// grid (Fp/256, Bp), block 256.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256)
void update_w_kernel(float* __restrict__ W, float* __restrict__ V,
                     float* __restrict__ Gt, __bf16* __restrict__ Wt,
                     const int* __restrict__ epoch_mask,
                     const float* __restrict__ lr,
                     const float* __restrict__ mom,
                     const float* __restrict__ l1v,
                     const float* __restrict__ l2v,
                     float inv_n, int Fp) {
  int c = blockIdx.y;
  long off = (long)c * Fp + blockIdx.x * 256 + threadIdx.x;
  float g = Gt[off];
  Gt[off] = 0.f;
  if (!epoch_mask[c]) return;
  float w = W[off];
  float grad = g * inv_n + l2v[c] * w +
               l1v[c] * ((w > 0.f) ? 1.f : ((w < 0.f) ? -1.f : 0.f));
  float v = mom[c] * V[off] - lr[c] * grad;
  V[off] = v;
  w += v;
  W[off] = w;
  Wt[off] = (__bf16)w;
}

// ---------------------------------------------------------------------------
// scores_kernel: per-candidate metric sums over idx rows.
// metric: 0=accuracy(correct count), 1=sum loglik, 2=sum squared residual.
// grid ceil(n_idx/32); block 256 (4 waves side by side over Bp cols).
// dynamic LDS: zbuf f32[32][Bp+8] + sX[32][72] + sW[Bp][72] + scand[ncand]
// ---------------------------------------------------------------------------
template <int P>
__global__ __launch_bounds__(256)
void scores_kernel(const __bf16* __restrict__ X, const float* __restrict__ y,
                   const long* __restrict__ idx,   // [n_idx] (no padding)
                   const __bf16* __restrict__ Wt, const float* __restrict__ bias,
                   const int* __restrict__ loss_id,
                   const int* __restrict__ target,
                   int C, int metric, long n_idx, int Fp, int Bp, int n_cand,
                   double* __restrict__ out_sum) {
  constexpr int AMAP = P >> 1, BMAP = P & 1;
  const int scand_pad = (n_cand + 3) & ~3;   // keep sX 16B-aligned
  extern __shared__ char smem[];
  long* sRow = reinterpret_cast<long*>(smem);                // [32]
  float* zbuf = reinterpret_cast<float*>(sRow + 32);         // [32][Bp+8]
  float* scand = zbuf + 32 * (Bp + 8);                       // [scand_pad]
  __bf16* sX = reinterpret_cast<__bf16*>(scand + scand_pad); // [32][72]
  __bf16* sW = sX + 32 * 72;                                 // [Bp][72]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int rt = blockIdx.x;
  const int colsPerWave = Bp / 4;            // Bp in {64,128,256}
  const int nfrag = colsPerWave / 16;

  if (tid < 32) {
    long r = (long)rt * 32 + tid;
    sRow[tid] = (r < n_idx) ? idx[r] : idx[n_idx - 1];
  }
  for (int i = tid; i < n_cand; i += 256) scand[i] = 0.f;
  __syncthreads();

  // accumulators: 2 m-frags x nfrag n-frags (nfrag <= 4)
  f32x4 acc[2][4];
  for (int i = 0; i < 2; ++i)
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int kc = 0; kc < Fp; kc += 64) {
    // stage 32 gathered X rows: 32 rows x 64 k = 2048 bf16; 256 thr x 8
    {
      int r = tid >> 3, seg = tid & 7;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          X + sRow[r] * Fp + kc + seg * 8);
      *reinterpret_cast<bf16x8*>(&sX[r * 72 + seg * 8]) = v;
    }
    // stage Wt rows for all Bp cols: Bp x 64 bf16; thread strided
    for (int e = tid; e < Bp * 4; e += 256) {
      int c = e >> 2, seg = e & 3;
      const uint4* gw = reinterpret_cast<const uint4*>(
          Wt + (long)c * Fp + kc + seg * 16);
      uint4* lw = reinterpret_cast<uint4*>(&sW[c * 72 + seg * 16]);
      lw[0] = gw[0];
      lw[1] = gw[1];
    }
    __syncthreads();
    for (int ks = 0; ks < 64; ks += 32) {
      bf16x8 a0 = lds_frag<AMAP>(&sX[(lane & 15) * 72 + ks], lane);
      bf16x8 a1 = lds_frag<AMAP>(&sX[(16 + (lane & 15)) * 72 + ks], lane);
      for (int fn = 0; fn < nfrag; ++fn) {
        int colbase = wave * colsPerWave + fn * 16 + (lane & 15);
        bf16x8 b = lds_frag<BMAP>(&sW[colbase * 72 + ks], lane);
        acc[0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, acc[0][fn], 0, 0, 0);
        acc[1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, acc[1][fn], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // spill z to LDS (add bias)
  for (int fm = 0; fm < 2; ++fm) {
    for (int fn = 0; fn < nfrag; ++fn) {
      int col = wave * colsPerWave + fn * 16 + (lane & 15);
      float bia = bias[col];
      for (int r = 0; r < 4; ++r) {
        int row = fm * 16 + ((lane >> 4) << 2) + r;
        zbuf[row * (Bp + 8) + col] = acc[fm][fn][r] + bia;
      }
    }
  }
  __syncthreads();

  // score (row, cand) pairs
  const float eps = 1e-12f;
  for (int p = tid; p < 32 * n_cand; p += 256) {
    int row = p & 31;
    int cand = p >> 5;
    long gr = (long)rt * 32 + row;
    if (gr >= n_idx) continue;
    float yv = y[sRow[row]];
    const float* zr = &zbuf[row * (Bp + 8) + cand * C];
    float val = 0.f;
    if (metric == 2) {                       // r2: squared residual
      float r = yv - zr[0];
      val = r * r;
    } else if (C == 1) {
      float z = zr[0];
      if (metric == 0) {
        int pred = (z > 0.f) ? 1 : 0;
        val = (pred == (int)yv) ? 1.f : 0.f;
      } else {
        float pr = 1.f / (1.f + __expf(-z));
        float t = (yv == 1.f) ? 1.f : 0.f;
        val = t * __logf(pr + eps) + (1.f - t) * __logf(1.f - pr + eps);
      }
    } else {
      if (metric == 0) {
        int best = 0;
        float bz = zr[0];
        for (int c = 1; c < C; ++c)
          if (zr[c] > bz) { bz = zr[c]; best = c; }
        val = (best == (int)yv) ? 1.f : 0.f;
      } else {
        float ssum = 0.f, sy = 0.f;
        for (int c = 0; c < C; ++c) {
          float s = 1.f / (1.f + __expf(-zr[c]));
          ssum += s;
          if (c == (int)yv) sy = s;
        }
        val = __logf(sy / fmaxf(ssum, eps) + eps);
      }
    }
    atomicAdd(&scand[cand], val);
  }
  __syncthreads();
  for (int cand = tid; cand < n_cand; cand += 256)
    atomicAdd(&out_sum[cand], (double)scand[cand]);
}

// ===========================================================================
// host wrappers
// ===========================================================================

static int g_layout = -1;   // P = AMAP*2+BMAP chosen by the runtime probe

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// run all 4 layout combos; returns tensor [4][16][16] of probe outputs
torch::Tensor gemm_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16);
  TORCH_CHECK(A.sizes() == std::vector<int64_t>({16, 32}));
  TORCH_CHECK(B.sizes() == std::vector<int64_t>({16, 32}));
  auto D = torch::zeros({4, 16, 16},
                        A.options().dtype(torch::kFloat32));
  const __bf16* a = reinterpret_cast<const __bf16*>(A.data_ptr());
  const __bf16* b = reinterpret_cast<const __bf16*>(B.data_ptr());
  float* d = D.data_ptr<float>();
  hipStream_t s = cur_stream();
  hipLaunchKernelGGL((probe_kernel<0, 0>), dim3(1), dim3(64), 0, s, a, b, d);
  hipLaunchKernelGGL((probe_kernel<0, 1>), dim3(1), dim3(64), 0, s, a, b, d + 256);
  hipLaunchKernelGGL((probe_kernel<1, 0>), dim3(1), dim3(64), 0, s, a, b, d + 512);
  hipLaunchKernelGGL((probe_kernel<1, 1>), dim3(1), dim3(64), 0, s, a, b, d + 768);
  HIP_CHECK(hipGetLastError());
  return D;
}

void set_layout(int64_t p) {
  TORCH_CHECK(p >= 0 && p < 4, "layout in [0,4)");
  g_layout = (int)p;
}

int64_t get_layout() { return g_layout; }

double msq_rows(torch::Tensor X) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kBFloat16 &&
              X.dim() == 2 && X.stride(1) == 1);
  long n = X.size(0);
  int Fp = (int)X.size(1);
  TORCH_CHECK(Fp % 8 == 0);
  auto out = torch::zeros({1}, X.options().dtype(torch::kFloat64));
  int blocks = (int)std::min<long>((n + 255) / 256, 2048);
  hipLaunchKernelGGL(msq_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     reinterpret_cast<const __bf16*>(X.data_ptr()), n, Fp,
                     out.data_ptr<double>());
  HIP_CHECK(hipGetLastError());
  return out.item<double>() / (double)n;
}

// linear_fit: the full epoch loop on device.
// Returns (W [Bp][Fp] f32, bias [Bp], n_iter [Bp] i32, cur_loss [Bp] f32,
//          Wt [Bp][Fp] bf16 -- kept on device for the scores kernel).
std::vector<torch::Tensor> linear_fit(
    torch::Tensor X, torch::Tensor y, torch::Tensor idx,
    torch::Tensor loss_id, torch::Tensor lr, torch::Tensor l1,
    torch::Tensor l2, torch::Tensor mom, torch::Tensor tol,
    torch::Tensor max_iter, torch::Tensor target, torch::Tensor fit_ic,
    int64_t nt, int64_t max_epochs, int64_t check_every) {
  TORCH_CHECK(g_layout >= 0, "MFMA layout not probed; call ensure_layout()");
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kBFloat16 && X.dim() == 2);
  TORCH_CHECK(X.stride(1) == 1 && X.stride(0) == X.size(1));
  const at::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  int Fp = (int)X.size(1);
  TORCH_CHECK(Fp % 64 == 0, "Fp must be a multiple of 64");
  int Bp = (int)loss_id.size(0);
  TORCH_CHECK(Bp % 64 == 0, "Bp must be a multiple of 64");
  long NTp = idx.size(0);
  TORCH_CHECK(NTp % 64 == 0 && nt <= NTp);
  TORCH_CHECK(idx.dtype() == torch::kInt32);
  TORCH_CHECK(y.dtype() == torch::kFloat32);

  auto optF = X.options().dtype(torch::kFloat32);
  auto optI = X.options().dtype(torch::kInt32);
  auto W = torch::zeros({Bp, Fp}, optF);
  auto V = torch::zeros({Bp, Fp}, optF);
  auto Gt = torch::zeros({Bp, Fp}, optF);
  auto Wt = torch::zeros({Bp, Fp}, X.options());
  auto Dt = torch::zeros({(long)Bp, NTp}, X.options());
  auto bias = torch::zeros({Bp}, optF);
  auto Vb = torch::zeros({Bp}, optF);
  auto loss_sum = torch::zeros({Bp}, optF);
  auto db_sum = torch::zeros({Bp}, optF);
  auto prev_loss = torch::full({Bp}, 3.0e38, optF);
  auto cur_loss = torch::zeros({Bp}, optF);
  auto active = torch::ones({Bp}, optI);
  auto epoch_mask = torch::zeros({Bp}, optI);
  auto n_iter = torch::zeros({Bp}, optI);
  auto n_active = torch::zeros({1}, optI);

  const __bf16* Xp = reinterpret_cast<const __bf16*>(X.data_ptr());
  float inv_n = 1.0f / (float)nt;
  hipStream_t s = cur_stream();

  // split-K segmentation: target ~1024 blocks for the bwd launch
  int ctiles = Bp / 64, ftiles = Fp / 64;
  long rtiles = NTp / 64;
  long want_nseg = std::max<long>(1, 1024 / (ctiles * ftiles));
  long seg_tiles = std::max<long>(1, (rtiles + want_nseg - 1) / want_nseg);
  long nseg = (rtiles + seg_tiles - 1) / seg_tiles;
  int seg_rows = (int)(seg_tiles * 64);

  dim3 fwd_grid((unsigned)(NTp / 64), (unsigned)(Bp / 64));
  dim3 bwd_grid((unsigned)ctiles, (unsigned)ftiles, (unsigned)nseg);
  dim3 fin_grid((unsigned)((Bp + 255) / 256));
  TORCH_CHECK(Fp % 256 == 0 || Fp == 64 || Fp == 128,
              "Fp must be 64, 128 or a multiple of 256");
  int upd_threads = Fp >= 256 ? 256 : Fp;
  dim3 upd_grid2((unsigned)(Fp / upd_threads), (unsigned)Bp);

  auto launch_epoch = [&](int epoch, auto fwd_k, auto bwd_k) {
    hipLaunchKernelGGL(fwd_k, fwd_grid, dim3(256), 0, s, Xp,
                       y.data_ptr<float>(), idx.data_ptr<int>(),
                       reinterpret_cast<const __bf16*>(Wt.data_ptr()),
                       bias.data_ptr<float>(), loss_id.data_ptr<int>(),
                       target.data_ptr<int>(),
                       reinterpret_cast<__bf16*>(Dt.data_ptr()),
                       loss_sum.data_ptr<float>(), db_sum.data_ptr<float>(),
                       (int)nt, (int)NTp, Fp);
    hipLaunchKernelGGL(bwd_k, bwd_grid, dim3(256), 0, s, Xp,
                       idx.data_ptr<int>(),
                       reinterpret_cast<const __bf16*>(Dt.data_ptr()),
                       Gt.data_ptr<float>(), (int)NTp, Fp, seg_rows);
    hipLaunchKernelGGL(col_finalize_kernel, fin_grid, dim3(256), 0, s,
                       loss_sum.data_ptr<float>(), db_sum.data_ptr<float>(),
                       bias.data_ptr<float>(), Vb.data_ptr<float>(),
                       prev_loss.data_ptr<float>(), cur_loss.data_ptr<float>(),
                       active.data_ptr<int>(), epoch_mask.data_ptr<int>(),
                       n_iter.data_ptr<int>(), max_iter.data_ptr<int>(),
                       lr.data_ptr<float>(), mom.data_ptr<float>(),
                       tol.data_ptr<float>(),
                       fit_ic.data_ptr<unsigned char>(), inv_n, epoch, Bp,
                       n_active.data_ptr<int>());
    hipLaunchKernelGGL(update_w_kernel, upd_grid2, dim3(upd_threads), 0, s,
                       W.data_ptr<float>(), V.data_ptr<float>(),
                       Gt.data_ptr<float>(),
                       reinterpret_cast<__bf16*>(Wt.data_ptr()),
                       epoch_mask.data_ptr<int>(), lr.data_ptr<float>(),
                       mom.data_ptr<float>(), l1.data_ptr<float>(),
                       l2.data_ptr<float>(), inv_n, Fp);
  };

  for (int epoch = 0; epoch < (int)max_epochs; ++epoch) {
    switch (g_layout) {
      case 0: launch_epoch(epoch, fwd_d_kernel<0>, bwd_g_kernel<0>); break;
      case 1: launch_epoch(epoch, fwd_d_kernel<1>, bwd_g_kernel<1>); break;
      case 2: launch_epoch(epoch, fwd_d_kernel<2>, bwd_g_kernel<2>); break;
      default: launch_epoch(epoch, fwd_d_kernel<3>, bwd_g_kernel<3>); break;
    }
    HIP_CHECK(hipGetLastError());
    if (check_every > 0 && (epoch + 1) % check_every == 0 &&
        epoch + 1 < max_epochs) {
      int na = n_active.item<int>();   // syncs the stream
      if (na == 0) break;
      n_active.zero_();
    } else {
      // zero without sync: n_active is recomputed every finalize
      hipMemsetAsync(n_active.data_ptr(), 0, sizeof(int), s);
    }
  }
  return {W, bias, n_iter, cur_loss, Wt};
}

// linear_scores: per-candidate sums; host finalizes means/r2.
torch::Tensor linear_scores(torch::Tensor X, torch::Tensor y,
                            torch::Tensor idx, torch::Tensor Wt_bf16,
                            torch::Tensor bias, torch::Tensor loss_id,
                            torch::Tensor target, int64_t C, int64_t metric,
                            int64_t n_cand) {
  TORCH_CHECK(g_layout >= 0, "MFMA layout not probed");
  const at::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  int Fp = (int)X.size(1);
  int Bp = (int)Wt_bf16.size(0);
  TORCH_CHECK(Bp == 64 || Bp == 128 || Bp == 256,
              "scores kernel supports Bp in {64,128,256}");
  TORCH_CHECK(idx.dtype() == torch::kInt64);
  long n_idx = idx.size(0);
  auto out = torch::zeros({n_cand}, X.options().dtype(torch::kFloat64));

  size_t scand_pad = ((size_t)n_cand + 3) & ~(size_t)3;
  size_t smem = 32 * 8 + 32 * (Bp + 8) * 4 + scand_pad * 4 +
                32 * 72 * 2 + (size_t)Bp * 72 * 2;
  dim3 grid((unsigned)((n_idx + 31) / 32));
  hipStream_t s = cur_stream();

  auto launch = [&](auto k) {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(k),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem));
    hipLaunchKernelGGL(k, grid, dim3(256), smem, s,
                       reinterpret_cast<const __bf16*>(X.data_ptr()),
                       y.data_ptr<float>(), idx.data_ptr<long>(),
                       reinterpret_cast<const __bf16*>(Wt_bf16.data_ptr()),
                       bias.data_ptr<float>(), loss_id.data_ptr<int>(),
                       target.data_ptr<int>(), (int)C, (int)metric, n_idx,
                       Fp, Bp, (int)n_cand, out.data_ptr<double>());
  };
  switch (g_layout) {
    case 0: launch(scores_kernel<0>); break;
    case 1: launch(scores_kernel<1>); break;
    case 2: launch(scores_kernel<2>); break;
    default: launch(scores_kernel<3>); break;
  }
  HIP_CHECK(hipGetLastError());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_probe", &gemm_probe, "MFMA layout probe (4 combos)");
  m.def("set_layout", &set_layout);
  m.def("get_layout", &get_layout);
  m.def("msq_rows", &msq_rows, "mean squared row norm");
  m.def("linear_fit", &linear_fit, "batched linear fit (device epoch loop)");
  m.def("linear_scores", &linear_scores, "batched per-candidate scores");
}
