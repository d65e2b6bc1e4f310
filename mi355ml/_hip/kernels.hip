// mi355ml CDNA4 (gfx950) kernels — hand-written HIP, MI355X-first.
//
// Kernel inventory: K1–K3 fused GLM objective+gradient,
// K4+K5 fused KMeans assign + centroid partials, K7/K8 column moments &
// min/max, K9 per-column affine transform.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64; blocks of 256 threads (4 waves).
//  - All cross-workgroup combining goes through per-WG partial slabs reduced
//    deterministically by the caller (no global atomics -> bitwise
//    run-to-run determinism).
//  - fp32 data, fp64 accumulators for statistics.
//  - Memory-bound kernels: coalesced row-major traversal, LDS staging where
//    a tile is read twice (GLM: X used for both z=X·w and g=Xᵀr).

#include <hip/hip_runtime.h>
#include <math.h>

#define NT 256      // threads per block
#define GLM_TR 64   // GLM row-tile height

// ---------------------------------------------------------------------------
// K7/K8: column moments (n, sum, sumsq->M2) — one pass, per-WG partials.
// Thread t covers column c = c0 + (t % cpad) for a power-of-2 cpad >= cols,
// so a wave's lanes read consecutive addresses within each row (coalesced).
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NT) void col_moments_partial_k(
    const float* __restrict__ x, long m, int f,
    double* __restrict__ ns, double* __restrict__ sums,
    double* __restrict__ m2s) {
  __shared__ double s_sum[NT];
  __shared__ double s_ssq[NT];
  __shared__ long s_cnt[NT];
  const int w = blockIdx.x;
  const int tid = threadIdx.x;
  for (int c0 = 0; c0 < f; c0 += NT) {
    const int cols = min(NT, f - c0);
    int cpad = 4;
    while (cpad < cols) cpad <<= 1;
    if (cpad > NT) cpad = NT;
    const int rpi = NT / cpad;  // rows advanced per iteration
    const int c = tid % cpad;
    const int rs = tid / cpad;
    double sum = 0.0, ssq = 0.0;
    long cnt = 0;
    if (c < cols) {
      long r = (long)w * rpi + rs;
      const long stride = (long)gridDim.x * rpi;
      const float* xp = x + (long)(c0 + c);
      for (; r < m; r += stride) {
        const double v = (double)xp[r * (long)f];
        sum += v;
        ssq += v * v;
        ++cnt;
      }
    }
    s_sum[tid] = sum;
    s_ssq[tid] = ssq;
    s_cnt[tid] = cnt;
    __syncthreads();
    if (tid < cpad && tid < cols) {
      double tsum = 0.0, tssq = 0.0;
      long tc = 0;
      for (int i = tid; i < cpad * rpi; i += cpad) {
        tsum += s_sum[i];
        tssq += s_ssq[i];
        tc += s_cnt[i];
      }
      const double mean = tc ? tsum / (double)tc : 0.0;
      double m2 = tc ? tssq - tsum * mean : 0.0;
      if (m2 < 0.0) m2 = 0.0;
      sums[(long)w * f + c0 + tid] = tsum;
      m2s[(long)w * f + c0 + tid] = m2;
      if (c0 == 0 && tid == 0) ns[w] = (double)tc;
    }
    __syncthreads();
  }
}

extern "C" __global__ __launch_bounds__(NT) void col_minmax_partial_k(
    const float* __restrict__ x, long m, int f,
    float* __restrict__ mins, float* __restrict__ maxs) {
  __shared__ float s_min[NT];
  __shared__ float s_max[NT];
  const int w = blockIdx.x;
  const int tid = threadIdx.x;
  for (int c0 = 0; c0 < f; c0 += NT) {
    const int cols = min(NT, f - c0);
    int cpad = 4;
    while (cpad < cols) cpad <<= 1;
    if (cpad > NT) cpad = NT;
    const int rpi = NT / cpad;
    const int c = tid % cpad;
    const int rs = tid / cpad;
    float mn = INFINITY, mx = -INFINITY;
    if (c < cols) {
      long r = (long)w * rpi + rs;
      const long stride = (long)gridDim.x * rpi;
      const float* xp = x + (long)(c0 + c);
      for (; r < m; r += stride) {
        const float v = xp[r * (long)f];
        mn = fminf(mn, v);
        mx = fmaxf(mx, v);
      }
    }
    s_min[tid] = mn;
    s_max[tid] = mx;
    __syncthreads();
    if (tid < cpad && tid < cols) {
      float tmn = INFINITY, tmx = -INFINITY;
      for (int i = tid; i < cpad * rpi; i += cpad) {
        tmn = fminf(tmn, s_min[i]);
        tmx = fmaxf(tmx, s_max[i]);
      }
      mins[(long)w * f + c0 + tid] = tmn;
      maxs[(long)w * f + c0 + tid] = tmx;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// K1+K2+K3: fused GLM objective + gradient.
// One pass over X per objective evaluation: each 64-row tile is staged once
// in LDS (coalesced), used for z = x·w (z-phase, 4 partial lanes per row,
// shfl-combined) and for the g += r·x rank-1 accumulation (grad-phase,
// per-thread f64 column accumulators held in registers across tiles).
// Families: 0 logistic, 1 normal, 2 poisson (ops/glm.py order).
// Requires d <= 2*NT = 512.
// LDS carve (all dynamic, 16B-aligned base): w[d] | xtile[TR*ld] | y[TR] |
// rres[TR] | red[2*NT doubles].
// ---------------------------------------------------------------------------
template <int FAM>
__global__ __launch_bounds__(NT) void glm_value_grad_k(
    const float* __restrict__ x, const float* __restrict__ y,
    const float* __restrict__ w, float b, long m, int d, int ld,
    double* __restrict__ loss_slab, double* __restrict__ gw_slab,
    double* __restrict__ gb_slab) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* sw = (float*)smem_raw;                    // [d]
  float* xt = sw + ((d + 3) & ~3);                 // [TR * ld]
  float* yt = xt + GLM_TR * ld;                    // [TR]
  float* rres = yt + GLM_TR;                       // [TR]
  double* red = (double*)(smem_raw +
      (((((d + 3) & ~3) + GLM_TR * ld + 2 * GLM_TR) * sizeof(float) + 15) & ~15));
  const int tid = threadIdx.x;
  for (int i = tid; i < d; i += NT) sw[i] = w[i];

  double lacc = 0.0, gbacc = 0.0;
  double gacc0 = 0.0, gacc1 = 0.0;  // column slots tid and tid+NT

  const long tiles = (m + GLM_TR - 1) / GLM_TR;
  for (long tile = blockIdx.x; tile < tiles; tile += gridDim.x) {
    const long r0 = tile * GLM_TR;
    const int rows = (int)min((long)GLM_TR, m - r0);
    __syncthreads();  // previous grad-phase done before overwriting xt
    for (int i = tid; i < rows * d; i += NT) {
      const int rr = i / d;
      const int cc = i - rr * d;
      xt[rr * ld + cc] = x[(r0 + rr) * (long)d + cc];
    }
    for (int i = tid; i < rows; i += NT) yt[i] = y[r0 + i];
    __syncthreads();
    // z-phase: 4 lanes per row
    {
      const int r = tid >> 2;
      const int q = tid & 3;
      float dot = 0.0f;
      if (r < rows) {
        const float* xrow = xt + r * ld;
        for (int j = q; j < d; j += 4) dot += xrow[j] * sw[j];
      }
      dot += __shfl_xor(dot, 1, 4);
      dot += __shfl_xor(dot, 2, 4);
      if (r < rows && q == 0) {
        const double z = (double)(dot + b);
        const double yv = (double)yt[r];
        double li, ri;
        if (FAM == 0) {
          const double sp = z > 0.0 ? z + log1p(exp(-z)) : log1p(exp(z));
          li = sp - yv * z;
          ri = 1.0 / (1.0 + exp(-z)) - yv;
        } else if (FAM == 1) {
          const double df = z - yv;
          li = 0.5 * df * df;
          ri = df;
        } else {
          const double ez = exp(z);
          li = ez - yv * z;
          ri = ez - yv;
        }
        lacc += li;
        gbacc += ri;
        rres[r] = (float)ri;
      }
    }
    __syncthreads();
    // grad-phase: thread owns columns tid and tid+NT
    if (tid < d) {
      double acc = 0.0;
      const float* xp = xt + tid;
      for (int rr = 0; rr < rows; ++rr)
        acc += (double)rres[rr] * (double)xp[rr * ld];
      gacc0 += acc;
    }
    if (tid + NT < d) {
      double acc = 0.0;
      const float* xp = xt + tid + NT;
      for (int rr = 0; rr < rows; ++rr)
        acc += (double)rres[rr] * (double)xp[rr * ld];
      gacc1 += acc;
    }
  }
  // block reduction of loss / gb
  red[tid] = lacc;
  red[NT + tid] = gbacc;
  __syncthreads();
  for (int s = NT / 2; s > 0; s >>= 1) {
    if (tid < s) {
      red[tid] += red[tid + s];
      red[NT + tid] += red[NT + tid + s];
    }
    __syncthreads();
  }
  if (tid == 0) {
    loss_slab[blockIdx.x] = red[0];
    gb_slab[blockIdx.x] = red[NT];
  }
  if (tid < d) gw_slab[(long)blockIdx.x * d + tid] = gacc0;
  if (tid + NT < d) gw_slab[(long)blockIdx.x * d + tid + NT] = gacc1;
}

// ---------------------------------------------------------------------------
// K4+K5: fused KMeans assignment (+ optional centroid partials).
// One thread per row, x-row register-resident (d <= 128), centers staged in
// LDS (lane-uniform reads broadcast). Centroid partial sums/counts
// accumulate in per-WG LDS bins (LDS atomics) and flush once per WG into a
// slab; inertia reduces per-WG in f64.
// ---------------------------------------------------------------------------
#define KM_MAXD 128

template <int UPDATE>
__global__ __launch_bounds__(NT) void kmeans_assign_k(
    const float* __restrict__ x, const float* __restrict__ cen, long m,
    int d, int k, int* __restrict__ labels, float* __restrict__ dists,
    float* __restrict__ sums_slab, float* __restrict__ cnt_slab,
    double* __restrict__ inertia_slab) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* sc = (float*)smem_raw;        // centers [k*d]
  float* cn = sc + k * d;              // center norms [k]
  float* bins = cn + ((k + 3) & ~3);   // UPDATE: [k*d]
  float* bcnt = bins + (UPDATE ? k * d : 0);  // UPDATE: [k]
  double* red = (double*)(smem_raw +
      (((k * d + ((k + 3) & ~3) + (UPDATE ? (k * d + ((k + 3) & ~3)) : 0)) *
        sizeof(float) + 15) & ~15));   // [NT]
  const int tid = threadIdx.x;
  for (int i = tid; i < k * d; i += NT) sc[i] = cen[i];
  if (UPDATE) {
    for (int i = tid; i < k * d; i += NT) bins[i] = 0.0f;
    for (int i = tid; i < k; i += NT) bcnt[i] = 0.0f;
  }
  __syncthreads();
  for (int c = tid; c < k; c += NT) {
    float s = 0.0f;
    const float* cp = sc + c * d;
    for (int j = 0; j < d; ++j) s += cp[j] * cp[j];
    cn[c] = s;
  }
  __syncthreads();

  float xr[KM_MAXD];
  double iacc = 0.0;
  long r = (long)blockIdx.x * NT + tid;
  const long stride = (long)gridDim.x * NT;
  for (; r < m; r += stride) {
    const float* xrow = x + r * (long)d;
    float xn = 0.0f;
#pragma unroll 4
    for (int j = 0; j < d; ++j) {
      const float v = xrow[j];
      xr[j] = v;
      xn += v * v;
    }
    float best = INFINITY;
    int bi = 0;
    for (int c = 0; c < k; ++c) {
      const float* cp = sc + c * d;
      float dot = 0.0f;
#pragma unroll 4
      for (int j = 0; j < d; ++j) dot = fmaf(xr[j], cp[j], dot);
      const float dist = cn[c] - 2.0f * dot;
      if (dist < best) {
        best = dist;
        bi = c;
      }
    }
    labels[r] = bi;
    const float d2 = fmaxf(best + xn, 0.0f);
    if (UPDATE) {
      iacc += (double)d2;
      float* bp = bins + bi * d;
      for (int j = 0; j < d; ++j) atomicAdd(&bp[j], xr[j]);
      atomicAdd(&bcnt[bi], 1.0f);
    } else {
      dists[r] = d2;
    }
  }

  if (UPDATE) {
    __syncthreads();
    for (int i = tid; i < k * d; i += NT)
      sums_slab[(long)blockIdx.x * k * d + i] = bins[i];
    for (int i = tid; i < k; i += NT)
      cnt_slab[(long)blockIdx.x * k + i] = bcnt[i];
    red[tid] = iacc;
    __syncthreads();
    for (int s = NT / 2; s > 0; s >>= 1) {
      if (tid < s) red[tid] += red[tid + s];
      __syncthreads();
    }
    if (tid == 0) inertia_slab[blockIdx.x] = red[0];
  }
}

// ---------------------------------------------------------------------------
// K9: per-column affine transform out = x*a[col] + b[col], one fused pass.
// float4-vectorized when the row length is a multiple of 4.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NT) void affine_transform_k(
    const float* __restrict__ x, const float* __restrict__ a,
    const float* __restrict__ b, float* __restrict__ out, long n, int d) {
  long i = (long)blockIdx.x * NT + threadIdx.x;
  const long stride = (long)gridDim.x * NT;
  if ((d & 3) == 0) {
    const long n4 = n >> 2;
    const float4* x4 = (const float4*)x;
    float4* o4 = (float4*)out;
    const int d4 = d >> 2;
    for (; i < n4; i += stride) {
      const int c4 = (int)(i % d4) << 2;
      const float4 v = x4[i];
      float4 o;
      o.x = fmaf(v.x, a[c4], b[c4]);
      o.y = fmaf(v.y, a[c4 + 1], b[c4 + 1]);
      o.z = fmaf(v.z, a[c4 + 2], b[c4 + 2]);
      o.w = fmaf(v.w, a[c4 + 3], b[c4 + 3]);
      o4[i] = o;
    }
  } else {
    for (; i < n; i += stride) {
      const int c = (int)(i % d);
      out[i] = fmaf(x[i], a[c], b[c]);
    }
  }
}

// ---------------------------------------------------------------------------
// C-visible launchers
// ---------------------------------------------------------------------------
extern "C" {

void launch_col_moments(const float* x, long m, int f, double* ns,
                        double* sums, double* m2s, int nblocks,
                        hipStream_t stream) {
  hipLaunchKernelGGL(col_moments_partial_k, dim3(nblocks), dim3(NT), 0,
                     stream, x, m, f, ns, sums, m2s);
}

void launch_col_minmax(const float* x, long m, int f, float* mins,
                       float* maxs, int nblocks, hipStream_t stream) {
  hipLaunchKernelGGL(col_minmax_partial_k, dim3(nblocks), dim3(NT), 0,
                     stream, x, m, f, mins, maxs);
}

void launch_glm_value_grad(const float* x, const float* y, const float* w,
                           float b, long m, int d, int family,
                           double* loss_slab, double* gw_slab,
                           double* gb_slab, int nblocks, hipStream_t stream) {
  const int ld = d | 1;  // odd LDS stride -> conflict-free row & col reads
  const size_t smem =
      ((((((d + 3) & ~3) + GLM_TR * ld + 2 * GLM_TR) * sizeof(float) + 15) &
        ~15)) + 2 * NT * sizeof(double);
  switch (family) {
    case 0:
      hipLaunchKernelGGL((glm_value_grad_k<0>), dim3(nblocks), dim3(NT), smem,
                         stream, x, y, w, b, m, d, ld, loss_slab, gw_slab,
                         gb_slab);
      break;
    case 1:
      hipLaunchKernelGGL((glm_value_grad_k<1>), dim3(nblocks), dim3(NT), smem,
                         stream, x, y, w, b, m, d, ld, loss_slab, gw_slab,
                         gb_slab);
      break;
    default:
      hipLaunchKernelGGL((glm_value_grad_k<2>), dim3(nblocks), dim3(NT), smem,
                         stream, x, y, w, b, m, d, ld, loss_slab, gw_slab,
                         gb_slab);
  }
}

size_t kmeans_smem_bytes(int k, int d, int update) {
  size_t f32s = (size_t)k * d + ((k + 3) & ~3);
  if (update) f32s += (size_t)k * d + ((k + 3) & ~3);
  return (((f32s * sizeof(float) + 15) & ~15)) + NT * sizeof(double);
}

void launch_kmeans_assign(const float* x, const float* cen, long m, int d,
                          int k, int* labels, float* sums_slab,
                          float* cnt_slab, double* inertia_slab, int nblocks,
                          hipStream_t stream) {
  const size_t smem = kmeans_smem_bytes(k, d, 1);
  hipLaunchKernelGGL((kmeans_assign_k<1>), dim3(nblocks), dim3(NT), smem,
                     stream, x, cen, m, d, k, labels, (float*)nullptr,
                     sums_slab, cnt_slab, inertia_slab);
}

void launch_closest_center(const float* x, const float* cen, long m, int d,
                           int k, int* labels, float* dists, int nblocks,
                           hipStream_t stream) {
  const size_t smem = kmeans_smem_bytes(k, d, 0);
  hipLaunchKernelGGL((kmeans_assign_k<0>), dim3(nblocks), dim3(NT), smem,
                     stream, x, cen, m, d, k, labels, dists, (float*)nullptr,
                     (float*)nullptr, (double*)nullptr);
}

void launch_affine_transform(const float* x, const float* a, const float* b,
                             float* out, long n, int d, int nblocks,
                             hipStream_t stream) {
  hipLaunchKernelGGL(affine_transform_k, dim3(nblocks), dim3(NT), 0, stream,
                     x, a, b, out, n, d);
}

}  // extern "C"
