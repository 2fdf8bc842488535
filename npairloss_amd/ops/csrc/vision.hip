// Fused vision kernels for the GoogLeNet path (gfx950).
//
// rocprof on the flagship bench showed the backbone's time sinks are NOT
// the convs: torch's eager LRN (pad + pow + avg_pool3d + div chain, fp32)
// and the atomic-based NHWC max-pool backward dominate.  These replace
// them with:
//   lrn_fwd / lrn_bwd       — Caffe across-channel LRN
//                             (scale_i = k + alpha/n * sum_{win} x_j^2,
//                              y = x * scale^-beta).  Three variants by
//                             shape: wave-shuffle (channel window taps come
//                             from NEIGHBOR LANES' registers — no LDS, no
//                             barriers; 4.7 TB/s), LDS pixel-group stencil,
//                             and a generic per-element fallback.  All use
//                             fast_powf (v_log+v_exp): hipcc's __powf is a
//                             ~750-instruction softfloat pow and was the
//                             entire kernel cost (profiles/kernels_r1.md).
//   maxpool3x3_fwd / _bwd   — kernel-3 max pool (stride 1 or 2, pad 1)
//                             storing a 1-byte argmax; backward is a
//                             deterministic GATHER over the <=9 covering
//                             windows per input element (no atomics).
//                             NHWC channels are processed 4 wide
//                             (8-byte bf16 loads) when C % 4 == 0.
// Both kernels address NHWC (channels_last, the training layout) or NCHW
// via strides; lanes walk the contiguous dimension so every access is
// coalesced.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#include "common.h"

// load/store helpers: compute in fp32 regardless of storage type
template <typename T> DEVINL float ldf(const T* p, long long i);
template <> DEVINL float ldf<float>(const float* p, long long i) { return p[i]; }
template <> DEVINL float ldf<__hip_bfloat16>(const __hip_bfloat16* p, long long i) {
  return __bfloat162float(p[i]);
}
template <typename T> DEVINL void stf(T* p, long long i, float v);
template <> DEVINL void stf<float>(float* p, long long i, float v) { p[i] = v; }
template <> DEVINL void stf<__hip_bfloat16>(__hip_bfloat16* p, long long i, float v) {
  p[i] = __float2bfloat16(v);
}

// 4-wide channel loads (NHWC fast path)
template <typename T> DEVINL void ld4(const T* p, float* o);
template <> DEVINL void ld4<float>(const float* p, float* o) {
  const float4 v = *reinterpret_cast<const float4*>(p);
  o[0] = v.x; o[1] = v.y; o[2] = v.z; o[3] = v.w;
}
template <> DEVINL void ld4<__hip_bfloat16>(const __hip_bfloat16* p, float* o) {
  const ushort4 v = *reinterpret_cast<const ushort4*>(p);
  o[0] = __bfloat162float(__hip_bfloat16_raw{v.x});
  o[1] = __bfloat162float(__hip_bfloat16_raw{v.y});
  o[2] = __bfloat162float(__hip_bfloat16_raw{v.z});
  o[3] = __bfloat162float(__hip_bfloat16_raw{v.w});
}
template <typename T> DEVINL void st4(T* p, const float* v);
template <> DEVINL void st4<float>(float* p, const float* v) {
  *reinterpret_cast<float4*>(p) = make_float4(v[0], v[1], v[2], v[3]);
}
template <> DEVINL void st4<__hip_bfloat16>(__hip_bfloat16* p, const float* v) {
  ushort4 u;
  u.x = static_cast<__hip_bfloat16_raw>(__float2bfloat16(v[0])).x;
  u.y = static_cast<__hip_bfloat16_raw>(__float2bfloat16(v[1])).x;
  u.z = static_cast<__hip_bfloat16_raw>(__float2bfloat16(v[2])).x;
  u.w = static_cast<__hip_bfloat16_raw>(__float2bfloat16(v[3])).x;
  *reinterpret_cast<ushort4*>(p) = u;
}

// ---------------------------------------------------------------------------
// LRN — LDS-stencil variant (C <= 256) + generic per-element fallback
// ---------------------------------------------------------------------------
// "Pixel" = one (b, h, w) site (NHWC: its C channels contiguous) or one
// (b, s) site for NCHW (channel stride S).  A 256-thread block processes
// 256/C_pad pixels at once: xsq (and the dy*y/scale cross terms in the
// backward) land in LDS once, each window sum reads 5 LDS floats.

template <typename T, bool BWD>
__global__ void lrn_tile_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                                T* __restrict__ out, long long npix, int C, int C_pad,
                                long long cstride, long long pix_to_base_mul,
                                long long S, int n, float alpha_over_n, float beta,
                                float k) {
  __shared__ float xsq[NPAIR_BLOCK];
  __shared__ float tbuf[NPAIR_BLOCK];
  const int half = n / 2;
  const int ppb = NPAIR_BLOCK / C_pad;        // pixels per block
  const int pl = threadIdx.x / C_pad;         // pixel slot in block
  const int c = threadIdx.x % C_pad;
  const int lds0 = pl * C_pad;
  const bool chan_ok = (c < C) && (pl < ppb);
  const long long tiles = (npix + ppb - 1) / ppb;
  for (long long tile = blockIdx.x; tile < tiles; tile += gridDim.x) {
    const long long pix = tile * ppb + pl;
    const bool act = chan_ok && pix < npix;
    long long base = 0, idx = 0;
    float xi = 0.f, gi = 0.f;
    if (act) {
      if (cstride == 1) {
        base = pix * C;                       // NHWC
      } else {
        base = (pix / S) * (C * S) + (pix % S);  // NCHW: pix = (b, s)
      }
      idx = base + (long long)c * cstride;
      xi = ldf(x, idx);
      if (BWD) gi = ldf(dy, idx);
    }
    xsq[threadIdx.x] = act ? xi * xi : 0.f;
    __syncthreads();
    const int lo = max(0, c - half);
    const int hi = min(C - 1, c + half);
    float ss = 0.f;
    for (int j = lo; j <= hi; ++j) ss += xsq[lds0 + j];
    const float scale = k + alpha_over_n * ss;
    const float p = fast_powf(scale, -beta);     // ONE powf per element
    if (!BWD) {
      if (act) stf(out, idx, xi * p);
      __syncthreads();
      continue;
    }
    const float yi = xi * p;
    tbuf[threadIdx.x] = act ? gi * yi / scale : 0.f;
    __syncthreads();
    float cross = 0.f;
    for (int j = lo; j <= hi; ++j) cross += tbuf[lds0 + j];
    if (act) stf(out, idx, gi * p - 2.f * alpha_over_n * beta * xi * cross);
    __syncthreads();
  }
}

// 4-channels-per-thread variant (NHWC, C % 4 == 0, n <= 9, C <= 1024):
// 8-byte bf16 global loads/stores; the channel rows live in LDS with a
// 4-float zero halo on each side so each thread's window sums come from
// THREE aligned float4 reads (conflict-free ds_read_b128 — the naive
// per-channel scalar reads at stride 4 are a 4-way bank conflict).
// Out-of-range window taps read halo zeros == the clamp semantics.
#define LRN_HALO 4

template <typename T, bool BWD, int HALF>
__global__ void lrn_tile4_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                                 T* __restrict__ out, long long npix, int C,
                                 int n, float alpha_over_n, float beta, float k) {
  __shared__ float xsq[NPAIR_BLOCK * 5];
  __shared__ float tbuf[NPAIR_BLOCK * 5];
  constexpr int half = HALF;
  const int C_q = C / 4;                      // threads per pixel
  const int ppb = NPAIR_BLOCK / C_q;          // pixels per block
  const int row = C + 2 * LRN_HALO;           // padded LDS row per pixel
  const int pl = threadIdx.x / C_q;
  const int c0 = (threadIdx.x % C_q) * 4;
  const int lds0 = pl * row + LRN_HALO;
  const bool chan_ok = pl < ppb;
  // zero the halos (and everything) once; interiors are rewritten per tile
  for (int i = threadIdx.x; i < ppb * row; i += blockDim.x) {
    xsq[i] = 0.f;
    tbuf[i] = 0.f;
  }
  __syncthreads();
  const long long tiles = (npix + ppb - 1) / ppb;
  for (long long tile = blockIdx.x; tile < tiles; tile += gridDim.x) {
    const long long pix = tile * ppb + pl;
    const bool act = chan_ok && pix < npix;
    const long long base = pix * C + c0;      // NHWC
    float xi[4] = {0.f, 0.f, 0.f, 0.f};
    float gi[4] = {0.f, 0.f, 0.f, 0.f};
    if (act) {
      ld4(x + base, xi);
      if (BWD) ld4(dy + base, gi);
    }
    if (chan_ok)
      *reinterpret_cast<float4*>(&xsq[lds0 + c0]) =
          make_float4(xi[0] * xi[0], xi[1] * xi[1], xi[2] * xi[2], xi[3] * xi[3]);
    __syncthreads();
    // window sums for channels [c0, c0+3] from the 12-float neighborhood
    float buf[12];
    {
      const float4 a = *reinterpret_cast<const float4*>(&xsq[lds0 + c0 - 4]);
      const float4 b = *reinterpret_cast<const float4*>(&xsq[lds0 + c0]);
      const float4 c = *reinterpret_cast<const float4*>(&xsq[lds0 + c0 + 4]);
      buf[0] = a.x; buf[1] = a.y; buf[2] = a.z; buf[3] = a.w;
      buf[4] = b.x; buf[5] = b.y; buf[6] = b.z; buf[7] = b.w;
      buf[8] = c.x; buf[9] = c.y; buf[10] = c.z; buf[11] = c.w;
    }
    float scale[4], p[4];
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      float ss = 0.f;
#pragma unroll
      for (int j = -half; j <= half; ++j) ss += buf[4 + v + j];  // halo zeros = clamp
      scale[v] = k + alpha_over_n * ss;
      p[v] = fast_powf(scale[v], -beta);
    }
    if (!BWD) {
      if (act) {
        float yv[4];
#pragma unroll
        for (int v = 0; v < 4; ++v) yv[v] = xi[v] * p[v];
        st4(out + base, yv);
      }
      __syncthreads();
      continue;
    }
    if (chan_ok)
      *reinterpret_cast<float4*>(&tbuf[lds0 + c0]) =
          make_float4(gi[0] * (xi[0] * p[0]) / scale[0], gi[1] * (xi[1] * p[1]) / scale[1],
                      gi[2] * (xi[2] * p[2]) / scale[2], gi[3] * (xi[3] * p[3]) / scale[3]);
    __syncthreads();
    {
      const float4 a = *reinterpret_cast<const float4*>(&tbuf[lds0 + c0 - 4]);
      const float4 b = *reinterpret_cast<const float4*>(&tbuf[lds0 + c0]);
      const float4 c = *reinterpret_cast<const float4*>(&tbuf[lds0 + c0 + 4]);
      buf[0] = a.x; buf[1] = a.y; buf[2] = a.z; buf[3] = a.w;
      buf[4] = b.x; buf[5] = b.y; buf[6] = b.z; buf[7] = b.w;
      buf[8] = c.x; buf[9] = c.y; buf[10] = c.z; buf[11] = c.w;
    }
    if (act) {
      float dxv[4];
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        float cross = 0.f;
#pragma unroll
        for (int j = -half; j <= half; ++j) cross += buf[4 + v + j];
        dxv[v] = gi[v] * p[v] - 2.f * alpha_over_n * beta * xi[v] * cross;
      }
      st4(out + base, dxv);
    }
    __syncthreads();
  }
}

// Wave-shuffle variant (NHWC, C % 4 == 0, 64 % (C/4) == 0, n <= 5): each
// lane owns 4 channels; the window's out-of-quad taps come from the
// NEIGHBOR LANES' registers via __shfl_up/__shfl_down — no LDS, no
// barriers at all.  Pixel boundaries align with lane groups (C/4 divides
// the 64-lane wave), so boundary lanes just zero their cross-pixel taps.
template <typename T, bool BWD>
__global__ void lrn_wave_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                                T* __restrict__ out, long long npix, int C,
                                int C_q_log2, int half, float alpha_over_n,
                                float beta, float k) {
  const int C_q = 1 << C_q_log2;
  const long long nquads = npix << C_q_log2;
  for (long long q = (long long)blockIdx.x * blockDim.x + threadIdx.x; q < nquads;
       q += (long long)gridDim.x * blockDim.x) {
    const long long pix = q >> C_q_log2;
    const int c0 = (int)(q & (C_q - 1)) * 4;
    const long long base = pix * C + c0;
    float xi[4], gi[4] = {0.f, 0.f, 0.f, 0.f};
    ld4(x + base, xi);
    if (BWD) ld4(dy + base, gi);
    float xsq[4];
#pragma unroll
    for (int v = 0; v < 4; ++v) xsq[v] = xi[v] * xi[v];
    // neighbor taps: left lane's top-2, right lane's bottom-2
    const bool at_lo = (c0 == 0);
    const bool at_hi = (c0 + 4 >= C);
    float l2 = __shfl_up(xsq[2], 1);
    float l3 = __shfl_up(xsq[3], 1);
    float r0 = __shfl_down(xsq[0], 1);
    float r1 = __shfl_down(xsq[1], 1);
    if (at_lo) l2 = l3 = 0.f;
    if (at_hi) r0 = r1 = 0.f;
    const float buf[8] = {l2, l3, xsq[0], xsq[1], xsq[2], xsq[3], r0, r1};
    float scale[4], p[4];
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      float ss = 0.f;
#pragma unroll
      for (int j = -2; j <= 2; ++j) {
        const int idx = 2 + v + j;
        // window half may be 1 (n=3): drop the +-2 taps then
        ss += (j >= -half && j <= half) ? buf[idx] : 0.f;
      }
      scale[v] = k + alpha_over_n * ss;
      p[v] = fast_powf(scale[v], -beta);
    }
    if (!BWD) {
      float yv[4];
#pragma unroll
      for (int v = 0; v < 4; ++v) yv[v] = xi[v] * p[v];
      st4(out + base, yv);
      continue;
    }
    float t4[4];
#pragma unroll
    for (int v = 0; v < 4; ++v) t4[v] = gi[v] * (xi[v] * p[v]) / scale[v];
    float tl2 = __shfl_up(t4[2], 1);
    float tl3 = __shfl_up(t4[3], 1);
    float tr0 = __shfl_down(t4[0], 1);
    float tr1 = __shfl_down(t4[1], 1);
    if (at_lo) tl2 = tl3 = 0.f;
    if (at_hi) tr0 = tr1 = 0.f;
    const float tb[8] = {tl2, tl3, t4[0], t4[1], t4[2], t4[3], tr0, tr1};
    float dxv[4];
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      float cross = 0.f;
#pragma unroll
      for (int j = -2; j <= 2; ++j) {
        const int idx = 2 + v + j;
        cross += (j >= -half && j <= half) ? tb[idx] : 0.f;
      }
      dxv[v] = gi[v] * p[v] - 2.f * alpha_over_n * beta * xi[v] * cross;
    }
    st4(out + base, dxv);
  }
}

// generic per-element fallback (any C): used when C > 256
template <typename T>
__global__ void lrn_fwd_generic(const T* __restrict__ x, T* __restrict__ y,
                                long long total, int C, long long cstride, int n,
                                float alpha_over_n, float beta, float k) {
  const int half = n / 2;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const long long c = (i / cstride) % C;
    const long long base = i - c * cstride;
    const int lo = max((long long)0, c - half);
    const int hi = min((long long)C - 1, c + half);
    float ss = 0.f;
    for (int j = lo; j <= hi; ++j) {
      const float v = ldf(x, base + (long long)j * cstride);
      ss += v * v;
    }
    stf(y, i, ldf(x, i) * fast_powf(k + alpha_over_n * ss, -beta));
  }
}

template <typename T>
__global__ void lrn_bwd_generic(const T* __restrict__ x, const T* __restrict__ dy,
                                T* __restrict__ dx, long long total, int C,
                                long long cstride, int n, float alpha_over_n,
                                float beta, float k) {
  const int half = n / 2;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const long long c = (i / cstride) % C;
    const long long base = i - c * cstride;
    const int lo = max((long long)0, c - half);
    const int hi = min((long long)C - 1, c + half);
    float cross = 0.f, scale_i = 0.f;
    for (int j = lo; j <= hi; ++j) {
      const int jlo = max(0, j - half);
      const int jhi = min(C - 1, j + half);
      float ss = 0.f;
      for (int m = jlo; m <= jhi; ++m) {
        const float v = ldf(x, base + (long long)m * cstride);
        ss += v * v;
      }
      const float scale_j = k + alpha_over_n * ss;
      if (j == (int)c) scale_i = scale_j;
      const float yj = ldf(x, base + (long long)j * cstride) * fast_powf(scale_j, -beta);
      cross += ldf(dy, base + (long long)j * cstride) * yj / scale_j;
    }
    const float xi = ldf(x, i);
    stf(dx, i, ldf(dy, i) * fast_powf(scale_i, -beta) - 2.f * alpha_over_n * beta * xi * cross);
  }
}

// ---------------------------------------------------------------------------
// 3x3 max pool, pad 1, stride 1 or 2 — V channels per thread (V = 4 or 1)
// ---------------------------------------------------------------------------

template <typename T, int V>
__global__ void maxpool3_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                    unsigned char* __restrict__ idx,
                                    int B, int Cv, int H, int W, int OH, int OW,
                                    int stride,
                                    long long xsb, long long xsh, long long xsw, long long xsc,
                                    long long ysb, long long ysh, long long ysw, long long ysc) {
  // Cv = C / V; i enumerates ((b*OH + oh)*OW + ow)*Cv + cv  (cv fastest).
  // All supported shapes fit 32 bits -> uint division (the 64-bit div/mod
  // chain showed up as ~40 VALU instructions per element in PMC counts).
  const unsigned int total = (unsigned int)((long long)B * OH * OW * Cv);
  for (unsigned int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    const int cv = i % (unsigned int)Cv;
    unsigned int r = i / (unsigned int)Cv;
    const int ow = r % (unsigned int)OW;
    r /= (unsigned int)OW;
    const int oh = r % (unsigned int)OH;
    const int b = r / (unsigned int)OH;
    const int h0 = oh * stride - 1;
    const int w0 = ow * stride - 1;
    float best[V];
    int besti[V];
#pragma unroll
    for (int v = 0; v < V; ++v) {
      best[v] = -FLT_MAX;
      besti[v] = 0;
    }
    const long long xb = (long long)b * xsb + (long long)(cv * V) * xsc;
#pragma unroll
    for (int dh = 0; dh < 3; ++dh) {
      const int h = h0 + dh;
      if (h < 0 || h >= H) continue;
#pragma unroll
      for (int dw = 0; dw < 3; ++dw) {
        const int w = w0 + dw;
        if (w < 0 || w >= W) continue;
        const long long a = xb + (long long)h * xsh + (long long)w * xsw;
        float vals[V];
        if (V == 4) {
          ld4(x + a, vals);
        } else {
          vals[0] = ldf(x, a);
        }
#pragma unroll
        for (int v = 0; v < V; ++v)
          if (vals[v] > best[v]) {
            best[v] = vals[v];
            besti[v] = dh * 3 + dw;
          }
      }
    }
    const long long yi = (long long)b * ysb + (long long)oh * ysh + (long long)ow * ysw
                       + (long long)(cv * V) * ysc;
    if (V == 4) {
      st4(y + yi, best);
      uchar4 u;
      u.x = besti[0]; u.y = besti[1]; u.z = besti[2]; u.w = besti[3];
      *reinterpret_cast<uchar4*>(idx + yi) = u;
    } else {
      stf(y, yi, best[0]);
      idx[yi] = (unsigned char)besti[0];
    }
  }
}

template <typename T, int V>
__global__ void maxpool3_bwd_kernel(const T* __restrict__ dy,
                                    const unsigned char* __restrict__ idx,
                                    T* __restrict__ dx,
                                    int B, int Cv, int H, int W, int OH, int OW,
                                    int stride,
                                    long long xsb, long long xsh, long long xsw, long long xsc,
                                    long long ysb, long long ysh, long long ysw, long long ysc) {
  const unsigned int total = (unsigned int)((long long)B * H * W * Cv);
  for (unsigned int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    const int cv = i % (unsigned int)Cv;
    unsigned int r = i / (unsigned int)Cv;
    const int w = r % (unsigned int)W;
    r /= (unsigned int)W;
    const int h = r % (unsigned int)H;
    const int b = r / (unsigned int)H;
    float acc[V];
#pragma unroll
    for (int v = 0; v < V; ++v) acc[v] = 0.f;
    const long long yb = (long long)b * ysb + (long long)(cv * V) * ysc;
    // output windows covering (h, w): oh*stride - 1 <= h <= oh*stride + 1,
    // i.e. oh in [ceil((h-1)/s), floor((h+1)/s)] (negative lower clamps to 0)
    const int oh_lo = max(0, (h - 1 + stride - 1) / stride);
    const int oh_hi = min(OH - 1, (h + 1) / stride);
    const int ow_lo = max(0, (w - 1 + stride - 1) / stride);
    const int ow_hi = min(OW - 1, (w + 1) / stride);
    for (int oh = oh_lo; oh <= oh_hi; ++oh) {
      const int dh = h - (oh * stride - 1);
      if (dh < 0 || dh > 2) continue;
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        const int dw = w - (ow * stride - 1);
        if (dw < 0 || dw > 2) continue;
        const long long yi = yb + (long long)oh * ysh + (long long)ow * ysw;
        const unsigned char want = (unsigned char)(dh * 3 + dw);
        if (V == 4) {
          const uchar4 u = *reinterpret_cast<const uchar4*>(idx + yi);
          if (u.x == want || u.y == want || u.z == want || u.w == want) {
            float g[4];
            ld4(dy + yi, g);
            if (u.x == want) acc[0] += g[0];
            if (u.y == want) acc[1] += g[1];
            if (u.z == want) acc[2] += g[2];
            if (u.w == want) acc[3] += g[3];
          }
        } else {
          if (idx[yi] == want) acc[0] += ldf(dy, yi);
        }
      }
    }
    const long long xi = (long long)b * xsb + (long long)h * xsh + (long long)w * xsw
                       + (long long)(cv * V) * xsc;
    if (V == 4) {
      st4(dx + xi, acc);
    } else {
      stf(dx, xi, acc[0]);
    }
  }
}

// ---------------------------------------------------------------------------
// stride-1 rolling-column FORWARD (NHWC, V=4): one thread walks a whole
// output row (b, oh, cv).  Consecutive 3x3 windows overlap in 2 columns:
// a 3-column register ring (3 rows x 4 channels) shifts left each step and
// loads only the NEW column — 1/3 the taps of the per-element kernel.
// ---------------------------------------------------------------------------

struct MpXCol {
  float v[3][4];  // x values for the 3 window rows x 4 channels
};

template <typename T>
DEVINL void mp_load_xcol(MpXCol& c, const T* __restrict__ x, int w, int W,
                         long long xsw, const long long xrow[3],
                         const int rvalid[3]) {
#pragma unroll
  for (int rr = 0; rr < 3; ++rr) {
    if (rvalid[rr] && w >= 0 && w < W) {
      ld4(x + xrow[rr] + (long long)w * xsw, c.v[rr]);
    } else {
      c.v[rr][0] = c.v[rr][1] = c.v[rr][2] = c.v[rr][3] = -FLT_MAX;
    }
  }
}

template <typename T>
__global__ void maxpool3_fwd_s1_row_kernel(const T* __restrict__ x,
                                           T* __restrict__ y,
                                           unsigned char* __restrict__ idx,
                                           int B, int Cv, int H, int W,
                                           int OH, int OW,
                                           long long xsb, long long xsh,
                                           long long ysb, long long ysh) {
  const unsigned int total = (unsigned int)((long long)B * OH * Cv);
  for (unsigned int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    const int cv = i % (unsigned int)Cv;
    const unsigned int r = i / (unsigned int)Cv;
    const int oh = r % (unsigned int)OH;
    const int b = r / (unsigned int)OH;
    long long xrow[3];
    int rvalid[3];
#pragma unroll
    for (int rr = 0; rr < 3; ++rr) {
      const int h = oh - 1 + rr;
      rvalid[rr] = (h >= 0 && h < H);
      xrow[rr] = (long long)b * xsb + (long long)h * xsh + (long long)(cv * 4);
    }
    const long long xsw = (long long)Cv * 4;  // NHWC w-stride = C
    MpXCol colA, colB, colC;  // w = ow-1, ow, ow+1
    mp_load_xcol(colA, x, -1, W, xsw, xrow, rvalid);
    mp_load_xcol(colB, x, 0, W, xsw, xrow, rvalid);
    long long yi = (long long)b * ysb + (long long)oh * ysh + (long long)(cv * 4);
    for (int ow = 0; ow < OW; ++ow, yi += (long long)Cv * 4) {
      mp_load_xcol(colC, x, ow + 1, W, xsw, xrow, rvalid);
      float best[4];
      int besti[4];
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        best[v] = -FLT_MAX;
        besti[v] = 0;
      }
#pragma unroll
      for (int rr = 0; rr < 3; ++rr) {
#pragma unroll
        for (int v = 0; v < 4; ++v) {
          if (colA.v[rr][v] > best[v]) { best[v] = colA.v[rr][v]; besti[v] = rr * 3 + 0; }
          if (colB.v[rr][v] > best[v]) { best[v] = colB.v[rr][v]; besti[v] = rr * 3 + 1; }
          if (colC.v[rr][v] > best[v]) { best[v] = colC.v[rr][v]; besti[v] = rr * 3 + 2; }
        }
      }
      st4(y + yi, best);
      uchar4 u;
      u.x = besti[0]; u.y = besti[1]; u.z = besti[2]; u.w = besti[3];
      *reinterpret_cast<uchar4*>(idx + yi) = u;
      colA = colB;
      colB = colC;
    }
  }
}

// ---------------------------------------------------------------------------
// stride-1 rolling-column backward (NHWC, V=4): one thread walks a whole
// input row (b, h, cv).  The <=9 covering windows of consecutive w overlap
// in 6 columns, so instead of re-gathering 9 (idx, dy) pairs per element
// (the generic kernel above), a 3-column register ring shifts left each
// step and loads only the NEW column (3 rows) — 1/3 the gather traffic.
// Fully deterministic (pure gather, fixed order).
// ---------------------------------------------------------------------------

template <typename T>
struct MpCol {
  float g[3][4];       // dy values for the 3 covering rows x 4 channels
  unsigned int id[3];  // packed uchar4 argmax codes per row (0xFF = invalid)
};

template <typename T>
DEVINL void mp_load_col(MpCol<T>& c, const T* __restrict__ dy,
                        const unsigned char* __restrict__ idx, int ow, int OW,
                        long long ysw, const long long yrow[3], const int rvalid[3]) {
#pragma unroll
  for (int rr = 0; rr < 3; ++rr) {
    if (rvalid[rr] && ow >= 0 && ow < OW) {
      const long long a = yrow[rr] + (long long)ow * ysw;
      c.id[rr] = *reinterpret_cast<const unsigned int*>(idx + a);
      ld4(dy + a, c.g[rr]);
    } else {
      c.id[rr] = 0xFFFFFFFFu;
      c.g[rr][0] = c.g[rr][1] = c.g[rr][2] = c.g[rr][3] = 0.f;
    }
  }
}

template <typename T>
__global__ void maxpool3_bwd_s1_row_kernel(const T* __restrict__ dy,
                                           const unsigned char* __restrict__ idx,
                                           T* __restrict__ dx,
                                           int B, int Cv, int H, int W,
                                           int OH, int OW,
                                           long long xsb, long long xsh,
                                           long long ysb, long long ysh) {
  // NHWC: xsw == ysw == C (channel stride 1); addresses below bake in V=4.
  const unsigned int total = (unsigned int)((long long)B * H * Cv);
  for (unsigned int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    const int cv = i % (unsigned int)Cv;
    const unsigned int r = i / (unsigned int)Cv;
    const int h = r % (unsigned int)H;
    const int b = r / (unsigned int)H;
    // covering output rows oh = h-1+rr, rr in 0..2; dh = h-oh+1 = 2-rr
    long long yrow[3];
    int rvalid[3];
#pragma unroll
    for (int rr = 0; rr < 3; ++rr) {
      const int oh = h - 1 + rr;
      rvalid[rr] = (oh >= 0 && oh < OH);
      yrow[rr] = (long long)b * ysb + (long long)oh * ysh + (long long)(cv * 4);
    }
    const long long ysw = (long long)Cv * 4;  // NHWC w-stride = C
    MpCol<T> colA, colB, colC;  // ow = w-1, w, w+1
    mp_load_col(colA, dy, idx, -1, OW, ysw, yrow, rvalid);
    mp_load_col(colB, dy, idx, 0, OW, ysw, yrow, rvalid);
    long long xi = (long long)b * xsb + (long long)h * xsh + (long long)(cv * 4);
    for (int w = 0; w < W; ++w, xi += (long long)Cv * 4) {
      mp_load_col(colC, dy, idx, w + 1, OW, ysw, yrow, rvalid);
      float acc[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int rr = 0; rr < 3; ++rr) {
        // want codes: (dh=2-rr)*3 + dw, dw = 2 (colA), 1 (colB), 0 (colC)
        const unsigned int wantA = (2 - rr) * 3 + 2;
        const unsigned int wantB = (2 - rr) * 3 + 1;
        const unsigned int wantC = (2 - rr) * 3 + 0;
#pragma unroll
        for (int v = 0; v < 4; ++v) {
          const int sh = 8 * v;
          if (((colA.id[rr] >> sh) & 0xFFu) == wantA) acc[v] += colA.g[rr][v];
          if (((colB.id[rr] >> sh) & 0xFFu) == wantB) acc[v] += colB.g[rr][v];
          if (((colC.id[rr] >> sh) & 0xFFu) == wantC) acc[v] += colC.g[rr][v];
        }
      }
      st4(dx + xi, acc);
      colA = colB;
      colB = colC;
    }
  }
}

// ---------------------------------------------------------------------------
// stride-2 rolling-column backward (NHWC, V=4): one thread walks an input
// row.  An input (h, w) is covered by <=2x2 windows: oh in
// [ceil((h-1)/2), floor((h+1)/2)] (fixed per thread), ow in
// [ceil((w-1)/2), floor((w+1)/2)] — the ow window advances every OTHER w
// step, so a 2-column ring loads one new column per TWO inputs (1/4 the
// gather traffic of the per-element kernel).
// ---------------------------------------------------------------------------

template <typename T>
__global__ void maxpool3_bwd_s2_row_kernel(const T* __restrict__ dy,
                                           const unsigned char* __restrict__ idx,
                                           T* __restrict__ dx,
                                           int B, int Cv, int H, int W,
                                           int OH, int OW,
                                           long long xsb, long long xsh,
                                           long long ysb, long long ysh) {
  const unsigned int total = (unsigned int)((long long)B * H * Cv);
  for (unsigned int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    const int cv = i % (unsigned int)Cv;
    const unsigned int r = i / (unsigned int)Cv;
    const int h = r % (unsigned int)H;
    const int b = r / (unsigned int)H;
    // covering output rows: oh0 = ceil((h-1)/2), oh1 = floor((h+1)/2)
    const int oh0 = (h == 0) ? 0 : (h - 1 + 1) / 2;  // ceil((h-1)/2), h>=0
    const int oh1 = (h + 1) / 2;
    long long yrow[2];
    int rvalid[2];
#pragma unroll
    for (int rr = 0; rr < 2; ++rr) {
      const int oh = oh0 + rr;
      rvalid[rr] = (oh <= oh1 && oh >= 0 && oh < OH);
      yrow[rr] = (long long)b * ysb + (long long)oh * ysh + (long long)(cv * 4);
    }
    const long long ysw = (long long)Cv * 4;
    // ring: colP = ow k-1, colQ = ow k with k = floor((w+1)/2)
    MpCol<T> colP, colQ;
    mp_load_col(colP, dy, idx, -1, OW, ysw, yrow, rvalid);  // k-1 at w=0 is -1... k(0)=0
    mp_load_col(colQ, dy, idx, 0, OW, ysw, yrow, rvalid);
    int k = 0;
    long long xi = (long long)b * xsb + (long long)h * xsh + (long long)(cv * 4);
    for (int w = 0; w < W; ++w, xi += (long long)Cv * 4) {
      const int knew = (w + 1) >> 1;  // floor((w+1)/2)
      if (knew != k) {
        colP = colQ;
        mp_load_col(colQ, dy, idx, knew, OW, ysw, yrow, rvalid);
        k = knew;
      }
      // cols covering w: even w -> only ow = k = w/2 with dw = 1;
      // odd w -> ow = k-1 (colP, dw = 2) and ow = k (colQ, dw = 0).
      // Invalid rows carry id bytes 0xFF which match no want code.
      const bool odd = (w & 1) != 0;
      const int dwQ = odd ? 0 : 1;
      float acc[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int rr = 0; rr < 2; ++rr) {
        const int dh = h - ((oh0 + rr) * 2 - 1);
        const unsigned int wantQ = (unsigned int)(dh * 3 + dwQ);
        const unsigned int wantP = (unsigned int)(dh * 3 + 2);
#pragma unroll
        for (int v = 0; v < 4; ++v) {
          const int sh = 8 * v;
          if (((colQ.id[rr] >> sh) & 0xFFu) == wantQ) acc[v] += colQ.g[rr][v];
          if (odd && ((colP.id[rr] >> sh) & 0xFFu) == wantP) acc[v] += colP.g[rr][v];
        }
      }
      st4(dx + xi, acc);
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static bool is_nhwc(const torch::Tensor& t) {
  return t.is_contiguous(at::MemoryFormat::ChannelsLast);
}

static int grid_for(long long total) {
  return (int)std::min<long long>((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 4096);
}

#define VISION_DISPATCH(TENSOR, NAME, ...)                                        \
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, \
      (TENSOR).scalar_type(), NAME, [&] {                                         \
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,          \
                                 __hip_bfloat16, float>;                          \
    TORCH_CHECK((std::is_same_v<scalar_t, at::BFloat16> ||                        \
                 std::is_same_v<scalar_t, float>), NAME ": bf16/fp32 only");      \
    __VA_ARGS__                                                                   \
  })


static bool wave_lrn_ok(bool nhwc, long long C, int64_t size, double k) {
  if (!(nhwc && C % 4 == 0 && C >= 4 && C <= 256 && size <= 5 && k > 0)) return false;
  const long long cq = C / 4;
  return (cq & (cq - 1)) == 0 && 64 % cq == 0;
}

static int ilog2(long long v) {
  int r = 0;
  while ((1LL << r) < v) ++r;
  return r;
}

torch::Tensor lrn_fwd(torch::Tensor x, int64_t size, double alpha, double beta, double k) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4);
  const bool nhwc = is_nhwc(x);
  auto xc = nhwc ? x : x.contiguous();
  auto y = torch::empty_like(xc);
  const long long B = x.size(0), C = x.size(1), S = x.size(2) * x.size(3);
  const long long total = B * C * S;
  const long long cstride = nhwc ? 1 : S;
  auto stream = at::hip::getCurrentHIPStream();
  const float aon = (float)(alpha / size);
  if (wave_lrn_ok(nhwc, C, size, k)) {
    const long long npix = B * S;
    const long long nquads = npix * (C / 4);
    const int blocks = (int)std::min<long long>((nquads + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 8192);
    VISION_DISPATCH(x, "lrn_fwd", {
      lrn_wave_kernel<T, false><<<blocks, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), nullptr,
          reinterpret_cast<T*>(y.data_ptr()), npix, (int)C, ilog2(C / 4),
          (int)size / 2, aon, (float)beta, (float)k);
    });
  } else if (nhwc && C % 4 == 0 && C <= 1024 && size <= 9 && k > 0) {
    const long long npix = B * S;
    const int ppb = NPAIR_BLOCK / (int)(C / 4);
    const int blocks = (int)std::min<long long>((npix + ppb - 1) / ppb, 8192);
    const int half = (int)size / 2;
    VISION_DISPATCH(x, "lrn_fwd", {
      auto launch = [&](auto kfn) {
        kfn<<<blocks, NPAIR_BLOCK, 0, stream>>>(
            reinterpret_cast<const T*>(xc.data_ptr()), nullptr,
            reinterpret_cast<T*>(y.data_ptr()), npix, (int)C,
            (int)size, aon, (float)beta, (float)k);
      };
      switch (half) {
        case 0: launch(lrn_tile4_kernel<T, false, 0>); break;
        case 1: launch(lrn_tile4_kernel<T, false, 1>); break;
        case 2: launch(lrn_tile4_kernel<T, false, 2>); break;
        case 3: launch(lrn_tile4_kernel<T, false, 3>); break;
        default: launch(lrn_tile4_kernel<T, false, 4>); break;
      }
    });
  } else if (C <= NPAIR_BLOCK) {
    const int C_pad = (int)((C + 63) / 64) * 64;
    const long long npix = B * S;
    const int ppb = NPAIR_BLOCK / C_pad;
    const int blocks = (int)std::min<long long>((npix + ppb - 1) / ppb, 8192);
    VISION_DISPATCH(x, "lrn_fwd", {
      lrn_tile_kernel<T, false><<<blocks, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), nullptr,
          reinterpret_cast<T*>(y.data_ptr()), npix, (int)C, C_pad, cstride, 0, S,
          (int)size, aon, (float)beta, (float)k);
    });
  } else {
    VISION_DISPATCH(x, "lrn_fwd", {
      lrn_fwd_generic<T><<<grid_for(total), NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), reinterpret_cast<T*>(y.data_ptr()),
          total, (int)C, cstride, (int)size, aon, (float)beta, (float)k);
    });
  }
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor lrn_bwd(torch::Tensor x, torch::Tensor dy, int64_t size, double alpha,
                      double beta, double k) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4);
  const bool nhwc = is_nhwc(x);
  auto xc = nhwc ? x : x.contiguous();
  auto dyc = nhwc ? (is_nhwc(dy) ? dy : dy.contiguous(at::MemoryFormat::ChannelsLast))
                  : dy.contiguous();
  auto dx = torch::empty_like(xc);
  const long long B = x.size(0), C = x.size(1), S = x.size(2) * x.size(3);
  const long long total = B * C * S;
  const long long cstride = nhwc ? 1 : S;
  auto stream = at::hip::getCurrentHIPStream();
  const float aon = (float)(alpha / size);
  if (wave_lrn_ok(nhwc, C, size, k)) {
    const long long npix = B * S;
    const long long nquads = npix * (C / 4);
    const int blocks = (int)std::min<long long>((nquads + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 8192);
    VISION_DISPATCH(x, "lrn_bwd", {
      lrn_wave_kernel<T, true><<<blocks, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()),
          reinterpret_cast<const T*>(dyc.data_ptr()),
          reinterpret_cast<T*>(dx.data_ptr()), npix, (int)C, ilog2(C / 4),
          (int)size / 2, aon, (float)beta, (float)k);
    });
  } else if (nhwc && C % 4 == 0 && C <= 1024 && size <= 9 && k > 0) {
    const long long npix = B * S;
    const int ppb = NPAIR_BLOCK / (int)(C / 4);
    const int blocks = (int)std::min<long long>((npix + ppb - 1) / ppb, 8192);
    const int half = (int)size / 2;
    VISION_DISPATCH(x, "lrn_bwd", {
      auto launch = [&](auto kfn) {
        kfn<<<blocks, NPAIR_BLOCK, 0, stream>>>(
            reinterpret_cast<const T*>(xc.data_ptr()),
            reinterpret_cast<const T*>(dyc.data_ptr()),
            reinterpret_cast<T*>(dx.data_ptr()), npix, (int)C,
            (int)size, aon, (float)beta, (float)k);
      };
      switch (half) {
        case 0: launch(lrn_tile4_kernel<T, true, 0>); break;
        case 1: launch(lrn_tile4_kernel<T, true, 1>); break;
        case 2: launch(lrn_tile4_kernel<T, true, 2>); break;
        case 3: launch(lrn_tile4_kernel<T, true, 3>); break;
        default: launch(lrn_tile4_kernel<T, true, 4>); break;
      }
    });
  } else if (C <= NPAIR_BLOCK) {
    const int C_pad = (int)((C + 63) / 64) * 64;
    const long long npix = B * S;
    const int ppb = NPAIR_BLOCK / C_pad;
    const int blocks = (int)std::min<long long>((npix + ppb - 1) / ppb, 8192);
    VISION_DISPATCH(x, "lrn_bwd", {
      lrn_tile_kernel<T, true><<<blocks, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()),
          reinterpret_cast<const T*>(dyc.data_ptr()),
          reinterpret_cast<T*>(dx.data_ptr()), npix, (int)C, C_pad, cstride, 0, S,
          (int)size, aon, (float)beta, (float)k);
    });
  } else {
    VISION_DISPATCH(x, "lrn_bwd", {
      lrn_bwd_generic<T><<<grid_for(total), NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()),
          reinterpret_cast<const T*>(dyc.data_ptr()),
          reinterpret_cast<T*>(dx.data_ptr()),
          total, (int)C, cstride, (int)size, aon, (float)beta, (float)k);
    });
  }
  HIP_CHECK_LAST();
  return dx;
}

static void pool_strides(const torch::Tensor& t, bool nhwc, long long* sb,
                         long long* sh, long long* sw, long long* sc) {
  const long long C = t.size(1), H = t.size(2), W = t.size(3);
  if (nhwc) {
    *sb = H * W * C; *sh = W * C; *sw = C; *sc = 1;
  } else {
    *sb = C * H * W; *sh = W; *sw = 1; *sc = H * W;
  }
}

std::vector<torch::Tensor> maxpool3_fwd(torch::Tensor x, int64_t stride, bool ceil_mode) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4);
  TORCH_CHECK(stride == 1 || stride == 2, "maxpool3: stride 1 or 2");
  const bool nhwc = is_nhwc(x);
  auto xc = nhwc ? x : x.contiguous();
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto odim = [&](int I) {
    const int num = I + 2 * 1 - 3;
    int o = (ceil_mode ? (num + (int)stride - 1) / (int)stride : num / (int)stride) + 1;
    if (ceil_mode && (o - 1) * stride >= I + 1) --o;  // torch/caffe clamp
    return o;
  };
  const int OH = odim(H), OW = odim(W);
  auto y = nhwc
      ? torch::empty({B, C, OH, OW}, x.options().memory_format(at::MemoryFormat::ChannelsLast))
      : torch::empty({B, C, OH, OW}, x.options());
  auto idx = nhwc
      ? torch::empty({B, C, OH, OW}, x.options().dtype(torch::kUInt8).memory_format(at::MemoryFormat::ChannelsLast))
      : torch::empty({B, C, OH, OW}, x.options().dtype(torch::kUInt8));
  long long xsb, xsh, xsw, xsc, ysb, ysh, ysw, ysc;
  pool_strides(xc, nhwc, &xsb, &xsh, &xsw, &xsc);
  pool_strides(y, nhwc, &ysb, &ysh, &ysw, &ysc);
  const int V = (nhwc && C % 4 == 0) ? 4 : 1;
  const long long total = (long long)B * OH * OW * (C / V);
  TORCH_CHECK(total < (1LL << 31) && (long long)B * H * W * (C / V) < (1LL << 31),
              "maxpool3: tensor exceeds the 32-bit index fast path");
  auto stream = at::hip::getCurrentHIPStream();
  VISION_DISPATCH(x, "maxpool3_fwd", {
    // note: a rolling-column s1 forward (maxpool3_fwd_s1_row_kernel) was
    // measured SLOWER than the per-element kernel — the serial row walk
    // loses too much parallelism at the 14^2/7^2 inception shapes
    if (V == 4)
      maxpool3_fwd_kernel<T, 4><<<grid_for(total), NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), reinterpret_cast<T*>(y.data_ptr()),
          idx.data_ptr<unsigned char>(), B, C / 4, H, W, OH, OW, (int)stride,
          xsb, xsh, xsw, xsc, ysb, ysh, ysw, ysc);
    else
      maxpool3_fwd_kernel<T, 1><<<grid_for(total), NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), reinterpret_cast<T*>(y.data_ptr()),
          idx.data_ptr<unsigned char>(), B, C, H, W, OH, OW, (int)stride,
          xsb, xsh, xsw, xsc, ysb, ysh, ysw, ysc);
  });
  HIP_CHECK_LAST();
  return {y, idx};
}

torch::Tensor maxpool3_bwd(torch::Tensor dy, torch::Tensor idx, int64_t stride,
                           int64_t H, int64_t W) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4);
  // layout is dictated by idx (saved from forward); coerce dy to match
  const bool nhwc = idx.is_contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = nhwc ? (is_nhwc(dy) ? dy : dy.contiguous(at::MemoryFormat::ChannelsLast))
                  : dy.contiguous();
  const int B = dy.size(0), C = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  auto dx = nhwc
      ? torch::empty({B, C, (int)H, (int)W}, dy.options().memory_format(at::MemoryFormat::ChannelsLast))
      : torch::empty({B, C, (int)H, (int)W}, dy.options());
  long long xsb, xsh, xsw, xsc, ysb, ysh, ysw, ysc;
  pool_strides(dx, nhwc, &xsb, &xsh, &xsw, &xsc);
  pool_strides(dyc, nhwc, &ysb, &ysh, &ysw, &ysc);
  const int V = (nhwc && C % 4 == 0) ? 4 : 1;
  const long long total = (long long)B * H * W * (C / V);
  TORCH_CHECK(total < (1LL << 31), "maxpool3: tensor exceeds the 32-bit index fast path");
  auto stream = at::hip::getCurrentHIPStream();
  VISION_DISPATCH(dy, "maxpool3_bwd", {
    if (V == 4 && stride == 1)
      // rolling-column row walker: 1/3 the gather traffic of the generic
      // per-element kernel (each thread owns a (b, h, cv) input row)
      maxpool3_bwd_s1_row_kernel<T><<<grid_for((long long)B * H * (C / 4)),
                                      NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(dyc.data_ptr()), idx.data_ptr<unsigned char>(),
          reinterpret_cast<T*>(dx.data_ptr()), B, C / 4, (int)H, (int)W, OH, OW,
          xsb, xsh, ysb, ysh);
    else if (V == 4)
      maxpool3_bwd_kernel<T, 4><<<grid_for(total), NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(dyc.data_ptr()), idx.data_ptr<unsigned char>(),
          reinterpret_cast<T*>(dx.data_ptr()), B, C / 4, (int)H, (int)W, OH, OW, (int)stride,
          xsb, xsh, xsw, xsc, ysb, ysh, ysw, ysc);
    else
      maxpool3_bwd_kernel<T, 1><<<grid_for(total), NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(dyc.data_ptr()), idx.data_ptr<unsigned char>(),
          reinterpret_cast<T*>(dx.data_ptr()), B, C, (int)H, (int)W, OH, OW, (int)stride,
          xsb, xsh, xsw, xsc, ysb, ysh, ysw, ysc);
  });
  HIP_CHECK_LAST();
  return dx;
}
