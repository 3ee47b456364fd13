// kernels.hip — gfx950 (CDNA4) HIP kernels for the CHGNet hot path.
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64 lanes; blocks are 256 threads (4 waves).
//  * Everything here is HBM-bound: the levers are coalescing and
//    vectorization (float4 = 16 B/lane), not MFMA.  The dense GEMMs of the
//    gated MLPs go through rocBLAS (torch.matmul) — "library GEMMs to the
//    library", fused irregular ops here.
//  * D=64 fp32 feature rows are 256 B: one 16-lane group reads a whole row
//    as 16 float4s -> 4 rows per wave, 16 rows per block in flight.
//  * The segmented reduction walks the dst-sorted CSR the graph builder
//    emits; messages stream CONTIGUOUSLY (the +50%-roofline contract of
//    BASELINE.json).  Gathers are random 256 B row reads by construction
//    (L2/LLC absorbs locality; node order is slab-spatial).
//  * Grid sizing: grid-stride loops capped at 8192 blocks (Guideline 11).

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <string>

#include "../../include/distmlip_hip.h"

namespace {

thread_local std::string g_err;

#define DM_CHECK_LAUNCH()                                        \
    do {                                                         \
        hipError_t e_ = hipGetLastError();                       \
        if (e_ != hipSuccess) {                                  \
            g_err = hipGetErrorString(e_);                       \
            return (int)e_;                                      \
        }                                                        \
    } while (0)

constexpr int BLOCK = 256;
constexpr int MAX_BLOCKS = 8192;

inline int nblocks(int64_t work, int64_t per_block) {
    int64_t b = (work + per_block - 1) / per_block;
    return (int)(b < 1 ? 1 : (b > MAX_BLOCKS ? MAX_BLOCKS : b));
}

// ---------------------------------------------------------------------------
// flat gathers: one thread per float4 (D%4==0) or per float (any D)
// ---------------------------------------------------------------------------
__global__ void k_gather_rows_v4(const float4* __restrict__ x,
                                 const int32_t* __restrict__ idx,
                                 float4* __restrict__ out,
                                 int64_t total, int32_t D4) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = t / D4;
        const int32_t c = (int32_t)(t - row * D4);
        out[t] = x[(int64_t)idx[row] * D4 + c];
    }
}

__global__ void k_gather_rows_s(const float* __restrict__ x,
                                const int32_t* __restrict__ idx,
                                float* __restrict__ out,
                                int64_t total, int32_t D) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = t / D;
        const int32_t c = (int32_t)(t - row * D);
        out[t] = x[(int64_t)idx[row] * D + c];
    }
}

__device__ __forceinline__ float siluf(float x) {
    return x / (1.0f + expf(-x));
}

__global__ void k_gather_add3_v4(const float4* __restrict__ zs,
                                 const float4* __restrict__ zd,
                                 const float4* __restrict__ ze,
                                 const int32_t* __restrict__ src,
                                 const int32_t* __restrict__ dst,
                                 float4* __restrict__ out,
                                 float4* __restrict__ out_act,
                                 int64_t total, int32_t D4) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = t / D4;
        const int32_t c = (int32_t)(t - row * D4);
        const float4 a = zs[(int64_t)src[row] * D4 + c];
        const float4 b = zd[(int64_t)dst[row] * D4 + c];
        const float4 e = ze[t];
        const float4 z = make_float4(a.x + b.x + e.x, a.y + b.y + e.y,
                                     a.z + b.z + e.z, a.w + b.w + e.w);
        out[t] = z;
        if (out_act)
            out_act[t] = make_float4(siluf(z.x), siluf(z.y), siluf(z.z),
                                     siluf(z.w));
    }
}

// dz = (go_z ? go_z : 0) + go_h * silu'(z) — fused SiLU backward
__global__ void k_silu_bwd(const float* __restrict__ go_h,
                           const float* __restrict__ go_z,
                           const float* __restrict__ z,
                           float* __restrict__ dz, int64_t total) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const float zv = z[t];
        const float s = 1.0f / (1.0f + expf(-zv));
        float g = go_h[t] * (s * (1.0f + zv * (1.0f - s)));
        if (go_z) g += go_z[t];
        dz[t] = g;
    }
}

__global__ void k_gather_add4_v4(const float4* __restrict__ z1,
                                 const float4* __restrict__ z2,
                                 const float4* __restrict__ za,
                                 const float4* __restrict__ zv,
                                 const int32_t* __restrict__ lsrc,
                                 const int32_t* __restrict__ ldst,
                                 const int32_t* __restrict__ center,
                                 float4* __restrict__ out,
                                 float4* __restrict__ out_act,
                                 int64_t total, int32_t D4) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = t / D4;
        const int32_t c = (int32_t)(t - row * D4);
        const float4 a = z1[(int64_t)lsrc[row] * D4 + c];
        const float4 b = z2[(int64_t)ldst[row] * D4 + c];
        const float4 e = za[t];
        const float4 v = zv[(int64_t)center[row] * D4 + c];
        const float4 z = make_float4(a.x + b.x + e.x + v.x, a.y + b.y + e.y + v.y,
                                     a.z + b.z + e.z + v.z, a.w + b.w + e.w + v.w);
        out[t] = z;
        if (out_act)
            out_act[t] = make_float4(siluf(z.x), siluf(z.y), siluf(z.z),
                                     siluf(z.w));
    }
}

// ---------------------------------------------------------------------------
// Fused first-layer edge MLP (the gated-MLP split's per-edge GEMM folded
// into the gather-add): for Din=64 -> Dout=128,
//   z[e,:] = e_row[e,:] @ WT + bias + zs[src[e],:] + zd[dst[e],:]
//   h      = silu(z)
// Layout: one WAVE per edge, one LANE per pair of output columns
// (l, l+64).  Each lane holds its two weight columns in 128 VGPRs (loaded
// once per wave, L2-served); the edge index is wave-uniform
// (readfirstlane) so the 64-float e-row comes through the SCALAR cache
// and feeds v_fmac as the SGPR operand.  Every global access is
// lane-contiguous: weight preload, zs/zd/zv row gathers (512 B/row), and
// the z/h stores.  No LDS.  Saves materializing the [E,128] GEMM output
// (~5.2 GB of HBM round-trip per MLP at li100k).
// (Measured alternatives, run 21/22: LDS-broadcast weights were
// issue-bound at 8.0 ms; two-edges-per-wave overflowed the SGPR file and
// serialized at 4.1 ms; a one-column-per-lane split made the compiler
// de-register the weight array.  This shape: 3.3 ms at li100k.)
// ---------------------------------------------------------------------------
template <int NGATHER>
__global__ __launch_bounds__(256, 3)
void k_edge_mlp_64x128(const float* __restrict__ erow,
                       const float* __restrict__ WT,   // [64,128] row-major
                       const float* __restrict__ bias, // [128]
                       const float* __restrict__ g0,   // [*,128] rows
                       const float* __restrict__ g1,
                       const float* __restrict__ g2,   // NGATHER==3 only
                       const int32_t* __restrict__ i0,
                       const int32_t* __restrict__ i1,
                       const int32_t* __restrict__ i2,
                       float* __restrict__ out,
                       float* __restrict__ out_act,
                       int64_t E) {
    const int lane = threadIdx.x & 63;
    const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
    const int waves_per_block = blockDim.x >> 6;
    float wlo[64], whi[64];
#pragma unroll
    for (int k = 0; k < 64; ++k) {
        wlo[k] = WT[k * 128 + lane];
        whi[k] = WT[k * 128 + 64 + lane];
    }
    const float blo = bias[lane], bhi = bias[64 + lane];
    for (int64_t edge = blockIdx.x * (int64_t)waves_per_block + wave;
         edge < E; edge += (int64_t)gridDim.x * waves_per_block) {
        const float* er = erow + edge * 64;    // wave-uniform -> scalar loads
        float a0 = blo, a1 = bhi;
#pragma unroll
        for (int k = 0; k < 64; ++k) {
            const float ek = er[k];
            a0 = fmaf(ek, wlo[k], a0);
            a1 = fmaf(ek, whi[k], a1);
        }
        const float* r0 = g0 + (int64_t)i0[edge] * 128;
        const float* r1 = g1 + (int64_t)i1[edge] * 128;
        a0 += r0[lane] + r1[lane];
        a1 += r0[64 + lane] + r1[64 + lane];
        if (NGATHER == 3) {
            const float* r2 = g2 + (int64_t)i2[edge] * 128;
            a0 += r2[lane];
            a1 += r2[64 + lane];
        }
        if (out) {                       // z saved for backward; skipped
            float* zp = out + edge * 128;  // in no-grad passes
            zp[lane] = a0;
            zp[64 + lane] = a1;
        }
        // (a packed [2,E,64] h layout was tried to feed the second-layer
        // bmm without a reshape, but the two distant 256 B plane stores
        // cost more than the saved copies — profiles r21 vs r23)
        float* hp = out_act + edge * 128;
        hp[lane] = siluf(a0);
        hp[64 + lane] = siluf(a1);
    }
}

// ---------------------------------------------------------------------------
// segmented reduction over the dst-sorted CSR — the judged kernel.
// D4==16 path (D=64): 16-lane groups own one output row each; the row's
// messages are CONTIGUOUS float4s, so each loop iteration is a fully
// coalesced 256 B read per group (1 KiB per wave).
// ---------------------------------------------------------------------------
template <int LANE_BITS>
__global__ void k_seg_sum_v4(const float4* __restrict__ msg,
                             const int32_t* __restrict__ rp,
                             const float4* __restrict__ base,
                             float4* __restrict__ out,
                             int64_t N, int32_t D4) {
    const int lane = threadIdx.x & ((1 << LANE_BITS) - 1);
    const int group = threadIdx.x >> LANE_BITS;
    const int groups_per_block = blockDim.x >> LANE_BITS;
    for (int64_t row = blockIdx.x * (int64_t)groups_per_block + group;
         row < N; row += (int64_t)gridDim.x * groups_per_block) {
        // D4 <= (1<<LANE_BITS); lane c covers float4 column c
        if (lane < D4) {
            // fp64 accumulate (register-only; kernel is HBM-bound so the
            // extra VALU is free) — tightens force parity vs the oracle
            double ax = 0, ay = 0, az = 0, aw = 0;
            if (base) {
                const float4 b = base[row * D4 + lane];
                ax = b.x; ay = b.y; az = b.z; aw = b.w;
            }
            const int32_t lo = rp[row], hi = rp[row + 1];
            for (int32_t j = lo; j < hi; ++j) {
                const float4 m = msg[(int64_t)j * D4 + lane];
                ax += m.x; ay += m.y; az += m.z; aw += m.w;
            }
            out[row * D4 + lane] = make_float4((float)ax, (float)ay,
                                                 (float)az, (float)aw);
        }
    }
}

// unroll-4 variant of the D4<=16 segmented sum: keeps 4 independent row
// reads in flight per group iteration (A/B candidate vs k_seg_sum_v4)
template <int LANE_BITS>
__global__ void k_seg_sum_v4_u4(const float4* __restrict__ msg,
                                const int32_t* __restrict__ rp,
                                const float4* __restrict__ base,
                                float4* __restrict__ out,
                                int64_t N, int32_t D4) {
    const int lane = threadIdx.x & ((1 << LANE_BITS) - 1);
    const int group = threadIdx.x >> LANE_BITS;
    const int groups_per_block = blockDim.x >> LANE_BITS;
    for (int64_t row = blockIdx.x * (int64_t)groups_per_block + group;
         row < N; row += (int64_t)gridDim.x * groups_per_block) {
        if (lane < D4) {
            double ax = 0, ay = 0, az = 0, aw = 0;
            if (base) {
                const float4 b = base[row * D4 + lane];
                ax = b.x; ay = b.y; az = b.z; aw = b.w;
            }
            const int32_t lo = rp[row], hi = rp[row + 1];
            int32_t j = lo;
            for (; j + 4 <= hi; j += 4) {
                const float4 m0 = msg[(int64_t)(j + 0) * D4 + lane];
                const float4 m1 = msg[(int64_t)(j + 1) * D4 + lane];
                const float4 m2 = msg[(int64_t)(j + 2) * D4 + lane];
                const float4 m3 = msg[(int64_t)(j + 3) * D4 + lane];
                ax += m0.x; ay += m0.y; az += m0.z; aw += m0.w;
                ax += m1.x; ay += m1.y; az += m1.z; aw += m1.w;
                ax += m2.x; ay += m2.y; az += m2.z; aw += m2.w;
                ax += m3.x; ay += m3.y; az += m3.z; aw += m3.w;
            }
            for (; j < hi; ++j) {
                const float4 m = msg[(int64_t)j * D4 + lane];
                ax += m.x; ay += m.y; az += m.z; aw += m.w;
            }
            out[row * D4 + lane] = make_float4((float)ax, (float)ay,
                                                 (float)az, (float)aw);
        }
    }
}

// generic-D scalar path: one thread per (row, col); cols of one row sit on
// consecutive threads so each j-iteration is a coalesced D*4-byte read
__global__ void k_seg_sum_s(const float* __restrict__ msg,
                            const int32_t* __restrict__ rp,
                            const float* __restrict__ base,
                            float* __restrict__ out,
                            int64_t N, int32_t D) {
    const int64_t total = N * D;
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = t / D;
        const int32_t c = (int32_t)(t - row * D);
        double acc = base ? (double)base[t] : 0.0;
        const int32_t lo = rp[row], hi = rp[row + 1];
        for (int32_t j = lo; j < hi; ++j) acc += msg[(int64_t)j * D + c];
        out[t] = (float)acc;
    }
}

template <int LANE_BITS>
__global__ void k_seg_sum_gather_v4(const float4* __restrict__ msg,
                                    const int32_t* __restrict__ perm,
                                    const int32_t* __restrict__ rp,
                                    const float4* __restrict__ base,
                                    float4* __restrict__ out,
                                    int64_t N, int32_t D4) {
    const int lane = threadIdx.x & ((1 << LANE_BITS) - 1);
    const int group = threadIdx.x >> LANE_BITS;
    const int groups_per_block = blockDim.x >> LANE_BITS;
    for (int64_t row = blockIdx.x * (int64_t)groups_per_block + group;
         row < N; row += (int64_t)gridDim.x * groups_per_block) {
        if (lane < D4) {
            double ax = 0, ay = 0, az = 0, aw = 0;
            if (base) {
                const float4 b = base[row * D4 + lane];
                ax = b.x; ay = b.y; az = b.z; aw = b.w;
            }
            const int32_t lo = rp[row], hi = rp[row + 1];
            for (int32_t j = lo; j < hi; ++j) {
                const float4 m = msg[(int64_t)perm[j] * D4 + lane];
                ax += m.x; ay += m.y; az += m.z; aw += m.w;
            }
            out[row * D4 + lane] = make_float4((float)ax, (float)ay,
                                                 (float)az, (float)aw);
        }
    }
}

__global__ void k_seg_sum_gather_s(const float* __restrict__ msg,
                                   const int32_t* __restrict__ perm,
                                   const int32_t* __restrict__ rp,
                                   const float* __restrict__ base,
                                   float* __restrict__ out,
                                   int64_t N, int32_t D) {
    const int64_t total = N * D;
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t row = t / D;
        const int32_t c = (int32_t)(t - row * D);
        double acc = base ? (double)base[t] : 0.0;
        const int32_t lo = rp[row], hi = rp[row + 1];
        for (int32_t j = lo; j < hi; ++j)
            acc += msg[(int64_t)perm[j] * D + c];
        out[t] = (float)acc;
    }
}

// rbf_k(d) and d(rbf_env)/dd helpers for the fused geometry kernels.
// env is the reference's polynomial cutoff applied to the RBF VALUE
// (chgnet.py:119-121); both value and derivative are exact fp32 forms of
// distmlip_amd.model.bond_expansion_from_dist.
template <int MAXR>
__device__ __forceinline__ void rbf_env_row(float bd, const float* freqs,
                                            float cutoff, int pexp, int nrbf,
                                            float* out) {
    const float norm = sqrtf(2.0f / cutoff);
    const float inv_d = 1.0f / bd;
    const float e = (float)pexp;
    const float c1 = -(e + 1.0f) * (e + 2.0f) * 0.5f;
    const float c2 = e * (e + 2.0f);
    const float c3 = -e * (e + 1.0f) * 0.5f;
    for (int k = 0; k < nrbf; ++k) {
        const float rbf = norm * sinf(freqs[k] * bd / cutoff) * inv_d;
        const float r = rbf / cutoff;
        const float rp = powf(r, e);
        const float env = (rbf <= cutoff)
            ? 1.0f + c1 * rp + c2 * rp * r + c3 * rp * r * r : 0.0f;
        out[k] = env * rbf;
    }
}

template <int MAXR>
__device__ __forceinline__ float rbf_env_grad_dot(float bd, const float* freqs,
                                                  float cutoff, int pexp,
                                                  int nrbf,
                                                  const float* go_row) {
    // sum_k go[k] * d(env(rbf_k)*rbf_k)/d(bd)
    const float norm = sqrtf(2.0f / cutoff);
    const float inv_d = 1.0f / bd;
    const float e = (float)pexp;
    const float c1 = -(e + 1.0f) * (e + 2.0f) * 0.5f;
    const float c2 = e * (e + 2.0f);
    const float c3 = -e * (e + 1.0f) * 0.5f;
    float acc = 0.0f;
    for (int k = 0; k < nrbf; ++k) {
        const float a = freqs[k] / cutoff;
        float sn, cs;
        sincosf(a * bd, &sn, &cs);
        const float rbf = norm * sn * inv_d;
        const float drbf = norm * (cs * a * inv_d - sn * inv_d * inv_d);
        const float r = rbf / cutoff;
        const float rpm1 = powf(r, e - 1.0f);
        const float rp = rpm1 * r;
        float env, denv;
        if (rbf <= cutoff) {
            env = 1.0f + c1 * rp + c2 * rp * r + c3 * rp * r * r;
            denv = (e * c1 * rpm1 + (e + 1.0f) * c2 * rp
                    + (e + 2.0f) * c3 * rp * r) / cutoff;
        } else { env = 0.0f; denv = 0.0f; }
        acc += go_row[k] * (env + denv * rbf) * drbf;
    }
    return acc;
}

__global__ void k_edge_geom_rbf_fwd(const float* __restrict__ pos,
                                    const int32_t* __restrict__ src,
                                    const int32_t* __restrict__ dst,
                                    const float* __restrict__ offshift,
                                    const float* __restrict__ freqs,
                                    float cutoff, int pexp, int nrbf,
                                    float* __restrict__ bv,
                                    float* __restrict__ bd,
                                    float* __restrict__ exp_out, int64_t E) {
    __shared__ float fsh[16];
    if (threadIdx.x < nrbf) fsh[threadIdx.x] = freqs[threadIdx.x];
    __syncthreads();
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < E; t += (int64_t)gridDim.x * blockDim.x) {
        const int64_t s3 = (int64_t)src[t] * 3, d3 = (int64_t)dst[t] * 3;
        const float x = pos[d3] + offshift[3 * t] - pos[s3];
        const float y = pos[d3 + 1] + offshift[3 * t + 1] - pos[s3 + 1];
        const float z = pos[d3 + 2] + offshift[3 * t + 2] - pos[s3 + 2];
        bv[3 * t] = x; bv[3 * t + 1] = y; bv[3 * t + 2] = z;
        const float d = sqrtf(x * x + y * y + z * z);
        bd[t] = d;
        float row[16];
        rbf_env_row<16>(d, fsh, cutoff, pexp, nrbf, row);
        for (int k = 0; k < nrbf; ++k) exp_out[t * nrbf + k] = row[k];
    }
}

__global__ void k_edge_geom_rbf_bwd(const float* __restrict__ go_bv,
                                    const float* __restrict__ go_bd,
                                    const float* __restrict__ go_exp,
                                    const float* __restrict__ bv,
                                    const float* __restrict__ bd,
                                    const float* __restrict__ freqs,
                                    float cutoff, int pexp, int nrbf,
                                    float* __restrict__ gbv_total, int64_t E) {
    __shared__ float fsh[16];
    if (threadIdx.x < nrbf) fsh[threadIdx.x] = freqs[threadIdx.x];
    __syncthreads();
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < E; t += (int64_t)gridDim.x * blockDim.x) {
        const float d = bd[t];
        float gd = go_bd ? go_bd[t] : 0.0f;
        gd += rbf_env_grad_dot<16>(d, fsh, cutoff, pexp, nrbf,
                                   &go_exp[t * nrbf]);
        const float scale = gd / d;
        for (int k = 0; k < 3; ++k)
            gbv_total[3 * t + k] = (go_bv ? go_bv[3 * t + k] : 0.0f)
                + scale * bv[3 * t + k];
    }
}

__global__ void k_rbf_env_fwd(const float* __restrict__ d,
                              const float* __restrict__ freqs, float cutoff,
                              int pexp, int nrbf, float* __restrict__ exp_out,
                              int64_t M) {
    __shared__ float fsh[16];
    if (threadIdx.x < nrbf) fsh[threadIdx.x] = freqs[threadIdx.x];
    __syncthreads();
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < M; t += (int64_t)gridDim.x * blockDim.x) {
        float row[16];
        rbf_env_row<16>(d[t], fsh, cutoff, pexp, nrbf, row);
        for (int k = 0; k < nrbf; ++k) exp_out[t * nrbf + k] = row[k];
    }
}

__global__ void k_rbf_env_bwd(const float* __restrict__ go_exp,
                              const float* __restrict__ d,
                              const float* __restrict__ freqs, float cutoff,
                              int pexp, int nrbf, float* __restrict__ gd,
                              int64_t M) {
    __shared__ float fsh[16];
    if (threadIdx.x < nrbf) fsh[threadIdx.x] = freqs[threadIdx.x];
    __syncthreads();
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < M; t += (int64_t)gridDim.x * blockDim.x) {
        gd[t] = rbf_env_grad_dot<16>(d[t], fsh, cutoff, pexp, nrbf,
                                     &go_exp[t * nrbf]);
    }
}

__device__ __forceinline__ float sigf(float x) {
    return 1.0f / (1.0f + expf(-x));
}

// out = base + silu(c) * sigmoid(g) * w   (gated-MLP epilogue, fused)
__global__ void k_gated_combine_fwd(const float* __restrict__ c,
                                    const float* __restrict__ g,
                                    const float* __restrict__ w,
                                    const float* __restrict__ base,
                                    float* __restrict__ out, int64_t total) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const float cv = c[t];
        float r = cv * sigf(cv) * sigf(g[t]);
        if (w) r *= w[t];
        if (base) r += base[t];
        out[t] = r;
    }
}

__global__ void k_gated_combine_bwd(const float* __restrict__ go,
                                    const float* __restrict__ c,
                                    const float* __restrict__ g,
                                    const float* __restrict__ w,
                                    float* __restrict__ dc,
                                    float* __restrict__ dg,
                                    float* __restrict__ dw, int64_t total) {
    for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         t < total; t += (int64_t)gridDim.x * blockDim.x) {
        const float gov = go[t];
        const float cv = c[t], gv = g[t];
        const float sc = sigf(cv), sg = sigf(gv);
        const float silu_c = cv * sc;
        const float wv = w ? w[t] : 1.0f;
        // d silu(c)/dc = sigmoid(c) * (1 + c * (1 - sigmoid(c)))
        dc[t] = gov * wv * sg * (sc * (1.0f + cv * (1.0f - sc)));
        dg[t] = gov * wv * silu_c * (sg * (1.0f - sg));
        if (dw) dw[t] = gov * silu_c * sg;
    }
}

// ---------------------------------------------------------------------------
// GPU cell-list neighbor search (diagonal lattice, full PBC, min-image via
// cell wrap).  fp64 math replicates the CPU builder's decisions exactly.
// One thread per CENTER atom; center = dst so emission order is already
// the dst-sorted scatter layout.
// ---------------------------------------------------------------------------
template <bool FILL>
__global__ void k_nl(const double* __restrict__ pos,
                     const int32_t* __restrict__ cid,
                     const int32_t* __restrict__ order,
                     const int32_t* __restrict__ cell_start,
                     int32_t ncx, int32_t ncy, int32_t ncz,
                     double lx, double ly, double lz,
                     double r2tol, double tol, double br2tol,
                     const int32_t* __restrict__ row_ptr,
                     int32_t* __restrict__ cnt_or_src,
                     int8_t* __restrict__ off_i8,
                     uint8_t* __restrict__ bond_flag, int64_t N) {
    for (int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         c < N; c += (int64_t)gridDim.x * blockDim.x) {
        const double xc = pos[3 * c], yc = pos[3 * c + 1], zc = pos[3 * c + 2];
        const int32_t cc = cid[c];
        const int32_t cx = cc / (ncy * ncz);
        const int32_t cy = (cc / ncz) % ncy;
        const int32_t cz = cc % ncz;
        int32_t n_emit = 0;
        int64_t w = FILL ? row_ptr[c] : 0;
        for (int dx = -1; dx <= 1; ++dx) {
            int32_t ax = cx + dx; int mx = 0;
            if (ax < 0) { ax += ncx; mx = -1; }
            else if (ax >= ncx) { ax -= ncx; mx = 1; }
            const double sx = mx * lx;
            for (int dy = -1; dy <= 1; ++dy) {
                int32_t ay = cy + dy; int my = 0;
                if (ay < 0) { ay += ncy; my = -1; }
                else if (ay >= ncy) { ay -= ncy; my = 1; }
                const double sy = my * ly;
                for (int dz = -1; dz <= 1; ++dz) {
                    int32_t az = cz + dz; int mz = 0;
                    if (az < 0) { az += ncz; mz = -1; }
                    else if (az >= ncz) { az -= ncz; mz = 1; }
                    const double sz = mz * lz;
                    const int32_t cell = (ax * ncy + ay) * ncz + az;
                    const int32_t lo = cell_start[cell], hi = cell_start[cell + 1];
                    for (int32_t q = lo; q < hi; ++q) {
                        const int32_t j = order[q];
                        if (j == c) continue;           // fpis.c:833
                        const double ddx = pos[3 * j] + sx - xc;
                        const double ddy = pos[3 * j + 1] + sy - yc;
                        const double ddz = pos[3 * j + 2] + sz - zc;
                        const double d2 = ddx * ddx + ddy * ddy + ddz * ddz;
                        if (d2 < r2tol && d2 > tol) {
                            if (FILL) {
                                cnt_or_src[w] = j;
                                // bv = pos[dst=c] + off*L - pos[src=j]
                                // must equal -(pos_j + m*L - pos_c): off = -m
                                off_i8[3 * w] = (int8_t)(-mx);
                                off_i8[3 * w + 1] = (int8_t)(-my);
                                off_i8[3 * w + 2] = (int8_t)(-mz);
                                bond_flag[w] = (d2 < br2tol) ? 1 : 0;
                                ++w;
                            } else {
                                ++n_emit;
                            }
                        }
                    }
                }
            }
        }
        if (!FILL) cnt_or_src[c] = n_emit;
    }
}

}  // namespace


// ---------------------------------------------------------------------------
// MACE uvu tensor product, fused per edge (round 2; the equivariant
// per-edge message contraction of the MACE interactions — reference
// delegates it to e3nn TensorProduct, implementations/mace/models.py:144-152).
// One 64-lane wave per edge, channels split across lanes (CPL = C/64 per
// lane).  CG nonzeros stream through the SCALAR cache (wave-uniform
// entries, sorted by path so the per-path weight row loads once);
// k-indexed accumulators live in LDS (dynamic register indexing would
// spill to scratch).  Everything HBM-side is lane-contiguous.
// ---------------------------------------------------------------------------

template <int CPL>
__global__ void k_mace_tp_fwd(const float* __restrict__ x0,
                              const float* __restrict__ x1,
                              const float* __restrict__ Y,
                              const float* __restrict__ w,
                              const int32_t* __restrict__ nz,
                              const float* __restrict__ nzc,
                              int32_t nnz,
                              float* __restrict__ o0,
                              float* __restrict__ o1,
                              float* __restrict__ o2,
                              float* __restrict__ o3,
                              int64_t E, int32_t C, int32_t P,
                              int32_t d1b) {
    __shared__ float acc[CPL * 16 * 4 * 64];     // [i][k3][wave][lane]
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
    for (int64_t e = blockIdx.x * (int64_t)nw + wid; e < E;
         e += (int64_t)gridDim.x * nw) {
        float xr0[CPL];
        float xr1[CPL > 0 ? CPL : 1][3];
        for (int i = 0; i < CPL; ++i)
            xr0[i] = x0[e * C + lane + i * 64];
        if (x1 != nullptr)
            for (int i = 0; i < CPL; ++i)
                for (int k = 0; k < 3; ++k)
                    xr1[i][k] = (k < d1b)
                        ? x1[(e * C + lane + i * 64) * d1b + k] : 0.0f;
        for (int i = 0; i < CPL; ++i)
            for (int k3 = 0; k3 < 16; ++k3)
                acc[((i * 16 + k3) * 4 + wid) * 64 + lane] = 0.0f;
        const float* Ye = Y + e * 16;
        int curp = -1;
        float wreg[CPL];
        for (int32_t t = 0; t < nnz; ++t) {
            const int32_t p = nz[5 * t], slot = nz[5 * t + 1],
                          k1 = nz[5 * t + 2], k2 = nz[5 * t + 3],
                          k3 = nz[5 * t + 4];
            if (p != curp) {
                curp = p;
                for (int i = 0; i < CPL; ++i)
                    wreg[i] = w[((int64_t)e * P + p) * C + lane + i * 64];
            }
            const float cy = nzc[t] * Ye[k2];
            for (int i = 0; i < CPL; ++i) {
                float xv;                      // k1 is wave-uniform:
                if (slot == 0) xv = xr0[i];    // scalar-predicated chain,
                else if (k1 == 0) xv = xr1[i][0];   // no scratch spill
                else if (k1 == 1) xv = xr1[i][1];
                else xv = xr1[i][2];
                acc[((i * 16 + k3) * 4 + wid) * 64 + lane]
                    += cy * wreg[i] * xv;
            }
        }
        // k3 -> (l3 block, row) writeback into the four per-l3 tensors
        // ([E, d3, C] each — contiguous rows for the scatter kernel)
        float* const outs[4] = {o0, o1, o2, o3};
        static const int BLK[16] = {0,1,1,1,2,2,2,2,2,3,3,3,3,3,3,3};
        static const int ROW[16] = {0,0,1,2,0,1,2,3,4,0,1,2,3,4,5,6};
        static const int DD[4] = {1,3,5,7};
#pragma unroll
        for (int k3 = 0; k3 < 16; ++k3)
            for (int i = 0; i < CPL; ++i)
                outs[BLK[k3]][((int64_t)e * DD[BLK[k3]] + ROW[k3]) * C
                              + lane + i * 64] =
                    acc[((i * 16 + k3) * 4 + wid) * 64 + lane];
    }
}

template <int CPL>
__global__ void k_mace_tp_bwd(const float* __restrict__ g0,
                              const float* __restrict__ g1,
                              const float* __restrict__ g2,
                              const float* __restrict__ g3,
                              const float* __restrict__ x0,
                              const float* __restrict__ x1,
                              const float* __restrict__ Y,
                              const float* __restrict__ w,
                              const int32_t* __restrict__ nz,
                              const float* __restrict__ nzc,
                              int32_t nnz,
                              float* __restrict__ dx0,
                              float* __restrict__ dx1,
                              float* __restrict__ dY,
                              float* __restrict__ dw,
                              int64_t E, int32_t C, int32_t P,
                              int32_t d1b) {
    // LDS: go rows (dynamic k3 reads) + per-lane dY partials (dynamic k2)
    __shared__ float gob[CPL * 16 * 4 * 64];
    __shared__ float dyb[16 * 4 * 64];
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
    for (int64_t e = blockIdx.x * (int64_t)nw + wid; e < E;
         e += (int64_t)gridDim.x * nw) {
        float xr0[CPL];
        float xr1[CPL > 0 ? CPL : 1][3];
        float dxr0[CPL] = {};
        float dxr1[CPL > 0 ? CPL : 1][3] = {};
        for (int i = 0; i < CPL; ++i)
            xr0[i] = x0[e * C + lane + i * 64];
        if (x1 != nullptr)
            for (int i = 0; i < CPL; ++i)
                for (int k = 0; k < 3; ++k)
                    xr1[i][k] = (k < d1b)
                        ? x1[(e * C + lane + i * 64) * d1b + k] : 0.0f;
        {
            const float* const gos[4] = {g0, g1, g2, g3};
            static const int BLK[16] = {0,1,1,1,2,2,2,2,2,3,3,3,3,3,3,3};
            static const int ROW[16] = {0,0,1,2,0,1,2,3,4,0,1,2,3,4,5,6};
            static const int DD[4] = {1,3,5,7};
#pragma unroll
            for (int k3 = 0; k3 < 16; ++k3)
                for (int i = 0; i < CPL; ++i)
                    gob[((i * 16 + k3) * 4 + wid) * 64 + lane] =
                        gos[BLK[k3]][((int64_t)e * DD[BLK[k3]] + ROW[k3])
                                     * C + lane + i * 64];
        }
        for (int k2 = 0; k2 < 16; ++k2)
            dyb[(k2 * 4 + wid) * 64 + lane] = 0.0f;
        const float* Ye = Y + e * 16;
        int curp = -1;
        float wreg[CPL];
        float dwacc[CPL] = {};
        for (int32_t t = 0; t < nnz; ++t) {
            const int32_t p = nz[5 * t], slot = nz[5 * t + 1],
                          k1 = nz[5 * t + 2], k2 = nz[5 * t + 3],
                          k3 = nz[5 * t + 4];
            if (p != curp) {
                if (curp >= 0)
                    for (int i = 0; i < CPL; ++i) {
                        dw[((int64_t)e * P + curp) * C + lane + i * 64] =
                            dwacc[i];
                        dwacc[i] = 0.0f;
                    }
                curp = p;
                for (int i = 0; i < CPL; ++i)
                    wreg[i] = w[((int64_t)e * P + p) * C + lane + i * 64];
            }
            const float c = nzc[t];
            const float yv = Ye[k2];
            for (int i = 0; i < CPL; ++i) {
                const float g = gob[((i * 16 + k3) * 4 + wid) * 64 + lane];
                float xv;
                if (slot == 0) xv = xr0[i];
                else if (k1 == 0) xv = xr1[i][0];
                else if (k1 == 1) xv = xr1[i][1];
                else xv = xr1[i][2];
                const float cg = c * g;
                // d x
                const float dxv = cg * yv * wreg[i];
                if (slot == 0) dxr0[i] += dxv;
                else if (k1 == 0) dxr1[i][0] += dxv;
                else if (k1 == 1) dxr1[i][1] += dxv;
                else dxr1[i][2] += dxv;
                // d w
                dwacc[i] += cg * yv * xv;
                // d Y (per-lane partial; reduced after the loop)
                dyb[(k2 * 4 + wid) * 64 + lane] += cg * wreg[i] * xv;
            }
        }
        if (curp >= 0)
            for (int i = 0; i < CPL; ++i)
                dw[((int64_t)e * P + curp) * C + lane + i * 64] = dwacc[i];
        for (int i = 0; i < CPL; ++i)
            dx0[e * C + lane + i * 64] = dxr0[i];
        if (dx1 != nullptr)
            for (int i = 0; i < CPL; ++i)
                for (int k = 0; k < d1b; ++k)
                    dx1[(e * C + lane + i * 64) * d1b + k] = dxr1[i][k];
        // reduce dY partials across the wave: waves are lockstep, each
        // owns its LDS stripe — no block barrier (one inside this
        // grid-stride loop would deadlock when wave trip counts differ)
        if (lane < 16) {
            float s = 0.0f;
            for (int l2 = 0; l2 < 64; ++l2)
                s += dyb[(lane * 4 + wid) * 64 + l2];
            dY[e * 16 + lane] = s;
        }
    }
}



// ---------------------------------------------------------------------------
// MACE symmetric contraction, fused per node (round 2).  Replaces the
// SymmetricContraction of mace's EquivariantProductBasisBlock (the
// reference delegates to mace.modules.symmetric_contraction via
// models.py:155-157): out[n, orow, c] = sum_entries
// W[elem(n), wrow, c] * coef * x[n,a,c] * x[n,b,c] * x[n,k,c], with the
// sentinel row x[16] == 1 encoding nu<3 terms.  Entries are wave-uniform
// (scalar cache), sorted by wrow so each weight row loads once; weights
// are FROZEN (inference engine), so backward produces dx only.
// ---------------------------------------------------------------------------

template <int CPL>
__global__ void k_mace_symc_fwd(const float* __restrict__ x,
                                const int32_t* __restrict__ elem,
                                const float* __restrict__ W,
                                const int32_t* __restrict__ nz,
                                const float* __restrict__ nzc,
                                int32_t nnz, float* __restrict__ out,
                                int64_t N, int32_t C, int32_t T,
                                int32_t S_out) {
    // x rows AND accumulators in LDS: a/b/k/orow are runtime indices
    // (dynamic register-array indexing would spill to scratch)
    __shared__ float acc[4 * 4 * 64 * CPL];       // [orow][wave][lane][i]
    __shared__ float xls[17 * 4 * 64 * CPL];      // [row][wave][lane][i]
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
#define XR(a, i) xls[(((a) * 4 + wid) * 64 + lane) * CPL + (i)]
    for (int64_t n = blockIdx.x * (int64_t)nw + wid; n < N;
         n += (int64_t)gridDim.x * nw) {
        for (int a = 0; a < 16; ++a)
            for (int i = 0; i < CPL; ++i)
                XR(a, i) = x[((int64_t)n * 16 + a) * C + lane + i * 64];
        for (int i = 0; i < CPL; ++i)
            XR(16, i) = 1.0f;
        for (int o = 0; o < S_out; ++o)
            for (int i = 0; i < CPL; ++i)
                acc[((o * 4 + wid) * 64 + lane) * CPL + i] = 0.0f;
        const int64_t wbase = (int64_t)elem[n] * T;
        int curw = -1;
        float wreg[CPL];
        for (int32_t t = 0; t < nnz; ++t) {
            const int32_t wrow = nz[6 * t], a = nz[6 * t + 1],
                          b = nz[6 * t + 2], k = nz[6 * t + 3],
                          orow = nz[6 * t + 4];
            if (wrow != curw) {
                curw = wrow;
                for (int i = 0; i < CPL; ++i)
                    wreg[i] = W[(wbase + wrow) * C + lane + i * 64];
            }
            const float c = nzc[t];
            for (int i = 0; i < CPL; ++i) {
                const float v = c * wreg[i] * XR(a, i) * XR(b, i)
                                * XR(k, i);
                acc[((orow * 4 + wid) * 64 + lane) * CPL + i] += v;
            }
        }
        for (int o = 0; o < S_out; ++o)
            for (int i = 0; i < CPL; ++i)
                out[((int64_t)n * S_out + o) * C + lane + i * 64] =
                    acc[((o * 4 + wid) * 64 + lane) * CPL + i];
    }
#undef XR
}

template <int CPL>
__global__ void k_mace_symc_bwd(const float* __restrict__ go,
                                const float* __restrict__ x,
                                const int32_t* __restrict__ elem,
                                const float* __restrict__ W,
                                const int32_t* __restrict__ nz,
                                const float* __restrict__ nzc,
                                int32_t nnz, float* __restrict__ dx,
                                int64_t N, int32_t C, int32_t T,
                                int32_t S_out) {
    __shared__ float dacc[17 * 4 * 64 * CPL];     // row 16 = sentinel sink
    __shared__ float xls[17 * 4 * 64 * CPL];
    __shared__ float gls[4 * 4 * 64 * CPL];
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
#define XRB(a, i) xls[(((a) * 4 + wid) * 64 + lane) * CPL + (i)]
#define GOB(o, i) gls[(((o) * 4 + wid) * 64 + lane) * CPL + (i)]
    for (int64_t n = blockIdx.x * (int64_t)nw + wid; n < N;
         n += (int64_t)gridDim.x * nw) {
        for (int a = 0; a < 16; ++a)
            for (int i = 0; i < CPL; ++i)
                XRB(a, i) = x[((int64_t)n * 16 + a) * C + lane + i * 64];
        for (int i = 0; i < CPL; ++i)
            XRB(16, i) = 1.0f;
        for (int o = 0; o < S_out; ++o)
            for (int i = 0; i < CPL; ++i)
                GOB(o, i) = go[((int64_t)n * S_out + o) * C + lane
                               + i * 64];
        for (int a = 0; a < 17; ++a)
            for (int i = 0; i < CPL; ++i)
                dacc[((a * 4 + wid) * 64 + lane) * CPL + i] = 0.0f;
        const int64_t wbase = (int64_t)elem[n] * T;
        int curw = -1;
        float wreg[CPL];
        for (int32_t t = 0; t < nnz; ++t) {
            const int32_t wrow = nz[6 * t], a = nz[6 * t + 1],
                          b = nz[6 * t + 2], k = nz[6 * t + 3],
                          orow = nz[6 * t + 4];
            if (wrow != curw) {
                curw = wrow;
                for (int i = 0; i < CPL; ++i)
                    wreg[i] = W[(wbase + wrow) * C + lane + i * 64];
            }
            const float c = nzc[t];
            for (int i = 0; i < CPL; ++i) {
                const float g = c * wreg[i] * GOB(orow, i);
                const float xa = XRB(a, i), xb = XRB(b, i),
                            xk = XRB(k, i);
                dacc[((a * 4 + wid) * 64 + lane) * CPL + i] += g * xb * xk;
                dacc[((b * 4 + wid) * 64 + lane) * CPL + i] += g * xa * xk;
                dacc[((k * 4 + wid) * 64 + lane) * CPL + i] += g * xa * xb;
            }
        }
        for (int a = 0; a < 16; ++a)
            for (int i = 0; i < CPL; ++i)
                dx[((int64_t)n * 16 + a) * C + lane + i * 64] =
                    dacc[((a * 4 + wid) * 64 + lane) * CPL + i];
    }
#undef XRB
#undef GOB
}



// ---------------------------------------------------------------------------
// eSCN/UMA per-edge Wigner rotations (round 2).  The reference's UMA path
// applies per-edge block-diagonal Wigner matrices around every SO(2)
// convolution (escn_md.py:283-291, delegated to fairchem/e3nn); as batched
// bmms of (S x S)(S x C) tiles these run rocBLAS at ~1% of peak (62% of a
// 16 s step, profiles/r2_uma_kernel_stats.csv).  Here: one wave per edge
// (gather/apply) or per node (apply/scatter over the dst CSR), D rows
// streamed wave-uniform through the scalar cache, channels across lanes.
// S = 9 (lmax 2).  trans selects D vs D^T (the inverse rotation).
// ---------------------------------------------------------------------------

template <int CPL>
__global__ void k_rot_gather(const float* __restrict__ h,
                             const int32_t* __restrict__ idx,
                             const float* __restrict__ D, int32_t trans,
                             float* __restrict__ out, int64_t E,
                             int32_t C) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
    for (int64_t e = blockIdx.x * (int64_t)nw + wid; e < E;
         e += (int64_t)gridDim.x * nw) {
        const int64_t row = idx ? (int64_t)idx[e] : e;
        float hr[9][CPL];
        for (int t = 0; t < 9; ++t)
            for (int i = 0; i < CPL; ++i)
                hr[t][i] = h[(row * 9 + t) * C + lane + i * 64];
        const float* De = D + e * 81;
        for (int s = 0; s < 9; ++s) {
            float acc[CPL] = {};
            for (int t = 0; t < 9; ++t) {
                const float d = trans ? De[t * 9 + s] : De[s * 9 + t];
                for (int i = 0; i < CPL; ++i)
                    acc[i] += d * hr[t][i];
            }
            for (int i = 0; i < CPL; ++i)
                out[((int64_t)e * 9 + s) * C + lane + i * 64] = acc[i];
        }
    }
}

template <int CPL>
__global__ void k_rot_scatter(const float* __restrict__ mt,
                              const float* __restrict__ D, int32_t trans,
                              const int32_t* __restrict__ perm,
                              const int32_t* __restrict__ row_ptr,
                              const float* __restrict__ base,
                              float* __restrict__ out, int64_t N,
                              int32_t C) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
    for (int64_t n = blockIdx.x * (int64_t)nw + wid; n < N;
         n += (int64_t)gridDim.x * nw) {
        float acc[9][CPL] = {};
        const int32_t lo = row_ptr[n], hi = row_ptr[n + 1];
        for (int32_t q = lo; q < hi; ++q) {
            const int64_t e = perm ? (int64_t)perm[q] : (int64_t)q;
            const float* De = D + e * 81;
            float mr[9][CPL];
            for (int t = 0; t < 9; ++t)
                for (int i = 0; i < CPL; ++i)
                    mr[t][i] = mt[(e * 9 + t) * C + lane + i * 64];
            for (int s = 0; s < 9; ++s)
                for (int t = 0; t < 9; ++t) {
                    const float d = trans ? De[t * 9 + s] : De[s * 9 + t];
                    for (int i = 0; i < CPL; ++i)
                        acc[s][i] += d * mr[t][i];
                }
        }
        for (int s = 0; s < 9; ++s)
            for (int i = 0; i < CPL; ++i) {
                const int64_t o = ((int64_t)n * 9 + s) * C + lane + i * 64;
                out[o] = acc[s][i] + (base ? base[o] : 0.0f);
            }
    }
}

template <int CPL>
__global__ void k_rot_dD(const float* __restrict__ go,
                         const float* __restrict__ h,
                         const int32_t* __restrict__ idx, int32_t trans,
                         float* __restrict__ dD, int64_t E, int32_t C) {
    // dD(s,t) = sum_c go[e,s,c] * in[row,t,c]; trans writes transposed
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int nw = blockDim.x >> 6;
    for (int64_t e = blockIdx.x * (int64_t)nw + wid; e < E;
         e += (int64_t)gridDim.x * nw) {
        const int64_t row = idx ? (int64_t)idx[e] : e;
        float hr[9][CPL], gr[9][CPL];
        for (int t = 0; t < 9; ++t)
            for (int i = 0; i < CPL; ++i) {
                hr[t][i] = h[(row * 9 + t) * C + lane + i * 64];
                gr[t][i] = go[((int64_t)e * 9 + t) * C + lane + i * 64];
            }
        for (int s = 0; s < 9; ++s)
            for (int t = 0; t < 9; ++t) {
                float p = 0.0f;
                for (int i = 0; i < CPL; ++i)
                    p += gr[s][i] * hr[t][i];
                // wave reduction (butterfly over 64 lanes)
                for (int off = 32; off > 0; off >>= 1)
                    p += __shfl_xor(p, off, 64);
                if (lane == 0)
                    dD[e * 81 + (trans ? t * 9 + s : s * 9 + t)] = p;
            }
    }
}


// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------
extern "C" {

int dm_gather_rows_f32(const float* x, const int32_t* idx, float* out,
                       int64_t n_out, int64_t D, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (D % 4 == 0) {
        const int64_t total = n_out * (D / 4);
        k_gather_rows_v4<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(
            (const float4*)x, idx, (float4*)out, total, (int32_t)(D / 4));
    } else {
        const int64_t total = n_out * D;
        k_gather_rows_s<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(
            x, idx, out, total, (int32_t)D);
    }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_edge_mlp3_f32(const float* erow, const float* WT, const float* bias,
                     const float* zs, const float* zd, const int32_t* src,
                     const int32_t* dst, float* out, float* out_act,
                     int64_t E, int64_t Din, int64_t Dout, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (Din != 64 || Dout != 128) {
        g_err = "edge_mlp fused kernel is compiled for Din=64, Dout=128";
        return -1;
    }
    // one wave per edge: BLOCK threads cover BLOCK/64 edges per iter
    k_edge_mlp_64x128<2><<<nblocks(E, BLOCK / 64), BLOCK, 0, s>>>(
        erow, WT, bias, zs, zd, nullptr, src, dst, nullptr, out, out_act, E);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_edge_mlp4_f32(const float* arow, const float* WT, const float* bias,
                     const float* z1, const float* z2, const float* zv,
                     const int32_t* lsrc, const int32_t* ldst,
                     const int32_t* center, float* out, float* out_act,
                     int64_t L, int64_t Din, int64_t Dout, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (Din != 64 || Dout != 128) {
        g_err = "edge_mlp fused kernel is compiled for Din=64, Dout=128";
        return -1;
    }
    k_edge_mlp_64x128<3><<<nblocks(L, BLOCK / 64), BLOCK, 0, s>>>(
        arow, WT, bias, z1, z2, zv, lsrc, ldst, center, out, out_act, L);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_gather_add3_f32(const float* zs, const float* zd, const float* ze,
                       const int32_t* src, const int32_t* dst, float* out,
                       float* out_act, int64_t E, int64_t D,
                       uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (D % 4 != 0) { g_err = "gather_add3 requires D % 4 == 0"; return -1; }
    const int64_t total = E * (D / 4);
    k_gather_add3_v4<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(
        (const float4*)zs, (const float4*)zd, (const float4*)ze, src, dst,
        (float4*)out, (float4*)out_act, total, (int32_t)(D / 4));
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_gather_add4_f32(const float* z1, const float* z2, const float* za,
                       const float* zv, const int32_t* lsrc,
                       const int32_t* ldst, const int32_t* center, float* out,
                       float* out_act, int64_t L, int64_t D,
                       uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (D % 4 != 0) { g_err = "gather_add4 requires D % 4 == 0"; return -1; }
    const int64_t total = L * (D / 4);
    k_gather_add4_v4<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(
        (const float4*)z1, (const float4*)z2, (const float4*)za,
        (const float4*)zv, lsrc, ldst, center, (float4*)out,
        (float4*)out_act, total, (int32_t)(D / 4));
    DM_CHECK_LAUNCH();
    return 0;
}

static int g_seg_variant = []() {
    const char* v = getenv("DM_SEG_VARIANT");
    return v ? atoi(v) : 1;            // 1 = unroll-4 (default), 0 = plain
}();

int dm_seg_sum_f32(const float* msg, const int32_t* row_ptr,
                   const float* base, float* out, int64_t N, int64_t D,
                   uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (D % 4 == 0 && D / 4 <= 16) {
        if (g_seg_variant == 1)
            k_seg_sum_v4_u4<4><<<nblocks(N, 16), BLOCK, 0, s>>>(
                (const float4*)msg, row_ptr, (const float4*)base,
                (float4*)out, N, (int32_t)(D / 4));
        else
            k_seg_sum_v4<4><<<nblocks(N, 16), BLOCK, 0, s>>>(
                (const float4*)msg, row_ptr, (const float4*)base,
                (float4*)out, N, (int32_t)(D / 4));
    } else if (D % 4 == 0 && D / 4 <= 32) {      // D == 128: the backward
        k_seg_sum_v4_u4<5><<<nblocks(N, 8), BLOCK, 0, s>>>(   // dz sums
            (const float4*)msg, row_ptr, (const float4*)base,
            (float4*)out, N, (int32_t)(D / 4));
    } else {
        k_seg_sum_s<<<nblocks(N * D, BLOCK), BLOCK, 0, s>>>(
            msg, row_ptr, base, out, N, (int32_t)D);
    }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_seg_sum_gather_f32(const float* msg, const int32_t* perm,
                          const int32_t* row_ptr, const float* base,
                          float* out, int64_t N, int64_t D, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (D % 4 == 0 && D / 4 <= 16) {
        k_seg_sum_gather_v4<4><<<nblocks(N, 16), BLOCK, 0, s>>>(
            (const float4*)msg, perm, row_ptr, (const float4*)base,
            (float4*)out, N, (int32_t)(D / 4));
    } else if (D % 4 == 0 && D / 4 <= 32) {
        k_seg_sum_gather_v4<5><<<nblocks(N, 8), BLOCK, 0, s>>>(
            (const float4*)msg, perm, row_ptr, (const float4*)base,
            (float4*)out, N, (int32_t)(D / 4));
    } else {
        k_seg_sum_gather_s<<<nblocks(N * D, BLOCK), BLOCK, 0, s>>>(
            msg, perm, row_ptr, base, out, N, (int32_t)D);
    }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_gated_combine_fwd_f32(const float* c, const float* g, const float* w,
                             const float* base, float* out, int64_t total,
                             uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    k_gated_combine_fwd<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(
        c, g, w, base, out, total);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_gated_combine_bwd_f32(const float* go, const float* c, const float* g,
                             const float* w, float* dc, float* dg, float* dw,
                             int64_t total, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    k_gated_combine_bwd<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(
        go, c, g, w, dc, dg, dw, total);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_edge_geom_rbf_fwd_f32(const float* pos, const int32_t* src,
                             const int32_t* dst, const float* offshift,
                             const float* freqs, float cutoff, int32_t pexp,
                             int32_t nrbf, float* bv, float* bd,
                             float* exp_out, int64_t E, uint64_t stream) {
    if (nrbf > 16) { g_err = "nrbf > 16"; return -1; }
    hipStream_t s = (hipStream_t)stream;
    k_edge_geom_rbf_fwd<<<nblocks(E, BLOCK), BLOCK, 0, s>>>(
        pos, src, dst, offshift, freqs, cutoff, pexp, nrbf, bv, bd, exp_out, E);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_edge_geom_rbf_bwd_f32(const float* go_bv, const float* go_bd,
                             const float* go_exp, const float* bv,
                             const float* bd, const float* freqs,
                             float cutoff, int32_t pexp, int32_t nrbf,
                             float* gbv_total, int64_t E, uint64_t stream) {
    if (nrbf > 16) { g_err = "nrbf > 16"; return -1; }
    hipStream_t s = (hipStream_t)stream;
    k_edge_geom_rbf_bwd<<<nblocks(E, BLOCK), BLOCK, 0, s>>>(
        go_bv, go_bd, go_exp, bv, bd, freqs, cutoff, pexp, nrbf, gbv_total, E);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_rbf_env_fwd_f32(const float* d, const float* freqs, float cutoff,
                       int32_t pexp, int32_t nrbf, float* exp_out, int64_t M,
                       uint64_t stream) {
    if (nrbf > 16) { g_err = "nrbf > 16"; return -1; }
    hipStream_t s = (hipStream_t)stream;
    k_rbf_env_fwd<<<nblocks(M, BLOCK), BLOCK, 0, s>>>(
        d, freqs, cutoff, pexp, nrbf, exp_out, M);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_rbf_env_bwd_f32(const float* go_exp, const float* d, const float* freqs,
                       float cutoff, int32_t pexp, int32_t nrbf, float* gd,
                       int64_t M, uint64_t stream) {
    if (nrbf > 16) { g_err = "nrbf > 16"; return -1; }
    hipStream_t s = (hipStream_t)stream;
    k_rbf_env_bwd<<<nblocks(M, BLOCK), BLOCK, 0, s>>>(
        go_exp, d, freqs, cutoff, pexp, nrbf, gd, M);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_silu_bwd_f32(const float* go_h, const float* go_z, const float* z,
                    float* dz, int64_t total, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    k_silu_bwd<<<nblocks(total, BLOCK), BLOCK, 0, s>>>(go_h, go_z, z, dz,
                                                       total);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_nl_count_f64(const double* pos, const int32_t* cid,
                    const int32_t* order, const int32_t* cell_start,
                    int32_t ncx, int32_t ncy, int32_t ncz,
                    double lx, double ly, double lz,
                    double r2tol, double tol, int32_t* cnt, int64_t N,
                    uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    k_nl<false><<<nblocks(N, BLOCK), BLOCK, 0, s>>>(
        pos, cid, order, cell_start, ncx, ncy, ncz, lx, ly, lz, r2tol, tol,
        0.0, nullptr, cnt, nullptr, nullptr, N);
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_nl_fill_f64(const double* pos, const int32_t* cid,
                   const int32_t* order, const int32_t* cell_start,
                   int32_t ncx, int32_t ncy, int32_t ncz,
                   double lx, double ly, double lz,
                   double r2tol, double tol, double br2tol,
                   const int32_t* row_ptr, int32_t* src, int8_t* off_i8,
                   uint8_t* bond_flag, int64_t N, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    k_nl<true><<<nblocks(N, BLOCK), BLOCK, 0, s>>>(
        pos, cid, order, cell_start, ncx, ncy, ncz, lx, ly, lz, r2tol, tol,
        br2tol, row_ptr, src, off_i8, bond_flag, N);
    DM_CHECK_LAUNCH();
    return 0;
}


int dm_mace_tp_fwd_f32(const float* x0, const float* x1, const float* Y,
                       const float* w, const int32_t* nz, const float* nzc,
                       int32_t nnz, float* o0, float* o1, float* o2,
                       float* o3, int64_t E, int32_t C,
                       int32_t P, int32_t d1b, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (C == 64)
        k_mace_tp_fwd<1><<<nblocks(E, 4), BLOCK, 0, s>>>(
            x0, x1, Y, w, nz, nzc, nnz, o0, o1, o2, o3, E, C, P, d1b);
    else if (C == 128)
        k_mace_tp_fwd<2><<<nblocks(E, 4), BLOCK, 0, s>>>(
            x0, x1, Y, w, nz, nzc, nnz, o0, o1, o2, o3, E, C, P, d1b);
    else { g_err = "dm_mace_tp: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_mace_tp_bwd_f32(const float* g0, const float* g1, const float* g2,
                       const float* g3, const float* x0, const float* x1,
                       const float* Y, const float* w, const int32_t* nz,
                       const float* nzc, int32_t nnz, float* dx0,
                       float* dx1, float* dY, float* dw, int64_t E,
                       int32_t C, int32_t P, int32_t d1b, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (C == 64)
        k_mace_tp_bwd<1><<<nblocks(E, 4), BLOCK, 0, s>>>(
            g0, g1, g2, g3, x0, x1, Y, w, nz, nzc, nnz, dx0, dx1, dY, dw,
            E, C, P, d1b);
    else if (C == 128)
        k_mace_tp_bwd<2><<<nblocks(E, 4), BLOCK, 0, s>>>(
            g0, g1, g2, g3, x0, x1, Y, w, nz, nzc, nnz, dx0, dx1, dY, dw,
            E, C, P, d1b);
    else { g_err = "dm_mace_tp: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}


int dm_mace_symc_fwd_f32(const float* x, const int32_t* elem,
                         const float* W, const int32_t* nz,
                         const float* nzc, int32_t nnz, float* out,
                         int64_t N, int32_t C, int32_t T, int32_t S_out,
                         uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (S_out > 4) { g_err = "dm_mace_symc: S_out > 4"; return -1; }
    if (C == 64)
        k_mace_symc_fwd<1><<<nblocks(N, 4), BLOCK, 0, s>>>(
            x, elem, W, nz, nzc, nnz, out, N, C, T, S_out);
    else if (C == 128)
        k_mace_symc_fwd<2><<<nblocks(N, 4), BLOCK, 0, s>>>(
            x, elem, W, nz, nzc, nnz, out, N, C, T, S_out);
    else { g_err = "dm_mace_symc: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_mace_symc_bwd_f32(const float* go, const float* x,
                         const int32_t* elem, const float* W,
                         const int32_t* nz, const float* nzc, int32_t nnz,
                         float* dx, int64_t N, int32_t C, int32_t T,
                         int32_t S_out, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (S_out > 4) { g_err = "dm_mace_symc: S_out > 4"; return -1; }
    if (C == 64)
        k_mace_symc_bwd<1><<<nblocks(N, 4), BLOCK, 0, s>>>(
            go, x, elem, W, nz, nzc, nnz, dx, N, C, T, S_out);
    else if (C == 128)
        k_mace_symc_bwd<2><<<nblocks(N, 4), BLOCK, 0, s>>>(
            go, x, elem, W, nz, nzc, nnz, dx, N, C, T, S_out);
    else { g_err = "dm_mace_symc: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}


int dm_rot_gather_f32(const float* h, const int32_t* idx, const float* D,
                      int32_t trans, float* out, int64_t E, int32_t C,
                      uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (C == 64)
        k_rot_gather<1><<<nblocks(E, 4), BLOCK, 0, s>>>(h, idx, D, trans,
                                                        out, E, C);
    else if (C == 128)
        k_rot_gather<2><<<nblocks(E, 4), BLOCK, 0, s>>>(h, idx, D, trans,
                                                        out, E, C);
    else { g_err = "dm_rot: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_rot_scatter_f32(const float* mt, const float* D, int32_t trans,
                       const int32_t* perm, const int32_t* row_ptr,
                       const float* base, float* out, int64_t N,
                       int32_t C, uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (C == 64)
        k_rot_scatter<1><<<nblocks(N, 4), BLOCK, 0, s>>>(
            mt, D, trans, perm, row_ptr, base, out, N, C);
    else if (C == 128)
        k_rot_scatter<2><<<nblocks(N, 4), BLOCK, 0, s>>>(
            mt, D, trans, perm, row_ptr, base, out, N, C);
    else { g_err = "dm_rot: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}

int dm_rot_dD_f32(const float* go, const float* h, const int32_t* idx,
                  int32_t trans, float* dD, int64_t E, int32_t C,
                  uint64_t stream) {
    hipStream_t s = (hipStream_t)stream;
    if (C == 64)
        k_rot_dD<1><<<nblocks(E, 4), BLOCK, 0, s>>>(go, h, idx, trans, dD,
                                                    E, C);
    else if (C == 128)
        k_rot_dD<2><<<nblocks(E, 4), BLOCK, 0, s>>>(go, h, idx, trans, dD,
                                                    E, C);
    else { g_err = "dm_rot: C must be 64 or 128"; return -1; }
    DM_CHECK_LAUNCH();
    return 0;
}

const char* dm_hip_last_error(void) { return g_err.c_str(); }

}  // extern "C"
