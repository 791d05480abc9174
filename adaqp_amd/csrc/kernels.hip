// CDNA4 (gfx950 / MI355X) HIP kernels for adaqp_amd.
//
// Reference parity: quant_cuda pack/unpack
// (/root/reference/AdaQP/util/quantization/src/quantization_cuda_kernel.cu:34-156)
// and DGL's update_all CSR SpMM (delegated by the reference at
// AdaQP/model/ops.py:30). MI355X-first redesign:
//
//  * quant_pack fuses per-node min/max + scale + stochastic round + bit-pack
//    + the per-(peer,bit-group) wire scatter into ONE kernel and ONE read
//    of x (lane-register staging for F<=1024; the reference computes
//    rmin/rmax with two torch reductions in Python first, then re-reads
//    x in the pack kernel).
//  * packing is along the FEATURE axis so each node is a contiguous byte
//    run: lane l owns (8/bits) consecutive features and emits whole bytes —
//    coalesced float4/float2/float loads and byte stores across the
//    64-wide wavefront (the reference packs along the node axis).
//  * RNG is a stateless counter hash (triple32) of (seed, node, feature) —
//    no curand state; bit-compatible with the CPU oracle in ops/quant.py.
//  * spmm_csr: one SUB-wavefront per destination-row segment (hub rows are
//    split and atomically combined), dtype-templated (fp32 / bf16 with fp32
//    accumulate), vector width chosen so row-base loads stay aligned, fused
//    src/dst degree scaling, XCD-aware block swizzle, int32 column indices
//    + segment bounds (half the index bytes), four gather loads in flight.
//
// Built for gfx950 only. No CUDA compatibility path.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>

#define WAVE 64
#define CHECK_DEV(x) TORCH_CHECK(x.is_cuda(), #x " must be on the GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

// ---------------------------------------------------------------------------
// RNG: triple32 hash -> U[0,1). Must match ops/quant.py::_hash_u32 exactly.
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint32_t hash_u32(uint32_t h) {
    h ^= h >> 16; h *= 0x7FEB352Du;
    h ^= h >> 15; h *= 0x846CA68Bu;
    h ^= h >> 16;
    return h;
}

__device__ __forceinline__ float uniform01(uint32_t seed, uint32_t tag, uint32_t feat) {
    uint32_t h = seed ^ (tag * 0x9E3779B9u) ^ (feat * 0x85EBCA6Bu);
    return (float)((double)hash_u32(h) * 2.3283064365386963e-10);
}

// round-to-nearest-even f32 -> bf16 bits (parity with torch .to(bfloat16))
__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
    uint32_t b = __float_as_uint(f);
    uint32_t r = (b + 0x7FFFu + ((b >> 16) & 1u)) >> 16;
    return (uint16_t)r;
}
__device__ __forceinline__ float bf16_to_f32(uint16_t h) {
    return __uint_as_float(((uint32_t)h) << 16);
}

// T = float (V=4 feats/lane) or ushort bf16 bits (V=8 feats/lane); fp32 accumulate.
template <typename T>
__device__ __forceinline__ float to_f32(T v);
template <> __device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ float to_f32<ushort>(ushort v) {
    return bf16_to_f32(v);
}


// ---------------------------------------------------------------------------
// quant_pack: one wave per node.
//   rows[i]   : row of x to quantize
//   pos[i]    : wire node position (params slot + RNG tag)
//   off[i]    : wire byte offset of this node's payload
// Lane l owns features [l*vpb + k*WAVE*vpb, ...): it loads vpb consecutive
// floats, reduces min/max across the wave, then emits one byte per k.
// ---------------------------------------------------------------------------
template <int BITS, typename T>
__global__ void quant_pack_kernel(
    const T* __restrict__ x, const int64_t* __restrict__ rows,
    const int64_t* __restrict__ pos, const int64_t* __restrict__ off,
    int64_t n, int64_t F, int64_t ld, uint32_t seed,
    uint8_t* __restrict__ payload, uint16_t* __restrict__ params) {
    constexpr int VPB = 8 / BITS;          // values per byte
    constexpr int MAXV = 16;               // staged floats per lane
    constexpr int MAXCH = MAXV / VPB;      // register path covers F <= WAVE*MAXV
    const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + threadIdx.x / WAVE;
    if (wid >= n) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int64_t row = rows[wid];
    const T* xr = x + row * ld;
    const int64_t bpn = (F * BITS + 7) / 8;

    // single pass over x: stage this lane's values in registers while
    // reducing min/max (the round-1 kernel read x twice). F beyond the
    // register budget falls back to the two-pass read below.
    const bool staged = F <= (int64_t)WAVE * MAXV;
    float vals[MAXV];
    float mn = 1e38f, mx = -1e38f;
    if (staged) {
#pragma unroll
        for (int c = 0; c < MAXCH; ++c) {
            const int64_t f0 = ((int64_t)lane + (int64_t)c * WAVE) * VPB;
            if (f0 >= F) break;
#pragma unroll
            for (int k = 0; k < VPB; ++k) {
                float v = 0.f;
                if (f0 + k < F) {
                    v = to_f32<T>(xr[f0 + k]);
                    mn = fminf(mn, v);
                    mx = fmaxf(mx, v);
                }
                vals[c * VPB + k] = v;
            }
        }
    } else {
        for (int64_t f0 = (int64_t)lane * VPB; f0 < F; f0 += (int64_t)WAVE * VPB) {
#pragma unroll
            for (int k = 0; k < VPB; ++k) {
                if (f0 + k < F) {
                    float v = to_f32<T>(xr[f0 + k]);
                    mn = fminf(mn, v);
                    mx = fmaxf(mx, v);
                }
            }
        }
    }
#pragma unroll
    for (int o = 32; o; o >>= 1) {
        mn = fminf(mn, __shfl_xor(mn, o));
        mx = fmaxf(mx, __shfl_xor(mx, o));
    }
    const float rng = mx - mn;
    float scale_f = rng > 0.f ? ((float)((1 << BITS) - 1)) / fmaxf(rng, 1e-30f) : 0.f;
    const uint16_t scale_h = f32_to_bf16(scale_f);
    const uint16_t rmin_h = f32_to_bf16(mn);
    const float scale = bf16_to_f32(scale_h);     // quantize with the wire scale
    const float rmin = bf16_to_f32(rmin_h);
    const int64_t p = pos[wid];
    if (lane == 0) {
        params[2 * p] = scale_h;
        params[2 * p + 1] = rmin_h;
    }
    const uint32_t tag = (uint32_t)p;

    // quantize + pack one byte per lane-step. The staged path is fully
    // unrolled so vals[] indices are compile-time constants (registers,
    // no scratch).
    uint8_t* out = payload + off[wid];
    if (staged) {
#pragma unroll
        for (int c = 0; c < MAXCH; ++c) {
            const int64_t b0 = (int64_t)lane + (int64_t)c * WAVE;
            if (b0 * VPB >= F) break;
            uint32_t byte = 0;
#pragma unroll
            for (int k = 0; k < VPB; ++k) {
                const int64_t f = b0 * VPB + k;
                if (f < F && scale > 0.f) {
                    float v = (vals[c * VPB + k] - rmin) * scale;
                    float u = uniform01(seed, tag, (uint32_t)f);
                    int q = (int)floorf(v + u);
                    q = max(0, min(q, (1 << BITS) - 1));
                    byte |= ((uint32_t)q) << (k * BITS);
                }
            }
            out[b0] = (uint8_t)byte;
        }
    } else {
        for (int64_t b0 = lane; b0 * VPB < F; b0 += WAVE) {
            uint32_t byte = 0;
#pragma unroll
            for (int k = 0; k < VPB; ++k) {
                const int64_t f = b0 * VPB + k;
                if (f < F && scale > 0.f) {
                    float v = (to_f32<T>(xr[f]) - rmin) * scale;
                    float u = uniform01(seed, tag, (uint32_t)f);
                    int q = (int)floorf(v + u);
                    q = max(0, min(q, (1 << BITS) - 1));
                    byte |= ((uint32_t)q) << (k * BITS);
                }
            }
            out[b0] = (uint8_t)byte;
        }
    }
}

// ---------------------------------------------------------------------------
// quant_unpack: one wave per node, fused scatter into out[rows[i]].
// ---------------------------------------------------------------------------
template <int BITS, typename T>
__global__ void quant_unpack_kernel(
    const uint8_t* __restrict__ payload, const uint16_t* __restrict__ params,
    const int64_t* __restrict__ rows, const int64_t* __restrict__ pos,
    const int64_t* __restrict__ off, int64_t n, int64_t F, int64_t ld,
    T* __restrict__ out) {
    constexpr bool BF = sizeof(T) == 2;
    constexpr int VPB = 8 / BITS;
    const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + threadIdx.x / WAVE;
    if (wid >= n) return;
    const int lane = threadIdx.x & (WAVE - 1);
    const int64_t p = pos[wid];
    const float scale = bf16_to_f32(params[2 * p]);
    const float rmin = bf16_to_f32(params[2 * p + 1]);
    const float inv = scale > 0.f ? 1.f / scale : 0.f;
    const uint8_t* in = payload + off[wid];
    T* o = out + rows[wid] * ld;
    for (int64_t b0 = lane; b0 * VPB < F; b0 += WAVE) {
        const uint32_t byte = in[b0];
#pragma unroll
        for (int k = 0; k < VPB; ++k) {
            const int64_t f = b0 * VPB + k;
            if (f < F) {
                const int q = (byte >> (k * BITS)) & ((1 << BITS) - 1);
                const float v = (float)q * inv + rmin;
                if constexpr (BF) o[f] = f32_to_bf16(v);
                else o[f] = v;
            }
        }
    }
}

// ---------------------------------------------------------------------------
// spmm_csr: y[r] = dst_scale[r] * sum_{e in row r} src_scale[c_e] * x[c_e]
//
// One SUB-wavefront (SW in {16,32,64}, chosen to cover F) per destination-row
// SEGMENT (seg_* arrays; hub rows are split to <=SEG_EDGES edges and combined
// with atomics into pre-zeroed rows). Lane sl owns features
// [VE*sl, VE*sl+VE) per chunk of VE*SW, VE picked host-side so the
// VE*sizeof(T)-byte row-base loads stay aligned (VE | F). Two edges are
// accumulated in flight (dual accumulators) to cover gather latency.
// Dual-tensor input: columns >= n_local read the REMOTE block directly
// (no [N,F] concat per layer). XCD-aware bijective block swizzle gives
// each XCD a contiguous chunk of work items so clustered neighbor rows
// hit the same L2 (guide §5.5 T1).
// ---------------------------------------------------------------------------
template <typename T>
__global__ void zero_rows_kernel(T* __restrict__ y,
                                 const int32_t* __restrict__ rows,
                                 int64_t n, int64_t F) {
    const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + threadIdx.x / WAVE;
    if (wid >= n) return;
    T* p = y + (int64_t)rows[wid] * F;
    for (int64_t f = threadIdx.x & (WAVE - 1); f < F; f += WAVE)
        p[f] = T(0);
}

// VE = elements per lane per load; chosen host-side as the largest of
// {8,4,2}(bf16) / {4,2,1}(f32) dividing F, so every row-base load of
// width VE*sizeof(T) is aligned (rows are x + c*F*sizeof(T)).
template <int NB> struct RawVec;
template <> struct RawVec<16> { using type = uint4; };
template <> struct RawVec<8>  { using type = uint2; };
template <> struct RawVec<4>  { using type = unsigned; };
template <> struct RawVec<2>  { using type = ushort; };

template <int SW, typename T, int VE>
__global__ void spmm_csr_kernel(
    const int32_t* __restrict__ indices,   // int32: halves index bytes
    const T* __restrict__ xl, const T* __restrict__ xr,
    T* __restrict__ y,
    const float* __restrict__ src_scale, const float* __restrict__ dst_scale,
    const int32_t* __restrict__ seg_row, const int32_t* __restrict__ seg_e0,
    const int32_t* __restrict__ seg_e1, const uint8_t* __restrict__ seg_multi,
    int64_t n_seg, int64_t F, int64_t n_local) {
    constexpr bool BF = sizeof(T) == 2;
    constexpr int NB = VE * sizeof(T);
    using Raw = typename RawVec<NB>::type;
    const int rows_per_block = blockDim.x / SW;
    // bijective XCD swizzle: blocks [0,nwg) -> xcd-contiguous chunks
    const int64_t nwg = gridDim.x;
    const int64_t q = nwg / 8, rem = nwg % 8;
    const int64_t xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
    const int64_t swz = (xcd < rem ? xcd * (q + 1) : rem * (q + 1) + (xcd - rem) * q) + idx;
    const int64_t sub0 = swz * rows_per_block + threadIdx.x / SW;
    const int64_t stride = nwg * rows_per_block;
    const int sl = threadIdx.x & (SW - 1);

    for (int64_t it = sub0; it < n_seg; it += stride) {
        const int64_t r = seg_row[it];
        const int64_t e0 = seg_e0[it], e1 = seg_e1[it];
        const bool multi = seg_multi[it];
        const float ds = dst_scale ? dst_scale[r] : 1.f;
        for (int64_t f0 = (int64_t)sl * VE; f0 < F; f0 += (int64_t)SW * VE) {
            float acc0[VE], acc1[VE];
#pragma unroll
            for (int k = 0; k < VE; ++k) { acc0[k] = 0.f; acc1[k] = 0.f; }
            int64_t e = e0;
            // FOUR edges in flight: the gather is latency-bound on L2
            // misses; independent loads + accumulator pairs keep more
            // of them outstanding (round-1 version had two)
            for (; e + 3 < e1; e += 4) {
                const int64_t c0 = indices[e],     c1 = indices[e + 1];
                const int64_t c2 = indices[e + 2], c3 = indices[e + 3];
                const float s0 = src_scale ? src_scale[c0] : 1.f;
                const float s1 = src_scale ? src_scale[c1] : 1.f;
                const float s2 = src_scale ? src_scale[c2] : 1.f;
                const float s3 = src_scale ? src_scale[c3] : 1.f;
                const T* p0 = (c0 < n_local ? xl + c0 * F : xr + (c0 - n_local) * F) + f0;
                const T* p1 = (c1 < n_local ? xl + c1 * F : xr + (c1 - n_local) * F) + f0;
                const T* p2 = (c2 < n_local ? xl + c2 * F : xr + (c2 - n_local) * F) + f0;
                const T* p3 = (c3 < n_local ? xl + c3 * F : xr + (c3 - n_local) * F) + f0;
                const Raw r0 = *reinterpret_cast<const Raw*>(p0);
                const Raw r1 = *reinterpret_cast<const Raw*>(p1);
                const Raw r2 = *reinterpret_cast<const Raw*>(p2);
                const Raw r3 = *reinterpret_cast<const Raw*>(p3);
                const T* v0 = reinterpret_cast<const T*>(&r0);
                const T* v1 = reinterpret_cast<const T*>(&r1);
                const T* v2 = reinterpret_cast<const T*>(&r2);
                const T* v3 = reinterpret_cast<const T*>(&r3);
#pragma unroll
                for (int k = 0; k < VE; ++k) {
                    acc0[k] = fmaf(to_f32<T>(v0[k]), s0, acc0[k]);
                    acc1[k] = fmaf(to_f32<T>(v1[k]), s1, acc1[k]);
                    acc0[k] = fmaf(to_f32<T>(v2[k]), s2, acc0[k]);
                    acc1[k] = fmaf(to_f32<T>(v3[k]), s3, acc1[k]);
                }
            }
            for (; e + 1 < e1; e += 2) {   // VE divides F: loads always in-bounds
                const int64_t c0 = indices[e];
                const int64_t c1 = indices[e + 1];
                const float s0 = src_scale ? src_scale[c0] : 1.f;
                const float s1 = src_scale ? src_scale[c1] : 1.f;
                const T* p0 = (c0 < n_local ? xl + c0 * F : xr + (c0 - n_local) * F) + f0;
                const T* p1 = (c1 < n_local ? xl + c1 * F : xr + (c1 - n_local) * F) + f0;
                const Raw r0 = *reinterpret_cast<const Raw*>(p0);
                const Raw r1 = *reinterpret_cast<const Raw*>(p1);
                const T* v0 = reinterpret_cast<const T*>(&r0);
                const T* v1 = reinterpret_cast<const T*>(&r1);
#pragma unroll
                for (int k = 0; k < VE; ++k) {
                    acc0[k] = fmaf(to_f32<T>(v0[k]), s0, acc0[k]);
                    acc1[k] = fmaf(to_f32<T>(v1[k]), s1, acc1[k]);
                }
            }
            for (; e < e1; ++e) {
                const int64_t c = indices[e];
                const float s = src_scale ? src_scale[c] : 1.f;
                const T* xc = (c < n_local ? xl + c * F : xr + (c - n_local) * F) + f0;
                const Raw rv = *reinterpret_cast<const Raw*>(xc);
                const T* v = reinterpret_cast<const T*>(&rv);
#pragma unroll
                for (int k = 0; k < VE; ++k)
                    acc0[k] = fmaf(to_f32<T>(v[k]), s, acc0[k]);
            }
            T* yr = y + r * F + f0;
            if (!multi) {
                T outv[VE];
#pragma unroll
                for (int k = 0; k < VE; ++k) {
                    const float o = (acc0[k] + acc1[k]) * ds;
                    if constexpr (BF) outv[k] = f32_to_bf16(o);
                    else outv[k] = o;
                }
                *reinterpret_cast<Raw*>(yr) = *reinterpret_cast<const Raw*>(outv);
            } else if constexpr (BF) {
                // long row split across segments: combine via packed atomics
#pragma unroll
                for (int k = 0; k < VE; k += 2) {
                    __hip_bfloat162 v;
                    v.x = __float2bfloat16((acc0[k] + acc1[k]) * ds);
                    v.y = __float2bfloat16((acc0[k + 1] + acc1[k + 1]) * ds);
                    unsafeAtomicAdd(reinterpret_cast<__hip_bfloat162*>(yr + k), v);
                }
            } else {
#pragma unroll
                for (int k = 0; k < VE; ++k)
                    atomicAdd(reinterpret_cast<float*>(yr) + k,
                              (acc0[k] + acc1[k]) * ds);
            }
        }
    }
}

// ---------------------------------------------------------------------------
// fused dual GEMM (bf16, MFMA): out[M,N] = A1@W1 + A2@W2 + bias
//
// The SAGE 'mean' layer computes fc_self(x) + fc_neigh(h) + biases — two
// rocBLAS GEMMs plus an elementwise add, writing/reading the [M,N] output
// three times. This kernel produces it in ONE pass: per 64-row block, the
// TRANSPOSED weights (Wt = W.T, [N,K]) are staged through LDS (shared by
// the block's 4 waves; W1 first, then W2), and each wave accumulates its
// 16xN rows with v_mfma_f32_16x16x32_bf16. fp32 accumulate, bf16 I/O.
// Fragment layouts per guide §3: A row = lane&15, k = (lane>>4)*8+j;
// B col = lane&15, same k; C/D col = lane&15, row = (lane>>4)*4+reg.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define FDG_MAXNT 16      // N <= 256

__device__ __forceinline__ bf16x8 load_bf16x8(const ushort* p) {
    uint4 raw = *reinterpret_cast<const uint4*>(p);
    return *reinterpret_cast<const bf16x8*>(&raw);
}

__global__ void __launch_bounds__(256)
fused_dual_gemm_bf16_kernel(
    const ushort* __restrict__ a1, const ushort* __restrict__ a2,
    const ushort* __restrict__ w1t, const ushort* __restrict__ w2t,
    const ushort* __restrict__ bias,     // [N] bf16 (b1+b2), may be null
    ushort* __restrict__ out,
    int64_t M, int64_t N, int64_t K1, int64_t K2) {
    // Wt fragments are read straight from global: the whole weight set
    // (<=256KB) is L2-resident and re-read by every block; LDS staging
    // was measured SLOWER (135KB/block -> 1 block/CU, no latency hiding).
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    const int64_t row0 = (int64_t)blockIdx.x * 64 + wid * 16;
    const int arow = lane & 15;          // A row / B col within a 16-tile
    const int kgrp = lane >> 4;          // k sub-block (8 elems each)
    const int NT = (int)(N / 16);

    f32x4 acc[FDG_MAXNT];
#pragma unroll
    for (int nt = 0; nt < FDG_MAXNT; ++nt) acc[nt] = {0.f, 0.f, 0.f, 0.f};

    for (int pass = 0; pass < 2; ++pass) {
        const ushort* A = pass ? a2 : a1;
        const ushort* Wt = pass ? w2t : w1t;
        const int64_t K = pass ? K2 : K1;
        const int64_t arow_g = row0 + arow;
        const ushort* arow_p = A + arow_g * K;
        const ushort* wt_p = Wt + arow * K;   // row (=out col) within 16-tile
        // double-buffered K pipeline: while the MFMAs of chunk i issue,
        // the loads of chunk i+1 are already in flight (round-1 version
        // had no K-pipelining — every chunk stalled on L2/HBM latency)
        bf16x8 af0 = {}, af1 = {};
        bf16x8 b0[FDG_MAXNT], b1[FDG_MAXNT];
        auto load_chunk = [&](int64_t k, bf16x8& af, bf16x8* bfr) {
            af = (arow_g < M && k < K) ? load_bf16x8(arow_p + k) : bf16x8{};
#pragma unroll
            for (int nt = 0; nt < FDG_MAXNT; ++nt) {
                if (nt >= NT) break;
                bfr[nt] = (k < K) ? load_bf16x8(wt_p + (int64_t)nt * 16 * K + k)
                                  : bf16x8{};
            }
        };
        auto mfma_chunk = [&](const bf16x8& af, const bf16x8* bfr) {
#pragma unroll
            for (int nt = 0; nt < FDG_MAXNT; ++nt) {
                if (nt >= NT) break;
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af, bfr[nt], acc[nt], 0, 0, 0);
            }
        };
        load_chunk(kgrp * 8, af0, b0);
        for (int64_t k0 = 0; k0 < K; k0 += 64) {
            if (k0 + 32 < K) load_chunk(k0 + 32 + kgrp * 8, af1, b1);
            mfma_chunk(af0, b0);
            if (k0 + 64 < K) load_chunk(k0 + 64 + kgrp * 8, af0, b0);
            if (k0 + 32 < K) mfma_chunk(af1, b1);
        }
    }

    // epilogue: bias + bf16 store. C/D: col = lane&15, row = kgrp*4 + q
#pragma unroll
    for (int nt = 0; nt < FDG_MAXNT; ++nt) {
        if (nt >= NT) break;
        const int col = nt * 16 + arow;
        const float bv = bias ? bf16_to_f32(bias[col]) : 0.f;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            const int64_t r = row0 + kgrp * 4 + q;
            if (r < M)
                out[r * N + col] = f32_to_bf16(acc[nt][q] + bv);
        }
    }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------
static inline hipStream_t cur_stream() {
    return at::cuda::getCurrentHIPStream().stream();
}

void quant_pack(torch::Tensor x, torch::Tensor rows, torch::Tensor pos,
                torch::Tensor off, int64_t bits, int64_t seed,
                torch::Tensor payload, torch::Tensor params) {
    CHECK_DEV(x); CHECK_CONTIG(x); CHECK_DEV(rows); CHECK_DEV(payload); CHECK_DEV(params);
    const bool bf16 = x.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat32,
                "quant_pack expects fp32 or bf16");
    const int64_t n = rows.numel();
    if (n == 0) return;
    const int64_t F = x.size(1), ld = x.stride(0);
    const int waves_per_block = 4;
    const dim3 block(WAVE * waves_per_block);
    const dim3 grid((n + waves_per_block - 1) / waves_per_block);
    auto s = cur_stream();
#define LAUNCH(B, T) quant_pack_kernel<B, T><<<grid, block, 0, s>>>( \
        reinterpret_cast<const T*>(x.data_ptr()), rows.data_ptr<int64_t>(), \
        pos.data_ptr<int64_t>(), \
        off.data_ptr<int64_t>(), n, F, ld, (uint32_t)seed, \
        payload.data_ptr<uint8_t>(), reinterpret_cast<uint16_t*>(params.data_ptr<at::BFloat16>()))
    switch (bits) {
        case 2: if (bf16) LAUNCH(2, ushort); else LAUNCH(2, float); break;
        case 4: if (bf16) LAUNCH(4, ushort); else LAUNCH(4, float); break;
        case 8: if (bf16) LAUNCH(8, ushort); else LAUNCH(8, float); break;
        default: TORCH_CHECK(false, "bits must be 2/4/8");
    }
#undef LAUNCH
}

void quant_unpack(torch::Tensor payload, torch::Tensor params, torch::Tensor rows,
                  torch::Tensor pos, torch::Tensor off, int64_t bits, int64_t F,
                  torch::Tensor out) {
    CHECK_DEV(payload); CHECK_DEV(out); CHECK_CONTIG(out);
    const bool bf16 = out.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(bf16 || out.scalar_type() == torch::kFloat32,
                "quant_unpack expects fp32 or bf16 out");
    const int64_t n = rows.numel();
    if (n == 0) return;
    const int waves_per_block = 4;
    const dim3 block(WAVE * waves_per_block);
    const dim3 grid((n + waves_per_block - 1) / waves_per_block);
    auto s = cur_stream();
#define LAUNCH(B, T) quant_unpack_kernel<B, T><<<grid, block, 0, s>>>( \
        payload.data_ptr<uint8_t>(), \
        reinterpret_cast<uint16_t*>(params.data_ptr<at::BFloat16>()), \
        rows.data_ptr<int64_t>(), pos.data_ptr<int64_t>(), off.data_ptr<int64_t>(), \
        n, F, out.stride(0), reinterpret_cast<T*>(out.data_ptr()))
    switch (bits) {
        case 2: if (bf16) LAUNCH(2, ushort); else LAUNCH(2, float); break;
        case 4: if (bf16) LAUNCH(4, ushort); else LAUNCH(4, float); break;
        case 8: if (bf16) LAUNCH(8, ushort); else LAUNCH(8, float); break;
        default: TORCH_CHECK(false, "bits must be 2/4/8");
    }
#undef LAUNCH
}

void spmm_csr(torch::Tensor indices, torch::Tensor xl,
              torch::Tensor xr, torch::Tensor y, torch::Tensor src_scale,
              torch::Tensor dst_scale, torch::Tensor seg_row,
              torch::Tensor seg_e0, torch::Tensor seg_e1,
              torch::Tensor seg_multi, torch::Tensor zero_rows) {
    CHECK_DEV(xl); CHECK_CONTIG(xl); CHECK_DEV(y); CHECK_CONTIG(y);
    const bool bf16 = xl.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(bf16 || xl.scalar_type() == torch::kFloat32,
                "spmm_csr expects fp32 or bf16");
    TORCH_CHECK(y.scalar_type() == xl.scalar_type(), "x/y dtype mismatch");
    const int64_t F = xl.size(1);
    const int64_t n_local = xl.size(0);
    const int64_t n_seg = seg_row.numel();
    const void* xr_ptr = nullptr;
    if (xr.numel()) {
        CHECK_DEV(xr); CHECK_CONTIG(xr);
        TORCH_CHECK(xr.size(1) == F, "remote feature dim mismatch");
        TORCH_CHECK(xr.scalar_type() == xl.scalar_type(), "x dtype mismatch");
        xr_ptr = xr.data_ptr();
    }
    auto s = cur_stream();
    if (zero_rows.numel()) {
        const dim3 zb(256), zg((zero_rows.numel() + 3) / 4);
        if (bf16)
            zero_rows_kernel<ushort><<<zg, zb, 0, s>>>(
                reinterpret_cast<ushort*>(y.data_ptr()),
                zero_rows.data_ptr<int32_t>(), zero_rows.numel(), F);
        else
            zero_rows_kernel<float><<<zg, zb, 0, s>>>(
                y.data_ptr<float>(), zero_rows.data_ptr<int32_t>(),
                zero_rows.numel(), F);
    }
    if (n_seg == 0) return;
    // largest vector width dividing F (keeps every row-base load aligned)
    int ve = bf16 ? 8 : 4;
    while (ve > 1 && F % ve) ve >>= 1;
    if (bf16 && ve == 1) ve = 2;              // bf16 atomics need even lanes
    TORCH_CHECK(!bf16 || F % 2 == 0, "bf16 spmm needs even feature dim");
    // sub-wavefront width: smallest of {16,32,64} covering F with ve/lane
    int sw = 16;
    while (sw < 64 && (int64_t)sw * ve < F) sw *= 2;
    const int block_threads = WAVE * 4;
    const int rows_per_block = block_threads / sw;
    int64_t blocks = (n_seg + rows_per_block - 1) / rows_per_block;
    blocks = std::max<int64_t>(std::min<int64_t>(blocks, 16384), 1);
    const dim3 grid(blocks), block(block_threads);
    TORCH_CHECK(indices.scalar_type() == torch::kInt32,
                "spmm_csr expects int32 indices (SpmmView builds them)");
    const int32_t* ind_p = indices.data_ptr<int32_t>();
    const float* ss_p = src_scale.numel() ? src_scale.data_ptr<float>() : nullptr;
    const float* ds_p = dst_scale.numel() ? dst_scale.data_ptr<float>() : nullptr;
    const int32_t* sr_p = seg_row.data_ptr<int32_t>();
    TORCH_CHECK(seg_e0.scalar_type() == torch::kInt32 &&
                seg_e1.scalar_type() == torch::kInt32,
                "seg_e0/e1 must be int32 (SpmmView builds them)");
    const int32_t* e0_p = seg_e0.data_ptr<int32_t>();
    const int32_t* e1_p = seg_e1.data_ptr<int32_t>();
    const uint8_t* sm_p = seg_multi.data_ptr<uint8_t>();
    auto run = [&](auto sw_tag, auto t_tag, auto ve_tag) {
        constexpr int SWC = decltype(sw_tag)::value;
        using TC = typename decltype(t_tag)::type;
        constexpr int VEC = decltype(ve_tag)::value;
        spmm_csr_kernel<SWC, TC, VEC><<<grid, block, 0, s>>>(
            ind_p, reinterpret_cast<const TC*>(xl.data_ptr()),
            reinterpret_cast<const TC*>(xr_ptr),
            reinterpret_cast<TC*>(y.data_ptr()), ss_p, ds_p,
            sr_p, e0_p, e1_p, sm_p, n_seg, F, n_local);
    };
    struct FT { using type = float; };
    struct BT { using type = ushort; };
    auto by_ve = [&](auto sw_tag) {
        if (bf16) {
            switch (ve) {
                case 8: run(sw_tag, BT{}, std::integral_constant<int,8>{}); break;
                case 4: run(sw_tag, BT{}, std::integral_constant<int,4>{}); break;
                default: run(sw_tag, BT{}, std::integral_constant<int,2>{}); break;
            }
        } else {
            switch (ve) {
                case 4: run(sw_tag, FT{}, std::integral_constant<int,4>{}); break;
                case 2: run(sw_tag, FT{}, std::integral_constant<int,2>{}); break;
                default: run(sw_tag, FT{}, std::integral_constant<int,1>{}); break;
            }
        }
    };
    switch (sw) {
        case 64: by_ve(std::integral_constant<int,64>{}); break;
        case 32: by_ve(std::integral_constant<int,32>{}); break;
        default: by_ve(std::integral_constant<int,16>{}); break;
    }
}


void fused_dual_gemm_bf16(torch::Tensor a1, torch::Tensor a2,
                          torch::Tensor w1t, torch::Tensor w2t,
                          torch::Tensor bias, torch::Tensor out) {
    CHECK_DEV(a1); CHECK_CONTIG(a1); CHECK_DEV(a2); CHECK_CONTIG(a2);
    CHECK_DEV(w1t); CHECK_CONTIG(w1t); CHECK_DEV(w2t); CHECK_CONTIG(w2t);
    CHECK_DEV(out); CHECK_CONTIG(out);
    TORCH_CHECK(a1.scalar_type() == torch::kBFloat16, "bf16 only");
    const int64_t M = a1.size(0), K1 = a1.size(1), K2 = a2.size(1);
    const int64_t N = out.size(1);
    TORCH_CHECK(a2.size(0) == M && out.size(0) == M, "M mismatch");
    TORCH_CHECK(w1t.size(0) == N && w1t.size(1) == K1, "w1t must be [N,K1]");
    TORCH_CHECK(w2t.size(0) == N && w2t.size(1) == K2, "w2t must be [N,K2]");
    TORCH_CHECK(N % 16 == 0 && N <= 256, "N must be mult of 16, <=256");
    TORCH_CHECK(K1 % 8 == 0 && K2 % 8 == 0, "K must be mult of 8");
    const dim3 block(256), grid((M + 63) / 64);
    fused_dual_gemm_bf16_kernel<<<grid, block, 0, cur_stream()>>>(
        reinterpret_cast<const ushort*>(a1.data_ptr()),
        reinterpret_cast<const ushort*>(a2.data_ptr()),
        reinterpret_cast<const ushort*>(w1t.data_ptr()),
        reinterpret_cast<const ushort*>(w2t.data_ptr()),
        bias.numel() ? reinterpret_cast<const ushort*>(bias.data_ptr()) : nullptr,
        reinterpret_cast<ushort*>(out.data_ptr()), M, N, K1, K2);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("quant_pack", &quant_pack,
          "fused minmax+stochastic-quantize+bitpack (CDNA4)");
    m.def("quant_unpack", &quant_unpack,
          "fused dequantize+scatter (CDNA4)");
    m.def("spmm_csr", &spmm_csr,
          "CSR SpMM with fused degree normalization (CDNA4)");
    m.def("fused_dual_gemm_bf16", &fused_dual_gemm_bf16,
          "out = A1@W1 + A2@W2 + bias, one-pass MFMA (CDNA4)");
}
