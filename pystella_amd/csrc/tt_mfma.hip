// Transverse-traceless projection of a symmetric rank-2 tensor field in
// k-space, on the CDNA4 matrix cores (__builtin_amdgcn_mfma_f64_4x4x4f64).
//
//   h_TT_ab = (P_ac P_db - 1/2 P_ab P_cd) h_cd,   P = I - khat khat^T
//
// computed per k-site as two 3x3 matrix products PHP = P.(H.P) plus a
// trace correction (reference pystella/fourier/projectors.py:198-216,
// tt_knl).  The f64 MFMA's four independent 4x4 blocks process four
// k-sites per instruction, one 3x3 (zero-padded to 4x4) per block, for
// the real and imaginary parts separately (P is real).
//
// Empirically verified fragment layout (tools/mfma_probe.py, gfx950):
//   A(block,i,k) at lane 16k + 4 block + i
//   B(block,k,j) at lane 16k + 4 block + j
//   C(block,i,j) at lane 16i + 4 block + j
// Note C's output layout equals B's input layout (i <-> k position), so
// the second product P.(H.P) consumes the first MFMA's result with NO
// cross-lane shuffle.

#include <hip/hip_runtime.h>

namespace {

// symmetric-pair component index, 0-based (sectors.tensor_index)
__device__ __forceinline__ int sym_idx(int a, int b)
{
    if (a > b) { int t = a; a = b; b = t; }
    // (0,0)->0 (0,1)->1 (0,2)->2 (1,1)->3 (1,2)->4 (2,2)->5
    return a * 3 - (a * (a + 1)) / 2 + b;
}

__global__ __launch_bounds__(256) void tt_project_mfma(
    const double *__restrict__ hij,   // [6][vol] complex128 interleaved
    double *__restrict__ out,         // [6][vol] complex128 (may == hij)
    const double *__restrict__ kx, const double *__restrict__ ky,
    const double *__restrict__ kz,
    int ny, int nz, long vol)
{
    const int lane = threadIdx.x & 63;
    // 2-D grid: HSA grid_size per dimension is uint32 WORK-ITEMS, so a
    // 1-D launch overflows for k-volumes above ~2.7e8 sites (>= 1024^3
    // r2c); blockIdx.y carries the high part.
    const long block = (long)blockIdx.y * gridDim.x + blockIdx.x;
    const long wave = (block * blockDim.x + threadIdx.x) >> 6;
    const int blk = (lane & 15) >> 2;
    long site = wave * 4 + blk;
    const bool active = site < vol;
    if (!active) site = vol - 1;

    const int iz = (int)(site % nz);
    const long t = site / nz;
    const int iy = (int)(t % ny);
    const int ix = (int)(t / ny);
    const double kxv = kx[ix], kyv = ky[iy], kzv = kz[iz];
    const double ksq = kxv * kxv + kyv * kyv + kzv * kzv;
    const bool kzero = (ksq == 0.0);
    const double kinv = kzero ? 0.0 : rsqrt(ksq);
    const double kh[3] = {kxv * kinv, kyv * kinv, kzv * kinv};

#define PEL(a, c) (((a) == (c) ? 1.0 : 0.0) - kh[a] * kh[c])

    // operand elements for this lane
    const int ai = lane & 3;          // A row
    const int ak = lane >> 4;         // A col (= B row)
    const int bj = lane & 3;          // B col
    double h_re = 0.0, h_im = 0.0;    // A = H for M1 = H.P
    if (ai < 3 && ak < 3) {
        typedef double d2 __attribute__((ext_vector_type(2)));
        const d2 *p = (const d2 *)(hij
            + ((long)sym_idx(ai, ak) * vol + site) * 2);
        const d2 v = __builtin_nontemporal_load(p);
        h_re = v.x;
        h_im = v.y;
    }
    const double b_p = (ak < 3 && bj < 3) ? PEL(ak, bj) : 0.0;
    const double a_p = (ai < 3 && ak < 3) ? PEL(ai, ak) : 0.0;

    // M1 = H.P  (C layout == B layout, so M1 feeds straight into M2)
    double m1_re = __builtin_amdgcn_mfma_f64_4x4x4f64(h_re, b_p, 0.0,
                                                      0, 0, 0);
    double m1_im = __builtin_amdgcn_mfma_f64_4x4x4f64(h_im, b_p, 0.0,
                                                      0, 0, 0);
    // M2 = P.(H.P) = P H P
    double m2_re = __builtin_amdgcn_mfma_f64_4x4x4f64(a_p, m1_re, 0.0,
                                                      0, 0, 0);
    double m2_im = __builtin_amdgcn_mfma_f64_4x4x4f64(a_p, m1_im, 0.0,
                                                      0, 0, 0);

    // tr(P H) = tr(H P) = sum_d M1(blk, d, d), M1(b,i,j) @ 16i + 4b + j
    double tr_re = 0.0, tr_im = 0.0;
#pragma unroll
    for (int d = 0; d < 3; ++d) {
        tr_re += __shfl(m1_re, 16 * d + 4 * blk + d, 64);
        tr_im += __shfl(m1_im, 16 * d + 4 * blk + d, 64);
    }

    // store: C(blk, i, j) at lane 16 i + 4 blk + j; write i <= j < 3
    const int ci = lane >> 4;
    const int cj = lane & 3;
    if (active && ci < 3 && cj < 3 && ci <= cj) {
        const double pab = PEL(ci, cj);
        double re = m2_re - 0.5 * pab * tr_re;
        double im = m2_im - 0.5 * pab * tr_im;
        if (kzero) { re = 0.0; im = 0.0; }
        typedef double d2 __attribute__((ext_vector_type(2)));
        d2 v;
        v.x = re;
        v.y = im;
        __builtin_nontemporal_store(
            v, (d2 *)(out + ((long)sym_idx(ci, cj) * vol + site) * 2));
    }
#undef PEL
}

}  // namespace

extern "C" void launch_tt_project_mfma(
    const double *hij, double *out, const double *kx, const double *ky,
    const double *kz, int ny, int nz, long vol, hipStream_t stream)
{
    const long waves = (vol + 3) / 4;
    const long threads = waves * 64;
    const int block = 256;
    const long nblocks = (threads + block - 1) / block;
    const long gx = nblocks < 32768 ? nblocks : 32768;
    const long gy = (nblocks + gx - 1) / gx;
    hipLaunchKernelGGL(tt_project_mfma, dim3((uint32_t)gx, (uint32_t)gy),
                       dim3(block), 0, stream, hij, out, kx, ky, kz, ny,
                       nz, vol);
}
