// Hand-written CDNA4 (gfx950 / MI355X) finite-difference stencil kernels.
//
// MI355X-native re-design of the reference's loopy-generated stencil
// kernels (reference: pystella/derivs.py:301-337, pystella/stencil.py:103-141
// "StreamingStencil").  Design:
//
//  * x-marching: each 256-thread block owns a (4 y × 64 z) tile and walks
//    the full x extent, keeping the 2H+1 x-planes needed by the x-stencil
//    in a register ring — the expensive (large-stride) axis never re-reads
//    HBM.  y/z neighbors are read from the current plane; consecutive
//    lanes read consecutive z addresses (fully coalesced), so these hit
//    L1/L2.
//  * fused grad+lap: one pass reads f once and emits all four outputs,
//    reusing each neighbor load for both the first- and second-derivative
//    coefficient (the reference fuses the same way: derivs.py:334-337).
//  * dtype-generic: every kernel is templated over the element type
//    (fp64 and fp32 instantiated), matching the reference's loopy dtype
//    parameter (reference derivs.py:234 takes any dtype); the kernels
//    are HBM-bandwidth bound by design either way.
//
// Centered-difference coefficients of truncation order 2H
// (reference derivs.py:127-131, 160-165; standard published tables).

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

namespace {

template <int H> struct FD;
template <> struct FD<1> {
    __device__ static constexpr double g(int s) {
        constexpr double c[1] = {1. / 2};
        return c[s - 1];
    }
    __device__ static constexpr double l(int s) {
        constexpr double c[2] = {-2., 1.};
        return c[s];
    }
};
template <> struct FD<2> {
    __device__ static constexpr double g(int s) {
        constexpr double c[2] = {8. / 12, -1. / 12};
        return c[s - 1];
    }
    __device__ static constexpr double l(int s) {
        constexpr double c[3] = {-30. / 12, 16. / 12, -1. / 12};
        return c[s];
    }
};
template <> struct FD<3> {
    __device__ static constexpr double g(int s) {
        constexpr double c[3] = {45. / 60, -9. / 60, 1. / 60};
        return c[s - 1];
    }
    __device__ static constexpr double l(int s) {
        constexpr double c[4] = {-490. / 180, 270. / 180, -27. / 180,
                                 2. / 180};
        return c[s];
    }
};
template <> struct FD<4> {
    __device__ static constexpr double g(int s) {
        constexpr double c[4] = {672. / 840, -168. / 840, 32. / 840,
                                 -3. / 840};
        return c[s - 1];
    }
    __device__ static constexpr double l(int s) {
        constexpr double c[5] = {-14350. / 5040, 8064. / 5040,
                                 -1008. / 5040, 128. / 5040, -9. / 5040};
        return c[s];
    }
};

// Block geometry: 64 lanes along z (one wave) x BY rows along y.
constexpr int BZ = 64;
constexpr int BY = 4;

// Fused gradient/Laplacian kernel, x-marching with a register ring.
// f is halo-padded (nx+2H, ny+2H, nz+2H) per field; outputs unpadded.
// Gradient outputs may live inside a packed (nf, 3, nx, ny, nz) "grd"
// array: g_fstride is the per-field stride of each gradient component
// (3*uvol when packed, uvol for standalone component arrays).
template <typename T, int H, bool LAP, bool GRAD>
__global__ __launch_bounds__(BZ * BY) void gradlap_knl(
    const T *__restrict__ f, T *__restrict__ lap,
    T *__restrict__ pdx, T *__restrict__ pdy,
    T *__restrict__ pdz, int64_t g_fstride, int nx, int ny, int nz,
    int nxch, int xchunk,
    double inv_dx, double inv_dy, double inv_dz,
    double inv_dx2, double inv_dy2, double inv_dz2)
{
    const int k = blockIdx.x * BZ + (threadIdx.x % BZ);
    const int j = blockIdx.y * BY + (threadIdx.x / BZ);
    const int fld = blockIdx.z / nxch;
    const int i0 = (blockIdx.z % nxch) * xchunk;
    const int i1 = (i0 + xchunk < nx) ? i0 + xchunk : nx;
    if (k >= nz || j >= ny) return;

    const T ix = (T)inv_dx, iy = (T)inv_dy, iz = (T)inv_dz;
    const T ix2 = (T)inv_dx2, iy2 = (T)inv_dy2, iz2 = (T)inv_dz2;

    const int64_t psz = nz + 2 * H;
    const int64_t psy = ny + 2 * H;
    const int64_t sx = psy * psz;                 // padded x-plane stride
    const int64_t pvol = (nx + 2 * H) * sx;
    const int64_t uvol = (int64_t)nx * ny * nz;

    // pointer to (x=-H, y=j, z=k) of this field
    const T *fp = f + (int64_t)fld * pvol + ((int64_t)(j + H)) * psz
                  + (k + H);
    T *outl = LAP ? lap + (int64_t)fld * uvol + (int64_t)j * nz + k
                  : nullptr;
    T *outx = GRAD ? pdx + fld * g_fstride + (int64_t)j * nz + k
                   : nullptr;
    T *outy = GRAD ? pdy + fld * g_fstride + (int64_t)j * nz + k
                   : nullptr;
    T *outz = GRAD ? pdz + fld * g_fstride + (int64_t)j * nz + k
                   : nullptr;
    const int64_t so = (int64_t)ny * nz;          // unpadded x stride

    // register ring r[p] holds f at x = i - H + p (center plane p = H)
    T r[2 * H + 1];
#pragma unroll
    for (int p = 0; p < 2 * H; ++p) r[p] = fp[(int64_t)(i0 + p) * sx];

    for (int i = i0; i < i1; ++i) {
        r[2 * H] = fp[(int64_t)(i + 2 * H) * sx];
        const T c = r[H];
        const T *cp = fp + (int64_t)(i + H) * sx;  // center plane

        T lap_acc = (T)0, gx = (T)0, gy = (T)0, gz = (T)0;
        if (LAP)
            lap_acc = (T)FD<H>::l(0) * c * (ix2 + iy2 + iz2);

#pragma unroll
        for (int s = 1; s <= H; ++s) {
            const T xm = r[H - s], xp = r[H + s];
            const T ym = cp[-(int64_t)s * psz], yp = cp[(int64_t)s * psz];
            const T zm = cp[-s], zp = cp[s];
            if (LAP)
                lap_acc += (T)FD<H>::l(s) * ((xp + xm) * ix2
                                             + (yp + ym) * iy2
                                             + (zp + zm) * iz2);
            if (GRAD) {
                gx += (T)FD<H>::g(s) * (xp - xm);
                gy += (T)FD<H>::g(s) * (yp - ym);
                gz += (T)FD<H>::g(s) * (zp - zm);
            }
        }
        if (LAP) outl[(int64_t)i * so] = lap_acc;
        if (GRAD) {
            outx[(int64_t)i * so] = gx * ix;
            outy[(int64_t)i * so] = gy * iy;
            outz[(int64_t)i * so] = gz * iz;
        }
        // rotate the ring
#pragma unroll
        for (int p = 0; p < 2 * H; ++p) r[p] = r[p + 1];
    }
}

// LDS-staged variant: the current x-plane tile (including the ±H ghost
// rim) is staged in LDS each i-iteration, so y/z neighbor reads come
// from LDS instead of L1 — measured 17 % faster than plain L1 reuse on
// gfx950 for the isolated Laplacian (profiles/r01_lds_vs_ring.txt).
// The x-axis still marches through the register ring.
constexpr int LBZ = 32;
constexpr int LBY = 8;

template <typename T, int H, bool LAP, bool GRAD>
__global__ __launch_bounds__(LBZ * LBY) void gradlap_lds_knl(
    const T *__restrict__ f, T *__restrict__ lap,
    T *__restrict__ pdx, T *__restrict__ pdy,
    T *__restrict__ pdz, int64_t g_fstride, int nx, int ny, int nz,
    int nxch, int xchunk,
    double inv_dx, double inv_dy, double inv_dz,
    double inv_dx2, double inv_dy2, double inv_dz2)
{
    __shared__ T tile[LBY + 2 * H][LBZ + 2 * H];
    const int lz = threadIdx.x % LBZ;
    const int ly = threadIdx.x / LBZ;
    const int k = blockIdx.x * LBZ + lz;
    const int j = blockIdx.y * LBY + ly;
    const int fld = blockIdx.z / nxch;
    const int i0 = (blockIdx.z % nxch) * xchunk;
    const int i1 = (i0 + xchunk < nx) ? i0 + xchunk : nx;
    const bool active = (k < nz) && (j < ny);

    const T ix = (T)inv_dx, iy = (T)inv_dy, iz = (T)inv_dz;
    const T ix2 = (T)inv_dx2, iy2 = (T)inv_dy2, iz2 = (T)inv_dz2;

    const int64_t psz = nz + 2 * H;
    const int64_t psy = ny + 2 * H;
    const int64_t sx = psy * psz;
    const int64_t pvol = (nx + 2 * H) * sx;
    const int64_t uvol = (int64_t)nx * ny * nz;

    const int jc = j < ny ? j : ny - 1;     // clamped (inactive lanes
    const int kc = k < nz ? k : nz - 1;     // still help staging)
    const T *fbase = f + (int64_t)fld * pvol;
    const T *fp = fbase + ((int64_t)(jc + H)) * psz + (kc + H);
    T *outl = LAP ? lap + (int64_t)fld * uvol + (int64_t)jc * nz + kc
                  : nullptr;
    T *outx = GRAD ? pdx + fld * g_fstride + (int64_t)jc * nz + kc
                   : nullptr;
    T *outy = GRAD ? pdy + fld * g_fstride + (int64_t)jc * nz + kc
                   : nullptr;
    T *outz = GRAD ? pdz + fld * g_fstride + (int64_t)jc * nz + kc
                   : nullptr;
    const int64_t so = (int64_t)ny * nz;

    T r[2 * H + 1];
#pragma unroll
    for (int p = 0; p < 2 * H; ++p) r[p] = fp[(int64_t)(i0 + p) * sx];

    for (int i = i0; i < i1; ++i) {
        r[2 * H] = fp[(int64_t)(i + 2 * H) * sx];
        __syncthreads();
        for (int t = threadIdx.x; t < (LBY + 2 * H) * (LBZ + 2 * H);
             t += LBZ * LBY) {
            const int tz = t % (LBZ + 2 * H);
            const int ty = t / (LBZ + 2 * H);
            int gj = blockIdx.y * LBY + ty;        // padded y index
            int gk = blockIdx.x * LBZ + tz;        // padded z index
            if (gj > ny + 2 * H - 1) gj = ny + 2 * H - 1;
            if (gk > nz + 2 * H - 1) gk = nz + 2 * H - 1;
            tile[ty][tz] = fbase[(int64_t)(i + H) * sx
                                 + (int64_t)gj * psz + gk];
        }
        __syncthreads();
        if (active) {
            const int ty = ly + H, tz = lz + H;
            const T c = r[H];
            T lap_acc = (T)0, gx = (T)0, gy = (T)0, gz = (T)0;
            if (LAP)
                lap_acc = (T)FD<H>::l(0) * c * (ix2 + iy2 + iz2);
#pragma unroll
            for (int s = 1; s <= H; ++s) {
                const T xm = r[H - s], xp = r[H + s];
                const T ym = tile[ty - s][tz];
                const T yp = tile[ty + s][tz];
                const T zm = tile[ty][tz - s];
                const T zp = tile[ty][tz + s];
                if (LAP)
                    lap_acc += (T)FD<H>::l(s) * ((xp + xm) * ix2
                                                 + (yp + ym) * iy2
                                                 + (zp + zm) * iz2);
                if (GRAD) {
                    gx += (T)FD<H>::g(s) * (xp - xm);
                    gy += (T)FD<H>::g(s) * (yp - ym);
                    gz += (T)FD<H>::g(s) * (zp - zm);
                }
            }
            if (LAP) outl[(int64_t)i * so] = lap_acc;
            if (GRAD) {
                outx[(int64_t)i * so] = gx * ix;
                outy[(int64_t)i * so] = gy * iy;
                outz[(int64_t)i * so] = gz * iz;
            }
        }
#pragma unroll
        for (int p = 0; p < 2 * H; ++p) r[p] = r[p + 1];
    }
}

// Single-axis first derivative (optionally accumulating, for divergence).
// AXIS: 0=x, 1=y, 2=z.
template <typename T, int H, int AXIS, bool ACCUM>
__global__ __launch_bounds__(BZ * BY) void pd_knl(
    const T *__restrict__ f, T *__restrict__ out,
    int nx, int ny, int nz, int nxch, int xchunk, double inv_d)
{
    const int k = blockIdx.x * BZ + (threadIdx.x % BZ);
    const int j = blockIdx.y * BY + (threadIdx.x / BZ);
    const int fld = blockIdx.z / nxch;
    const int i0 = (blockIdx.z % nxch) * xchunk;
    const int i1 = (i0 + xchunk < nx) ? i0 + xchunk : nx;
    if (k >= nz || j >= ny) return;

    const T inv = (T)inv_d;
    const int64_t psz = nz + 2 * H;
    const int64_t psy = ny + 2 * H;
    const int64_t sx = psy * psz;
    const int64_t pvol = (nx + 2 * H) * sx;
    const int64_t uvol = (int64_t)nx * ny * nz;
    const int64_t nstride = (AXIS == 0) ? sx : (AXIS == 1) ? psz : 1;

    const T *fp = f + (int64_t)fld * pvol + ((int64_t)(j + H)) * psz
                  + (k + H);
    T *op = out + (int64_t)fld * uvol + (int64_t)j * nz + k;
    const int64_t so = (int64_t)ny * nz;

    for (int i = i0; i < i1; ++i) {
        const T *cp = fp + (int64_t)(i + H) * sx;
        T g = (T)0;
#pragma unroll
        for (int s = 1; s <= H; ++s)
            g += (T)FD<H>::g(s) * (cp[(int64_t)s * nstride]
                                   - cp[-(int64_t)s * nstride]);
        g *= inv;
        if (ACCUM)
            op[(int64_t)i * so] += g;
        else
            op[(int64_t)i * so] = g;
    }
}

// XCHUNK: split the x-march into chunks for more resident blocks
// (latency hiding); tunable via PYSTELLA_XCHUNK host-side.
inline int xchunk_size(int nx)
{
    const char *env = getenv("PYSTELLA_XCHUNK");
    int c = env ? atoi(env) : 32;
    if (c <= 0 || c > nx) c = nx;
    return c;
}

inline dim3 tile_grid(int ny, int nz, int nf, int nxch)
{
    return dim3((nz + BZ - 1) / BZ, (ny + BY - 1) / BY, nf * nxch);
}

template <typename T>
int gradlap_t(const void *f, void *lap, void *pdx, void *pdy, void *pdz,
              long long g_fstride, int h, int nx, int ny, int nz, int nf,
              double dx, double dy, double dz, void *stream_)
{
    hipStream_t stream = (hipStream_t)stream_;
    const int xchunk = xchunk_size(nx);
    const int nxch = (nx + xchunk - 1) / xchunk;
    const double ix = 1. / dx, iy = 1. / dy, iz = 1. / dz;
    const double ix2 = ix * ix, iy2 = iy * iy, iz2 = iz * iz;
    const bool do_lap = lap != nullptr;
    const bool do_grad = pdx != nullptr;

    // LDS-staged form by default (see gradlap_lds_knl note); set
    // PYSTELLA_LDS=0 for the plain-L1 form.
    const char *env = getenv("PYSTELLA_LDS");
    const bool use_lds = !(env && atoi(env) == 0);

#define ARGS (const T *)f, (T *)lap, (T *)pdx, (T *)pdy, (T *)pdz,      \
             (int64_t)g_fstride, nx, ny, nz, nxch, xchunk, ix, iy, iz,  \
             ix2, iy2, iz2
#define DISPATCH_H(H_, ...)                                              \
    switch (H_) {                                                        \
    case 1: { constexpr int H = 1; __VA_ARGS__; break; }                 \
    case 2: { constexpr int H = 2; __VA_ARGS__; break; }                 \
    case 3: { constexpr int H = 3; __VA_ARGS__; break; }                 \
    case 4: { constexpr int H = 4; __VA_ARGS__; break; }                 \
    default: return 1;                                                   \
    }

    if (use_lds) {
        const dim3 grid((nz + LBZ - 1) / LBZ, (ny + LBY - 1) / LBY,
                        nf * nxch);
        const dim3 block(LBZ * LBY);
        DISPATCH_H(h, {
            if (do_lap && do_grad)
                hipLaunchKernelGGL((gradlap_lds_knl<T, H, true, true>),
                                   grid, block, 0, stream, ARGS);
            else if (do_lap)
                hipLaunchKernelGGL((gradlap_lds_knl<T, H, true, false>),
                                   grid, block, 0, stream, ARGS);
            else
                hipLaunchKernelGGL((gradlap_lds_knl<T, H, false, true>),
                                   grid, block, 0, stream, ARGS);
        });
        return (int)hipGetLastError();
    }

    const dim3 grid = tile_grid(ny, nz, nf, nxch);
    const dim3 block(BZ * BY);
    DISPATCH_H(h, {
        if (do_lap && do_grad)
            hipLaunchKernelGGL((gradlap_knl<T, H, true, true>), grid,
                               block, 0, stream, ARGS);
        else if (do_lap)
            hipLaunchKernelGGL((gradlap_knl<T, H, true, false>), grid,
                               block, 0, stream, ARGS);
        else
            hipLaunchKernelGGL((gradlap_knl<T, H, false, true>), grid,
                               block, 0, stream, ARGS);
    });
    return (int)hipGetLastError();
#undef ARGS
}

template <typename T>
int pd_t(const void *f, void *out, int h, int axis, int accum,
         int nx, int ny, int nz, int nf, double d, void *stream_)
{
    hipStream_t stream = (hipStream_t)stream_;
    const int xchunk = xchunk_size(nx);
    const int nxch = (nx + xchunk - 1) / xchunk;
    const dim3 grid = tile_grid(ny, nz, nf, nxch);
    const dim3 block(BZ * BY);
    const double inv = 1. / d;

#define PDARGS (const T *)f, (T *)out, nx, ny, nz, nxch, xchunk, inv
    DISPATCH_H(h, {
        switch (axis * 2 + (accum ? 1 : 0)) {
        case 0: hipLaunchKernelGGL((pd_knl<T, H, 0, false>), grid, block,
                                   0, stream, PDARGS); break;
        case 1: hipLaunchKernelGGL((pd_knl<T, H, 0, true>), grid, block,
                                   0, stream, PDARGS); break;
        case 2: hipLaunchKernelGGL((pd_knl<T, H, 1, false>), grid, block,
                                   0, stream, PDARGS); break;
        case 3: hipLaunchKernelGGL((pd_knl<T, H, 1, true>), grid, block,
                                   0, stream, PDARGS); break;
        case 4: hipLaunchKernelGGL((pd_knl<T, H, 2, false>), grid, block,
                                   0, stream, PDARGS); break;
        case 5: hipLaunchKernelGGL((pd_knl<T, H, 2, true>), grid, block,
                                   0, stream, PDARGS); break;
        }
    });
    return (int)hipGetLastError();
#undef PDARGS
}

}  // namespace

// dtype: 0 = fp64, 1 = fp32 (matches backend/hip.py derivs()).
extern "C" int pystella_gradlap(
    const void *f, void *lap, void *pdx, void *pdy, void *pdz,
    long long g_fstride, int h, int nx, int ny, int nz, int nf,
    double dx, double dy, double dz, int dtype, void *stream_)
{
    if (dtype == 0)
        return gradlap_t<double>(f, lap, pdx, pdy, pdz, g_fstride, h,
                                 nx, ny, nz, nf, dx, dy, dz, stream_);
    if (dtype == 1)
        return gradlap_t<float>(f, lap, pdx, pdy, pdz, g_fstride, h,
                                nx, ny, nz, nf, dx, dy, dz, stream_);
    return 2;
}

extern "C" int pystella_pd(
    const void *f, void *out, int h, int axis, int accum,
    int nx, int ny, int nz, int nf, double d, int dtype, void *stream_)
{
    if (dtype == 0)
        return pd_t<double>(f, out, h, axis, accum, nx, ny, nz, nf, d,
                            stream_);
    if (dtype == 1)
        return pd_t<float>(f, out, h, axis, accum, nx, ny, nz, nf, d,
                           stream_);
    return 2;
}
