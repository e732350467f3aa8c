"""Design experiment: LDS-staged ghost-cell tile vs register-ring
x-march for the h=2 fp64 Laplacian on gfx950.

The north-star design memo assumes LDS-staged ghost tiles; this
measures both forms of the same Laplacian so the shipped choice
(register ring + L1/L2 neighbor reuse) is evidence-based, not
assumed.  Run on a GPU box; results go to profiles/.
"""

import sys
import time

import torch

sys.path.insert(0, ".")
from pystella_amd.backend.hip import ext, _stream  # noqa: E402

N = 512
H = 2

# LDS variant: (TZ+2h)(TY+2h) tile of the current x-plane staged in
# LDS per i-iteration (classic ghost-cell prefetch); x neighbors from a
# register ring, like the shipped kernel.
LDS_SRC = r"""
#define H 2
#define NX 512
#define NY 512
#define NZ 512
#define PSY ((long)(NY + 2*H))
#define PSZ ((long)(NZ + 2*H))
#define PVOL ((long)(NX + 2*H)*PSY*PSZ)
#define TZ 32
#define TY 8
#define XCHUNK 32
extern "C" __global__ __launch_bounds__(TZ * TY) void lap_lds(
    const double* __restrict__ f, double* __restrict__ lap,
    double c0, double c1, double c2)
{
    __shared__ double tile[TY + 2 * H][TZ + 2 * H];
    const int lz = threadIdx.x % TZ;
    const int ly = threadIdx.x / TZ;
    const int k = blockIdx.x * TZ + lz;
    const int j = blockIdx.y * TY + ly;
    const int i0 = blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < NX) ? i0 + XCHUNK : NX;
    const long sx = PSY * PSZ;
    const double* fp = f + (long)(j + H) * PSZ + (k + H);
    double ring[2 * H + 1];
    #pragma unroll
    for (int p = 0; p < 2 * H; ++p)
        ring[p] = fp[(long)(i0 + p) * sx];
    for (int i = i0; i < i1; ++i) {
        ring[2 * H] = fp[(long)(i + 2 * H) * sx];
        // stage the current x-plane tile (incl. ghost rim) into LDS
        __syncthreads();
        for (int t = threadIdx.x; t < (TY + 2 * H) * (TZ + 2 * H);
             t += TZ * TY) {
            const int tz = t % (TZ + 2 * H);
            const int ty = t / (TZ + 2 * H);
            const int gk = blockIdx.x * TZ + tz;      // padded k idx
            const int gj = blockIdx.y * TY + ty;      // padded j idx
            tile[ty][tz] = f[(long)(i + H) * sx
                             + (long)gj * PSZ + gk];
        }
        __syncthreads();
        if (k < NZ && j < NY) {
            const int ty = ly + H, tz = lz + H;
            double la = ring[H] * c0 * 3.0;
            la += c1 * (ring[H+1] + ring[H-1]
                        + tile[ty+1][tz] + tile[ty-1][tz]
                        + tile[ty][tz+1] + tile[ty][tz-1]);
            la += c2 * (ring[H+2] + ring[H-2]
                        + tile[ty+2][tz] + tile[ty-2][tz]
                        + tile[ty][tz+2] + tile[ty][tz-2]);
            lap[((long)i * NY + j) * NZ + k] = la;
        }
        #pragma unroll
        for (int p = 0; p < 2 * H; ++p)
            ring[p] = ring[p + 1];
    }
}
"""

# register-ring variant (the shipped form, csrc/derivs.hip style)
RING_SRC = LDS_SRC.replace("lap_lds", "lap_ring").replace(
    r"""        // stage the current x-plane tile (incl. ghost rim) into LDS
        __syncthreads();
        for (int t = threadIdx.x; t < (TY + 2 * H) * (TZ + 2 * H);
             t += TZ * TY) {
            const int tz = t % (TZ + 2 * H);
            const int ty = t / (TZ + 2 * H);
            const int gk = blockIdx.x * TZ + tz;      // padded k idx
            const int gj = blockIdx.y * TY + ty;      // padded j idx
            tile[ty][tz] = f[(long)(i + H) * sx
                             + (long)gj * PSZ + gk];
        }
        __syncthreads();
        if (k < NZ && j < NY) {
            const int ty = ly + H, tz = lz + H;
            double la = ring[H] * c0 * 3.0;
            la += c1 * (ring[H+1] + ring[H-1]
                        + tile[ty+1][tz] + tile[ty-1][tz]
                        + tile[ty][tz+1] + tile[ty][tz-1]);
            la += c2 * (ring[H+2] + ring[H-2]
                        + tile[ty+2][tz] + tile[ty-2][tz]
                        + tile[ty][tz+2] + tile[ty][tz-2]);
            lap[((long)i * NY + j) * NZ + k] = la;
        }""",
    r"""        if (k < NZ && j < NY) {
            const double* cp = fp + (long)(i + H) * sx;
            double la = ring[H] * c0 * 3.0;
            la += c1 * (ring[H+1] + ring[H-1]
                        + cp[PSZ] + cp[-PSZ] + cp[1] + cp[-1]);
            la += c2 * (ring[H+2] + ring[H-2]
                        + cp[2*PSZ] + cp[-2*PSZ] + cp[2] + cp[-2]);
            lap[((long)i * NY + j) * NZ + k] = la;
        }""")


def timeit(fn, n=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    e = ext()
    dev = torch.device("cuda", 0)
    pad = (N + 2 * H,) * 3
    f = torch.rand(pad, dtype=torch.float64, device=dev)
    lap = torch.zeros((N, N, N), dtype=torch.float64, device=dev)
    c0, c1, c2 = -30 / 12, 16 / 12, -1 / 12
    gb = (f.numel() + lap.numel()) * 8 / 1e9

    outs = {}
    for name, src in (("lap_lds", LDS_SRC), ("lap_ring", RING_SRC)):
        key = e.jit_compile(src, name)
        grid = ((N + 31) // 32, (N + 7) // 8, (N + 31) // 32)
        fn = lambda: e.jit_launch(key, grid[0], grid[1], grid[2],
                                  256, 1, 1, 0, _stream(),
                                  [f.data_ptr(), lap.data_ptr()], [],
                                  [c0, c1, c2])
        fn()
        torch.cuda.synchronize()
        outs[name] = lap.clone()
        lap.zero_()
        ms = timeit(fn)
        print(f"{name:10s} {ms:7.3f} ms  {gb/ms:5.2f} TB/s "
              f"(~2 passes of 512^3 fp64)")
    err = (outs["lap_lds"] - outs["lap_ring"]).abs().max().item()
    print("max |lds - ring| =", err)
    assert err == 0.0


if __name__ == "__main__":
    main()
