"""HBM3E streaming-bandwidth probe: what is the real mixed read+write
roofline?  Variants: scalar 8B/lane, vectorized 16B/lane (double2),
nontemporal, and read-only/write-only, via tiny JIT kernels."""

import sys
import time

import torch

sys.path.insert(0, ".")
from pystella_amd.backend.hip import ext, _stream  # noqa: E402

N = 512**3 * 2          # doubles (2 GiB)

SRC = r"""
typedef double d2 __attribute__((ext_vector_type(2)));
extern "C" __global__ __launch_bounds__(256) void copy8(
    const double* __restrict__ a, double* __restrict__ b, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long s = (long)gridDim.x * blockDim.x;
  for (; i < n; i += s) b[i] = a[i];
}
extern "C" __global__ __launch_bounds__(256) void copy16(
    const d2* __restrict__ a, d2* __restrict__ b, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long s = (long)gridDim.x * blockDim.x;
  for (; i < n; i += s) b[i] = a[i];
}
extern "C" __global__ __launch_bounds__(256) void copy16nt(
    const d2* __restrict__ a, d2* __restrict__ b, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long s = (long)gridDim.x * blockDim.x;
  for (; i < n; i += s)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&a[i]), &b[i]);
}
extern "C" __global__ __launch_bounds__(256) void read16(
    const d2* __restrict__ a, d2* __restrict__ b, long n) {
  d2 acc = {0.0, 0.0};
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long s = (long)gridDim.x * blockDim.x;
  for (; i < n; i += s) acc += __builtin_nontemporal_load(&a[i]);
  if (acc[0] == 1.0e30) b[0] = acc;   // never true; defeats DCE
}
extern "C" __global__ __launch_bounds__(256) void write16(
    const d2* __restrict__ a, d2* __restrict__ b, long n) {
  d2 v = a[0];
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long s = (long)gridDim.x * blockDim.x;
  for (; i < n; i += s) __builtin_nontemporal_store(v, &b[i]);
}
"""


def timeit(fn, n=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    e = ext()
    dev = torch.device("cuda", 0)
    a = torch.rand(N, dtype=torch.float64, device=dev)
    b = torch.empty_like(a)
    tot = N * 8

    ms = timeit(lambda: b.copy_(a)) * 1e3
    print(f"torch copy_        {ms:7.3f} ms  {2*tot/ms/1e9:6.2f} TB/s")

    for name, nelem, bytes_moved in (
            ("copy8", N, 2 * tot), ("copy16", N // 2, 2 * tot),
            ("copy16nt", N // 2, 2 * tot), ("read16", N // 2, tot),
            ("write16", N // 2, tot)):
        key = e.jit_compile(SRC, name)
        for grid in (2048, 4096, 8192, 16384):
            fn = lambda: e.jit_launch(key, grid, 1, 1, 256, 1, 1, 0,
                                      _stream(),
                                      [a.data_ptr(), b.data_ptr()],
                                      [nelem], [])
            ms = timeit(fn) * 1e3
            print(f"{name:10s} grid={grid:6d} {ms:7.3f} ms  "
                  f"{bytes_moved/ms/1e9:6.2f} TB/s")


if __name__ == "__main__":
    main()
