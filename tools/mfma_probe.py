"""Empirical layout probe for __builtin_amdgcn_mfma_f64_4x4x4f64 on
gfx950: loads A/B with lane-coded values, prints which (i,j,k,block)
each lane's accumulator corresponds to.  Run on a GPU box."""

import sys

import torch

sys.path.insert(0, ".")
from pystella_amd.backend.hip import ext, _stream  # noqa: E402

SRC = r"""
extern "C" __global__ void probe(const double* A, const double* B,
                                 double* C, double* D) {
    int l = threadIdx.x;
    double a = A[l];
    double b = B[l];
    double acc = 0.0;
    acc = __builtin_amdgcn_mfma_f64_4x4x4f64(a, b, acc, 0, 0, 0);
    C[l] = acc;
    // identity-A pass: a = 1 where lane's A element is on the diagonal
    // is not known yet, so D gets a*b probe with b=1
    D[l] = a;
}
"""


def main():
    e = ext()
    key = e.jit_compile(SRC, "probe")
    dev = torch.device("cuda", 0)
    # encode A[lane] = 100 + lane, B[lane] = 1: C tells which A-elements
    # a lane's acc sums (row picks A row etc.)
    for tag, amode, bmode in (("A=lane,B=1", "lane", "one"),
                              ("A=1,B=lane", "one", "lane"),
                              ("A=e0,B=lane", "e0", "lane")):
        A = torch.ones(64, dtype=torch.float64, device=dev)
        B = torch.ones(64, dtype=torch.float64, device=dev)
        if amode == "lane":
            A = torch.arange(64, dtype=torch.float64, device=dev) + 100
        if amode == "e0":
            A = torch.zeros(64, dtype=torch.float64, device=dev)
            A[0] = 1
        if bmode == "lane":
            B = torch.arange(64, dtype=torch.float64, device=dev) + 100
        C = torch.zeros(64, dtype=torch.float64, device=dev)
        D = torch.zeros(64, dtype=torch.float64, device=dev)
        e.jit_launch(key, 1, 1, 1, 64, 1, 1, 0, _stream(),
                     [A.data_ptr(), B.data_ptr(), C.data_ptr(),
                      D.data_ptr()], [], [])
        torch.cuda.synchronize()
        print(tag)
        print(C.cpu().numpy().reshape(4, 16))


if __name__ == "__main__":
    main()
