"""Bisect the mm host overhead + capture failure: BLAS backend variants."""

import os
import sys
import time

import torch


def bench_mm(dtype, n=200):
    a = torch.randn(4096, 128, device="cuda", dtype=dtype)
    b = torch.randn(128, 128, device="cuda", dtype=dtype)
    bt = b.t().contiguous().t()  # transposed-layout variant
    for _ in range(20):
        a @ b
        a.t() @ a
        a @ bt
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        a @ b
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    for _ in range(n):
        a.t() @ a  # the backward-style (T,N) shape
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    print(f"  mm {dtype}: NN {(t1-t0)/n*1e6:.1f}us  TN {(t2-t1)/n*1e6:.1f}us",
          flush=True)


def try_capture_mm(dtype):
    a = torch.randn(512, 128, device="cuda", dtype=dtype)
    b = torch.randn(128, 128, device="cuda", dtype=dtype)
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                a @ b
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            a @ b
        g.replay()
        torch.cuda.synchronize()
        print(f"  capture mm {dtype}: OK", flush=True)
    except Exception as e:
        print(f"  capture mm {dtype}: FAIL {e}".splitlines()[0], flush=True)


def run_suite(tag):
    print(f"== {tag}", flush=True)
    bench_mm(torch.bfloat16)
    bench_mm(torch.float32)
    try_capture_mm(torch.bfloat16)
    try_capture_mm(torch.float32)


if __name__ == "__main__":
    mode = sys.argv[1] if len(sys.argv) > 1 else "default"
    if mode == "rocblas":
        torch.backends.cuda.preferred_blas_library("cublas")
    elif mode == "ck":
        try:
            torch.backends.cuda.preferred_blas_library("ck")
        except Exception as e:
            print("ck not available:", e)
            sys.exit(0)
    run_suite(f"{mode} (env TORCH_BLAS_PREFER_HIPBLASLT={os.environ.get('TORCH_BLAS_PREFER_HIPBLASLT')})")
