"""On-box diagnostic: phase timings, capture bisection, profiler table."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from npf import CNPFLoss
from npf.zoo import attncnp_1d


def mm_bwd_microbench():
    """Isolated bf16 mm backward on the autograd thread."""
    for dtype in (torch.bfloat16, torch.float32):
        a = torch.randn(4096, 128, device="cuda", dtype=dtype, requires_grad=True)
        w = torch.randn(128, 128, device="cuda", dtype=dtype, requires_grad=True)
        for _ in range(10):
            (a @ w).sum().backward()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(50):
            (a @ w).sum().backward()
        torch.cuda.synchronize()
        print(f"mm+backward {dtype}: {(time.perf_counter()-t0)/50*1000:.3f} ms",
              flush=True)


def main():
    mm_bwd_microbench()
    device = "cuda:0"
    torch.manual_seed(0)
    model = attncnp_1d().to(device)
    crit = CNPFLoss()
    crit.train()
    model.train()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, capturable=True, foreach=True)
    opt_plain = torch.optim.Adam(model.parameters(), lr=1e-3)

    B, C, T = 32, 50, 128
    Xc = torch.rand(B, C, 1, device=device) * 2 - 1
    Yc = torch.randn(B, C, 1, device=device)
    Xt = torch.rand(B, T, 1, device=device) * 2 - 1
    Yt = torch.randn(B, T, 1, device=device)

    def fwd():
        with torch.autocast("cuda", dtype=torch.bfloat16):
            return model(Xc, Yc, Xt, Yt)

    def phase_time(fn, n=50):
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1000

    print(f"forward only: {phase_time(lambda: fwd()):.3f} ms", flush=True)
    print(f"forward+loss: {phase_time(lambda: crit(fwd(), Yt)):.3f} ms", flush=True)

    def fwd_bwd():
        opt.zero_grad(set_to_none=False)
        loss = crit(fwd(), Yt)
        loss.backward()
        return loss

    print(f"fwd+loss+bwd: {phase_time(fwd_bwd):.3f} ms", flush=True)

    def full():
        loss = fwd_bwd()
        opt.step()
        return loss

    print(f"full step (capturable adam): {phase_time(full):.3f} ms", flush=True)

    def full_plain():
        opt_plain.zero_grad(set_to_none=False)
        loss = crit(fwd(), Yt)
        loss.backward()
        opt_plain.step()
        return loss

    print(f"full step (plain adam):      {phase_time(full_plain):.3f} ms", flush=True)

    # no-autocast fp32 comparison
    def full_fp32():
        opt_plain.zero_grad(set_to_none=False)
        loss = crit(model(Xc, Yc, Xt, Yt), Yt)
        loss.backward()
        opt_plain.step()
        return loss

    print(f"full step (fp32, plain adam): {phase_time(full_fp32):.3f} ms", flush=True)

    # ---- capture bisection ----
    import gc

    def try_capture(name, fn):
        gc.collect()
        torch.cuda.synchronize()
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    fn()
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                fn()
            g.replay()
            torch.cuda.synchronize()
            # time replay
            t0 = time.perf_counter()
            for _ in range(100):
                g.replay()
            torch.cuda.synchronize()
            print(f"capture {name}: OK, replay {(time.perf_counter()-t0)/100*1000:.3f} ms",
                  flush=True)
            return g
        except Exception as e:
            print(f"capture {name}: FAIL {type(e).__name__}: {e}", flush=True)
            return None

    try_capture("forward", lambda: fwd())
    try_capture("forward+loss", lambda: crit(fwd(), Yt))
    try_capture("fwd+loss+bwd", fwd_bwd)
    try_capture("full(capturable)", full)

    # ---- profiler: top ops by cuda time for one eager step ----
    from torch.profiler import ProfilerActivity, profile

    for _ in range(5):
        full()
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for _ in range(10):
            full()
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=25),
          flush=True)
    print(prof.key_averages().table(sort_by="self_cpu_time_total", row_limit=15),
          flush=True)


if __name__ == "__main__":
    main()
