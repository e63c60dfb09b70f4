"""Time convblock2d fwd/bwd in isolation across shapes — is the cost
per-element (bandwidth/compute) or per-workgroup (occupancy/latency)?"""
import sys, time, torch
sys.path.insert(0, "/root/repo")
import npf._hip_C as ext

def t(fn, iters=10):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

und = torch.Tensor()
for (N, C, H, W) in [(256, 128, 64, 64), (64, 128, 64, 64), (256, 128, 32, 32),
                     (256, 32, 64, 64), (16, 128, 32, 32)]:
    for dt in (torch.bfloat16, torch.float32):
        K = 9
        x = torch.randn(N, C, H, W, device="cuda", dtype=dt)
        dy = torch.randn_like(x)
        w = torch.randn(C, K, K, device="cuda")
        gamma = torch.rand(C, device="cuda") + 0.5
        beta = torch.zeros(C, device="cuda")
        mean = torch.zeros(C, device="cuda")
        rstd = torch.ones(C, device="cuda")
        ms_f = t(lambda: ext.convblock2d_fwd(x, und, w, und, gamma, beta, mean, rstd))
        ms_b = t(lambda: ext.convblock2d_bwd(x, w, dy, gamma, beta, mean, rstd, False, True))
        gb = N*C*H*W*x.element_size()/1e9
        print(f"[{N},{C},{H},{W}] {str(dt)[6:]}: fwd {ms_f:.2f}ms bwd {ms_b:.2f}ms"
              f" ({gb:.2f} GB/tensor; bwd eff {3*gb/ms_b*1e3:.0f} GB/s)", flush=True)
