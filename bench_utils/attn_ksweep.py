"""Exhaustive K sweep of the attention kernels at the 1D training shapes:
B=256 (32 tasks x 8 heads), Q=128, D=16, K=1..50, fp32 and bf16,
forward+backward vs the composed oracle."""
import sys

sys.path.insert(0, "/root/repo")
import torch

from npf.ops.functional import _attention_ref, attention_qkv

torch.manual_seed(0)
bad = 0
for dtype, atol_o, atol_g in [(torch.float32, 5e-4, 5e-3), (torch.bfloat16, 5e-2, 1e-1)]:
    for K in range(1, 51):
        B, Q, D = 256, 128, 16
        k = torch.randn(B, K, D, device="cuda", dtype=dtype, requires_grad=True)
        q = torch.randn(B, Q, D, device="cuda", dtype=dtype, requires_grad=True)
        v = torch.randn(B, K, D, device="cuda", dtype=dtype, requires_grad=True)
        out = attention_qkv(k, q, v)
        g = torch.randn_like(out)
        out.backward(g)
        k0 = k.detach().float().cpu().requires_grad_()
        q0 = q.detach().float().cpu().requires_grad_()
        v0 = v.detach().float().cpu().requires_grad_()
        o0 = _attention_ref(k0, q0, v0, 1.0 / D ** 0.5)
        o0.backward(g.float().cpu())
        eo = float((out.float().cpu() - o0).abs().max())
        ek = float((k.grad.float().cpu() - k0.grad).abs().max())
        eq = float((q.grad.float().cpu() - q0.grad).abs().max())
        ev = float((v.grad.float().cpu() - v0.grad).abs().max())
        if eo > atol_o or ek > atol_g or eq > atol_g or ev > atol_g:
            print(f"BAD {dtype} K={K}: out={eo:.4g} dk={ek:.4g} dq={eq:.4g} dv={ev:.4g}",
                  flush=True)
            bad += 1
print("DONE, bad:", bad, flush=True)
