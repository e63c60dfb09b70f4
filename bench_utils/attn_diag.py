"""Per-quantity error report for the MFMA attention path."""
import sys, torch
sys.path.insert(0, "/root/repo")
from npf.ops import functional as F_ops

def _oracle(k, q, v, scale):
    logits = torch.einsum("bkd,bqd->bqk", k, q) * scale
    return torch.bmm(logits.softmax(-1), v)

def run(B, K, Q, D):
    g = torch.Generator(device="cuda").manual_seed(0)
    k = torch.randn(B, K, D, device="cuda", generator=g, requires_grad=True)
    q = torch.randn(B, Q, D, device="cuda", generator=g, requires_grad=True)
    v = torch.randn(B, K, D, device="cuda", generator=g, requires_grad=True)
    scale = D ** -0.5
    out = F_ops.attention_qkv(k, q, v, scale)
    k0 = k.detach().cpu().requires_grad_(True)
    q0 = q.detach().cpu().requires_grad_(True)
    v0 = v.detach().cpu().requires_grad_(True)
    ref = _oracle(k0, q0, v0, scale)
    e_out = (out.cpu() - ref).abs().max()
    dout = torch.randn_like(out)
    out.backward(dout)
    ref.backward(dout.cpu())
    e_dk = (k.grad.cpu() - k0.grad).abs().max()
    e_dq = (q.grad.cpu() - q0.grad).abs().max()
    e_dv = (v.grad.cpu() - v0.grad).abs().max()
    print(f"B={B} K={K} Q={Q}: out={float(e_out):.2e} dq={float(e_dq):.2e} "
          f"dk={float(e_dk):.2e} dv={float(e_dv):.2e}", flush=True)

for shape in [(8, 96, 64, 16), (4, 130, 50, 16), (16, 307, 307, 16),
              (256, 300, 1024, 16), (8, 1024, 1024, 16)]:
    run(*shape)
