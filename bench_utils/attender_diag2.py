"""Localize the module-level dK failure: compare each stage's backward on
identical inputs (fused vs composed), exact failing shapes."""
import sys

sys.path.insert(0, "/root/repo")
import torch

torch.manual_seed(3)


def rel(a, b):
    return float((a.float().cpu() - b.float().cpu()).abs().max() / (b.abs().max() + 1e-6))


def attention_bwd_unit():
    """Fused small-K attention vs composed, exact module shapes."""
    from npf.ops.functional import _attention_ref, attention_qkv

    HB, K, Q, D = 16, 13, 37, 16
    for dtype in (torch.bfloat16, torch.float32):
        kh = torch.randn(HB, K, D, device="cuda", dtype=dtype, requires_grad=True)
        qh = torch.randn(HB, Q, D, device="cuda", dtype=dtype, requires_grad=True)
        vh = torch.randn(HB, K, D, device="cuda", dtype=dtype, requires_grad=True)
        out = attention_qkv(kh, qh, vh)
        g = torch.randn_like(out)
        out.backward(g)
        k0 = kh.detach().clone().requires_grad_()
        q0 = qh.detach().clone().requires_grad_()
        v0 = vh.detach().clone().requires_grad_()
        out0 = _attention_ref(k0, q0, v0, 1.0 / D ** 0.5)
        out0.backward(g)
        print(f"attn {dtype}: out={rel(out, out0.detach()):.4g} "
              f"dk={rel(kh.grad, k0.grad):.4g} dq={rel(qh.grad, q0.grad):.4g} "
              f"dv={rel(vh.grad, v0.grad):.4g}", flush=True)


def module_grads():
    from npf.architectures.attention import TransformerAttender

    torch.manual_seed(0)
    m = TransformerAttender(128, 128, 128).cuda()
    g = torch.Generator().manual_seed(3)
    keys = torch.randn(2, 13, 128, generator=g).cuda().requires_grad_()
    queries = torch.randn(2, 37, 128, generator=g).cuda().requires_grad_()
    values = torch.randn(2, 13, 128, generator=g).cuda().requires_grad_()
    out = m(keys, queries, values)
    out.square().sum().backward()
    gk_f = keys.grad.clone()
    gq_f = queries.grad.clone()
    gv_f = values.grad.clone()
    print("fused dk stats:", float(gk_f.abs().max()), float(gk_f.abs().mean()), flush=True)

    # composed on GPU: disable the fused block
    keys.grad = queries.grad = values.grad = None
    m._fused_block_ok = lambda *a: False
    out2 = m(keys, queries, values)
    out2.square().sum().backward()
    print("composed-gpu dk stats:", float(keys.grad.abs().max()), flush=True)
    print(f"module: out={rel(out, out2.detach()):.4g} dk={rel(gk_f, keys.grad):.4g} "
          f"dq={rel(gq_f, queries.grad):.4g} dv={rel(gv_f, values.grad):.4g}", flush=True)


if __name__ == "__main__":
    attention_bwd_unit()
    module_grads()
