"""Localize the fused-attender dK bug + the graphed-step step-2 divergence."""
import sys

sys.path.insert(0, "/root/repo")
import torch

torch.manual_seed(0)


def check(name, a, b, tol=2e-2):
    rel = (a.float().cpu() - b.float().cpu()).abs().max() / (b.abs().max() + 1e-6)
    print(f"{name}: rel={float(rel):.4g} {'OK' if rel < tol else 'FAIL'}", flush=True)
    return float(rel)


def qkv_unit():
    from npf.ops.functional import qkv_project_headsplit

    B, Kn, D, H = 2, 13, 128, 8
    x = torch.randn(B, Kn, D, device="cuda", requires_grad=True)
    w = torch.randn(D, D, device="cuda") * 0.05
    w.requires_grad_()
    (out,) = qkv_project_headsplit([x], [w], [None], H)
    # oracle: torch linear + head split
    x0 = x.detach().clone().requires_grad_()
    w0 = w.detach().clone().requires_grad_()
    proj = torch.nn.functional.linear(x0.to(torch.bfloat16), w0.to(torch.bfloat16))
    hs = proj.view(B, Kn, H, 16).permute(2, 0, 1, 3).reshape(H * B, Kn, 16)
    check("qkv fwd", out, hs.detach())
    g = torch.randn_like(out.float()).to(torch.bfloat16)
    out.backward(g)
    hs.backward(g)
    check("qkv dx", x.grad, x0.grad)
    check("qkv dw", w.grad, w0.grad)

    # 3-problem variant with bias on problem 1
    xs = [torch.randn(B, n, D, device="cuda", requires_grad=True) for n in (13, 37, 13)]
    ws = [(torch.randn(D, D, device="cuda") * 0.05).requires_grad_() for _ in range(3)]
    bias = torch.randn(D, device="cuda", requires_grad=True)
    outs = qkv_project_headsplit(xs, ws, [None, bias, None], H)
    o0 = []
    xs0 = [x.detach().clone().requires_grad_() for x in xs]
    ws0 = [w.detach().clone().requires_grad_() for w in ws]
    b0 = bias.detach().clone().requires_grad_()
    for i in range(3):
        p = torch.nn.functional.linear(
            xs0[i].to(torch.bfloat16), ws0[i].to(torch.bfloat16),
            b0.to(torch.bfloat16) if i == 1 else None,
        )
        n = p.shape[1]
        o0.append(p.view(B, n, H, 16).permute(2, 0, 1, 3).reshape(H * B, n, 16))
    for i in range(3):
        check(f"qkv3 fwd[{i}]", outs[i], o0[i].detach())
    gs = [torch.randn_like(o.float()).to(torch.bfloat16) for o in outs]
    torch.autograd.backward(outs, gs)
    torch.autograd.backward(o0, gs)
    for i in range(3):
        check(f"qkv3 dx[{i}]", xs[i].grad, xs0[i].grad)
        check(f"qkv3 dw[{i}]", ws[i].grad, ws0[i].grad)
    check("qkv3 db", bias.grad, b0.grad)


def add_ln_unit():
    from npf.ops.functional import add_layernorm

    B, N, D, H = 3, 37, 128, 8
    gamma = torch.randn(D, device="cuda", requires_grad=True)
    beta = torch.randn(D, device="cuda", requires_grad=True)
    # head-split a
    a = torch.randn(H * B, N, 16, device="cuda", requires_grad=True)
    b = torch.randn(B, N, D, device="cuda", requires_grad=True)
    y = add_layernorm(a, b, gamma, beta, headsplit=(B, N, H))
    a0 = a.detach().clone().requires_grad_()
    b0 = b.detach().clone().requires_grad_()
    g0 = gamma.detach().clone().requires_grad_()
    be0 = beta.detach().clone().requires_grad_()
    merged = a0.view(H, B, N, 16).permute(1, 2, 0, 3).reshape(B, N, D)
    y0 = torch.nn.functional.layer_norm(
        (merged.to(torch.bfloat16).float() + b0.to(torch.bfloat16).float()),
        (D,), g0, be0,
    )
    check("addln fwd", y, y0.detach())
    g = torch.randn(B, N, D, device="cuda")
    y.backward(g.to(torch.bfloat16))
    y0.backward(g)
    check("addln da", a.grad, a0.grad, tol=5e-2)
    check("addln db", b.grad, b0.grad, tol=5e-2)
    check("addln dgamma", gamma.grad, g0.grad, tol=5e-2)
    check("addln dbeta", beta.grad, be0.grad, tol=5e-2)


def graph_probe():
    """Param-checksum trace: find where graphed and eager diverge."""
    from npf import CNPFLoss
    from npf.train.trainer import NPFTrainer
    from npf.zoo import attncnp_1d

    def mk(h):
        torch.manual_seed(0)
        return NPFTrainer(
            attncnp_1d(), CNPFLoss(), device="cuda", batch_size=8,
            amp_dtype=torch.bfloat16, hipgraphs=h, seed=0,
        )

    def eps(n):
        g = torch.Generator().manual_seed(7)
        out = []
        for i in range(n):
            n_c = [5, 5, 9, 9, 5, 9, 13, 13][i % 8]
            out.append((
                dict(
                    X_cntxt=(torch.rand(8, n_c, 1, generator=g) * 2 - 1).cuda(),
                    Y_cntxt=torch.randn(8, n_c, 1, generator=g).cuda(),
                    X_trgt=(torch.rand(8, 64, 1, generator=g) * 2 - 1).cuda(),
                    Y_trgt=torch.randn(8, 64, 1, generator=g).cuda(),
                ),
                torch.randn(8, 64, 1).cuda(),
            ))
        # fix Y_trgt to match inputs
        for inputs, y in out:
            y.copy_(inputs["Y_trgt"])
        return out

    episodes = eps(8)
    t_e, t_g = mk(False), mk(True)
    t_g.module.load_state_dict(t_e.module.state_dict())

    def ck(t):
        return float(
            torch.cat([p.detach().double().reshape(-1) for p in t.module.parameters()]).sum()
        )

    for i, (inputs, y) in enumerate(episodes):
        le = float(t_e.train_step(inputs, y, _first=i == 0))
        lg = float(t_g.train_step(inputs, y, _first=i == 0))
        torch.cuda.synchronize()
        print(f"step {i}: loss_e={le:.6f} loss_g={lg:.6f} "
              f"ck_e={ck(t_e):.8f} ck_g={ck(t_g):.8f}", flush=True)


if __name__ == "__main__":
    mode = sys.argv[1] if len(sys.argv) > 1 else "all"
    if mode in ("all", "qkv"):
        qkv_unit()
    if mode in ("all", "addln"):
        add_ln_unit()
    if mode in ("all", "graph"):
        graph_probe()
