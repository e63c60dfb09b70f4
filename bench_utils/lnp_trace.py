"""attnlnp2d loss-component trace: find what goes non-finite."""
import os, sys, torch
sys.path.insert(0, "/root/repo")
import bench
from npf import ELBOLossLNPF
from npf.losses import sum_log_prob
from npf.ops import functional as F_ops
from npf import zoo

def run(steps=200, autocast=True, eager_ops=False):
    os.environ["NPF_FORCE_EAGER"] = "1" if eager_ops else "0"
    torch.manual_seed(123)
    device = torch.device("cuda:0")
    pool = bench._img_point_pool(device, 32, 1234)
    model = zoo.attnlnp_2d().to(device); model.train()
    crit = ELBOLossLNPF(); crit.train()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    tag = f"ac={autocast} eager={eager_ops}"
    for i in range(steps):
        Xc, Yc, Xt, Yt = pool[i % len(pool)]
        opt.zero_grad(set_to_none=True)
        if autocast:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                out = model(X_cntxt=Xc, Y_cntxt=Yc, X_trgt=Xt, Y_trgt=Yt)
        else:
            out = model(X_cntxt=Xc, Y_cntxt=Yc, X_trgt=Xt, Y_trgt=Yt)
        p_yCc, z_samples, q_zCc, q_zCct = out
        nll = -sum_log_prob(p_yCc, Yt).mean(0)
        kl = F_ops.gaussian_kl_sum(
            q_zCct.base_dist.loc.float(), q_zCct.base_dist.scale.float(),
            q_zCc.base_dist.loc.float(), q_zCc.base_dist.scale.float())
        loss = (nll + kl).mean(0)
        loss.backward()
        gn = torch.nn.utils.clip_grad_norm_(model.parameters(), 1e9)
        opt.step()
        l, n, k = float(loss), float(nll.mean()), float(kl.mean())
        if i % 10 == 0 or not (l == l):
            print(f"[{tag}] {i}: loss={l:.1f} nll={n:.1f} kl={k:.2f} gn={float(gn):.1f}",
                  flush=True)
        if not (l == l and abs(l) < 1e30):
            sc = p_yCc.base_dist.scale
            print(f"[{tag}] NONFINITE at {i}: scale[min={float(sc.min()):.2e},"
                  f"max={float(sc.max()):.2e}] qs_min={float(q_zCc.base_dist.scale.min()):.3f}",
                  flush=True)
            bad = [n_ for n_, p in model.named_parameters()
                   if p.grad is not None and not torch.isfinite(p.grad).all()]
            print(f"[{tag}] bad grads: {bad[:6]}", flush=True)
            return
    print(f"[{tag}] clean, final {l:.1f}", flush=True)

if __name__ == "__main__":
    mode = sys.argv[1] if len(sys.argv) > 1 else "all"
    if mode in ("all", "ac"): run(autocast=True, eager_ops=False)
    if mode in ("all", "fp32"): run(autocast=False, eager_ops=False)
    if mode in ("all", "aceager"): run(autocast=True, eager_ops=True)
