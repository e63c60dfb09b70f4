"""Bisect the GPU-bench NaN: which ingredient (bf16 replica, HIP kernels,
hipGraph, fused Adam) makes the AttnCNP-1D training loss go NaN.

Runs several 200-step variants of the bench step, printing loss every 20
steps and reporting the first non-finite step.
"""

import copy
import os
import sys

import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from npf import CNPFLoss
from npf.zoo import attncnp_1d

STEPS = 220


def make_pool(device, batch=32, seed=1234):
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "bench_mod", os.path.join(os.path.dirname(__file__), "..", "bench.py")
    )
    bench = importlib.util.module_from_spec(spec)
    argv = sys.argv
    sys.argv = ["bench.py"]
    spec.loader.exec_module(bench)
    sys.argv = argv
    return bench.make_task_pool(device, batch, seed)


def run_variant(name, pool, device, bf16=True, eager_ops=False, fused_adam=True):
    os.environ["NPF_FORCE_EAGER"] = "1" if eager_ops else "0"
    torch.manual_seed(123)
    model = attncnp_1d().to(device)
    crit = CNPFLoss()
    crit.train()
    model.train()
    comp_dtype = torch.bfloat16 if bf16 else torch.float32

    if bf16:
        model_c = copy.deepcopy(model).to(torch.bfloat16)
        model_c.train()
    else:
        model_c = model
    master = [p for p in model.parameters() if p.requires_grad]
    computep = [p for p in model_c.parameters() if p.requires_grad]
    try:
        opt = torch.optim.Adam(master, lr=1e-3, fused=fused_adam, capturable=fused_adam)
    except Exception:
        opt = torch.optim.Adam(master, lr=1e-3)

    first_bad = None
    for i in range(STEPS):
        Xc, Yc, Xt, Yt = pool[i % len(pool)]
        if bf16:
            with torch.no_grad():
                torch._foreach_copy_(computep, master)
        opt.zero_grad(set_to_none=True)
        out = model_c(
            X_cntxt=Xc.to(comp_dtype), Y_cntxt=Yc.to(comp_dtype), X_trgt=Xt.to(comp_dtype)
        )
        loss = crit(out, Yt)
        loss.backward()
        if bf16:
            with torch.no_grad():
                for mp, cp in zip(master, computep):
                    mp.grad = cp.grad.float() if cp.grad is not None else None
        opt.step()
        l = float(loss.detach())
        if i % 20 == 0:
            print(f"  [{name}] step {i}: {l:.2f}", flush=True)
        if first_bad is None and not (l == l and abs(l) < 1e30):
            first_bad = i
            # diagnose which tensor went bad
            loc, scale = out[0].base_dist.loc, out[0].base_dist.scale
            print(
                f"  [{name}] FIRST NON-FINITE at step {i}: loss={l} "
                f"loc finite={bool(torch.isfinite(loc).all())} "
                f"scale finite={bool(torch.isfinite(scale).all())} "
                f"scale min={float(scale.min()):.3e}",
                flush=True,
            )
            gbad = [
                n
                for (n, p) in model_c.named_parameters()
                if p.grad is not None and not torch.isfinite(p.grad).all()
            ]
            pbad = [
                n
                for (n, p) in model_c.named_parameters()
                if not torch.isfinite(p).all()
            ]
            print(f"  [{name}] nonfinite grads: {gbad[:8]}", flush=True)
            print(f"  [{name}] nonfinite params: {pbad[:8]}", flush=True)
            break
    print(f"[{name}] -> {'NaN at step %d' % first_bad if first_bad is not None else 'clean (final %.2f)' % l}", flush=True)
    return first_bad


def main():
    device = torch.device("cuda:0")
    pool = make_pool(device)
    run_variant("bf16+hip+fused", pool, device, bf16=True, eager_ops=False, fused_adam=True)
    run_variant("bf16+eagerops", pool, device, bf16=True, eager_ops=True, fused_adam=True)
    run_variant("fp32+hip", pool, device, bf16=False, eager_ops=False, fused_adam=True)
    run_variant("bf16+hip+plainadam", pool, device, bf16=True, eager_ops=False, fused_adam=False)


if __name__ == "__main__":
    main()
