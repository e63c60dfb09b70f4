"""RCCL-on-hardware smoke (VERDICT round-1 item 2).

Two ranks on one GPU is 'invalid usage' for RCCL (duplicate-device), so the
1-GPU lease evidence is:
  world-1 RCCL: real process group init + all-reduce/broadcast kernels
  world-2 gloo with CUDA gradients: full FlatDDP replica-sync semantics,
  both ranks on cuda:0 (the collective transport differs; the DDP code
  path — flat buffer, views, reduce_, broadcast — is the one the 8-GPU
  driver run executes).
"""
import os
import sys

sys.path.insert(0, "/root/repo")

import torch
import torch.distributed as dist


def world1_rccl():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29581")
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    dist.init_process_group("nccl", rank=0, world_size=1)
    from npf.parallel import FlatDDP
    from npf.zoo import attncnp_1d

    torch.manual_seed(0)
    m = attncnp_1d().cuda()
    ddp = FlatDDP(m)  # broadcast over RCCL
    x = torch.rand(4, 9, 1, device="cuda") * 2 - 1
    y = torch.randn(4, 9, 1, device="cuda")
    xt = torch.rand(4, 32, 1, device="cuda") * 2 - 1
    yt = torch.randn(4, 32, 1, device="cuda")
    from npf import CNPFLoss

    crit = CNPFLoss()
    crit.train()
    m.train()
    ddp.zero_grad_()
    crit(m(x, y, xt, yt), yt).backward()
    ddp.reduce_()  # RCCL all-reduce kernel
    t = torch.ones(1 << 20, device="cuda")
    dist.all_reduce(t)
    torch.cuda.synchronize()
    assert float(t.sum()) == float(1 << 20)
    print("world1-rccl OK: init + broadcast + all_reduce on librccl")
    dist.destroy_process_group()


def world2_gloo_cuda(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    from npf import CNPFLoss
    from npf.parallel import FlatDDP
    from npf.train import set_seed
    from npf.zoo import attncnp_1d

    set_seed(0)
    m = attncnp_1d().cuda()
    ddp = FlatDDP(m)
    opt = torch.optim.Adam(m.parameters(), lr=1e-3)
    crit = CNPFLoss()
    crit.train()
    m.train()
    for step in range(3):
        g = torch.Generator().manual_seed(step * 100 + rank)
        x = (torch.rand(4, 9, 1, generator=g) * 2 - 1).cuda()
        y = torch.randn(4, 9, 1, generator=g).cuda()
        xt = (torch.rand(4, 32, 1, generator=g) * 2 - 1).cuda()
        yt = torch.randn(4, 32, 1, generator=g).cuda()
        ddp.zero_grad_()
        crit(m(x, y, xt, yt), yt).backward()
        ddp.reduce_()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in m.parameters()]).cpu()
    gathered = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    if rank == 0:
        assert torch.equal(gathered[0], gathered[1]), "replicas diverged"
        print("world2-gloo-cuda OK: replicas bitwise identical after 3 steps")
    dist.destroy_process_group()


if __name__ == "__main__":
    mode = sys.argv[1] if len(sys.argv) > 1 else "all"
    if mode in ("all", "world1"):
        world1_rccl()
    if mode in ("all", "world2"):
        import torch.multiprocessing as mp

        mp.spawn(world2_gloo_cuda, args=(2, 29583), nprocs=2, join=True)
