"""Ops-layer tests: torch reference correctness + autograd, and (on GPU)
HIP-kernel parity against the fp32 composed-torch oracle."""

import math

import pytest
import torch

from npf.ops import functional as F_ops


def _attn_oracle(keys, queries, values, scale):
    logits = torch.einsum("bkd,bqd->bqk", keys, queries) * scale
    return torch.bmm(logits.softmax(-1), values)


def _setconv_oracle(keys, queries, values, sigma):
    diff = keys.unsqueeze(1) - queries.unsqueeze(2)
    dist = torch.norm(diff, p=2, dim=-1, keepdim=True)
    inp = -((dist / sigma) ** 2)
    w = torch.softmax(inp, dim=-2)
    density = torch.exp(inp).sum(dim=-2)
    out = (w * values.unsqueeze(1)).sum(dim=2)
    return torch.cat([out, density], dim=-1)


class TestAttention:
    def test_matches_oracle(self):
        g = torch.Generator().manual_seed(0)
        k = torch.randn(4, 9, 16, generator=g)
        q = torch.randn(4, 33, 16, generator=g)
        v = torch.randn(4, 9, 16, generator=g)
        scale = 1 / math.sqrt(16)
        out = F_ops.attention_qkv(k, q, v, scale)
        assert torch.allclose(out, _attn_oracle(k, q, v, scale), atol=1e-6)

    def test_autograd(self):
        g = torch.Generator().manual_seed(1)
        k = torch.randn(2, 5, 8, generator=g, dtype=torch.float64, requires_grad=True)
        q = torch.randn(2, 7, 8, generator=g, dtype=torch.float64, requires_grad=True)
        v = torch.randn(2, 5, 8, generator=g, dtype=torch.float64, requires_grad=True)
        assert torch.autograd.gradcheck(
            lambda k, q, v: F_ops.attention_qkv(k, q, v, 0.35), (k, q, v)
        )


class TestSetConv:
    def test_matches_oracle(self):
        g = torch.Generator().manual_seed(0)
        k = torch.rand(3, 11, 1, generator=g) * 2 - 1
        q = torch.rand(3, 19, 1, generator=g) * 2 - 1
        v = torch.randn(3, 11, 4, generator=g)
        sigma = torch.tensor(0.13)
        out = F_ops.setconv_gaussian(k, q, v, sigma)
        assert torch.allclose(out, _setconv_oracle(k, q, v, sigma), atol=1e-6)

    def test_autograd_matches_composed(self):
        g = torch.Generator().manual_seed(2)
        k = torch.rand(2, 6, 1, generator=g, requires_grad=True)
        q = torch.rand(2, 9, 1, generator=g, requires_grad=True)
        v = torch.randn(2, 6, 3, generator=g, requires_grad=True)
        sigma = torch.tensor(0.2, requires_grad=True)
        out = F_ops.setconv_gaussian(k, q, v, sigma)
        loss = (out * torch.arange(out.numel()).float().view_as(out)).sum()
        gk, gq, gv, gs = torch.autograd.grad(loss, (k, q, v, sigma))

        k2 = k.detach().requires_grad_(True)
        q2 = q.detach().requires_grad_(True)
        v2 = v.detach().requires_grad_(True)
        s2 = sigma.detach().requires_grad_(True)
        out2 = _setconv_oracle(k2, q2, v2, s2)
        loss2 = (out2 * torch.arange(out2.numel()).float().view_as(out2)).sum()
        gk2, gq2, gv2, gs2 = torch.autograd.grad(loss2, (k2, q2, v2, s2))
        for a, b in [(gk, gk2), (gq, gq2), (gv, gv2), (gs, gs2)]:
            assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()


class TestGaussianLL:
    def test_matches_distribution_api(self):
        g = torch.Generator().manual_seed(0)
        loc = torch.randn(2, 3, 7, 2, generator=g)
        scale = torch.rand(2, 3, 7, 2, generator=g) + 0.1
        y = torch.randn(3, 7, 2, generator=g)
        dist = torch.distributions.Independent(
            torch.distributions.Normal(loc, scale), 1
        )
        expected = dist.log_prob(y).reshape(2, 3, -1).sum(-1)
        got = F_ops.gaussian_nll_sum(loc, scale, y)
        assert torch.allclose(got, expected, atol=1e-5)

    def test_autograd(self):
        g = torch.Generator().manual_seed(1)
        loc = torch.randn(2, 2, 5, 1, generator=g, dtype=torch.float64,
                          requires_grad=True)
        scale = (torch.rand(2, 2, 5, 1, generator=g, dtype=torch.float64) + 0.1
                 ).requires_grad_(True)
        y = torch.randn(2, 5, 1, generator=g, dtype=torch.float64)
        assert torch.autograd.gradcheck(
            lambda l, s: F_ops.gaussian_nll_sum(l, s, y), (loc, scale)
        )


# --------------------------------------------------------------------------- #
# GPU: HIP kernels vs fp32 torch oracle
# --------------------------------------------------------------------------- #


@pytest.mark.gpu
class TestHIPKernels:
    def setup_method(self):
        from npf.ops import has_extension

        if not has_extension():
            pytest.fail("HIP extension not built/loadable on a GPU box")

    @pytest.mark.parametrize("B,K,Q,D,dtype", [
        # fp32 -> exact VALU kernel, tight tolerance
        (8, 13, 128, 16, "f32"), (32, 50, 128, 16, "f32"),
        (256, 300, 1024, 16, "f32"), (8, 96, 64, 16, "f32"),
        # bf16 large-K -> MFMA flash kernel, bf16-level tolerance
        (8, 96, 64, 16, "bf16"), (4, 130, 50, 16, "bf16"),
        (16, 307, 307, 16, "bf16"), (256, 300, 1024, 16, "bf16"),
        (8, 1024, 1024, 16, "bf16"),
        # bf16 small-K -> VALU kernel on bf16 inputs
        (32, 50, 128, 16, "bf16"),
    ])
    def test_attention_fwd_bwd(self, B, K, Q, D, dtype):
        td = torch.float32 if dtype == "f32" else torch.bfloat16
        atol_o, atol_g = (2e-4, 2e-3) if dtype == "f32" else (3e-2, 6e-2)
        g = torch.Generator(device="cuda").manual_seed(0)
        k = torch.randn(B, K, D, device="cuda", generator=g, dtype=td,
                        requires_grad=True)
        q = torch.randn(B, Q, D, device="cuda", generator=g, dtype=td,
                        requires_grad=True)
        v = torch.randn(B, K, D, device="cuda", generator=g, dtype=td,
                        requires_grad=True)
        scale = 1 / math.sqrt(D)
        out = F_ops.attention_qkv(k, q, v, scale)
        k0 = k.detach().cpu().float().requires_grad_(True)
        q0 = q.detach().cpu().float().requires_grad_(True)
        v0 = v.detach().cpu().float().requires_grad_(True)
        ref = _attn_oracle(k0, q0, v0, scale)
        assert torch.allclose(out.cpu().float(), ref, atol=atol_o), (
            (out.cpu().float() - ref).abs().max()
        )
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu().float())
        for a, b in [(k.grad, k0.grad), (q.grad, q0.grad), (v.grad, v0.grad)]:
            assert torch.allclose(a.cpu().float(), b, atol=atol_g), (
                (a.cpu().float() - b).abs().max()
            )

    @pytest.mark.parametrize("B,K,Q,C", [(4, 17, 33, 1), (32, 50, 192, 1),
                                         (64, 192, 128, 128)])
    def test_setconv_fwd_bwd(self, B, K, Q, C):
        g = torch.Generator(device="cuda").manual_seed(0)
        k = (torch.rand(B, K, 1, device="cuda", generator=g) * 2 - 1).requires_grad_(True)
        q = (torch.rand(B, Q, 1, device="cuda", generator=g) * 2 - 1).requires_grad_(True)
        v = torch.randn(B, K, C, device="cuda", generator=g, requires_grad=True)
        sigma = torch.tensor(0.1, device="cuda", requires_grad=True)
        out = F_ops.setconv_gaussian(k, q, v, sigma)
        k0 = k.detach().cpu().requires_grad_(True)
        q0 = q.detach().cpu().requires_grad_(True)
        v0 = v.detach().cpu().requires_grad_(True)
        s0 = sigma.detach().cpu().requires_grad_(True)
        ref = _setconv_oracle(k0, q0, v0, s0)
        assert torch.allclose(out.cpu(), ref, atol=2e-4), (out.cpu() - ref).abs().max()
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu())
        for a, b in [(k.grad, k0.grad), (q.grad, q0.grad), (v.grad, v0.grad),
                     (sigma.grad, s0.grad)]:
            assert torch.allclose(a.cpu(), b, atol=3e-3), (a.cpu() - b).abs().max()

    @pytest.mark.parametrize("Z,B,T,Y", [(1, 4, 128, 1), (16, 32, 128, 1),
                                         (8, 16, 1024, 3)])
    def test_gaussian_ll_fwd_bwd(self, Z, B, T, Y):
        g = torch.Generator(device="cuda").manual_seed(0)
        loc = torch.randn(Z, B, T, Y, device="cuda", generator=g, requires_grad=True)
        scale = (torch.rand(Z, B, T, Y, device="cuda", generator=g) + 0.05
                 ).requires_grad_(True)
        y = torch.randn(B, T, Y, device="cuda", generator=g)
        out = F_ops.gaussian_nll_sum(loc, scale, y)
        l0 = loc.detach().cpu().requires_grad_(True)
        s0 = scale.detach().cpu().requires_grad_(True)
        y0 = y.cpu()
        dist = torch.distributions.Independent(torch.distributions.Normal(l0, s0), 1)
        ref = dist.log_prob(y0).reshape(Z, B, -1).sum(-1)
        assert torch.allclose(out.cpu(), ref, atol=1e-3), (out.cpu() - ref).abs().max()
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu())
        assert torch.allclose(loc.grad.cpu(), l0.grad, atol=1e-3)
        assert torch.allclose(scale.grad.cpu(), s0.grad, atol=1e-3)

    @pytest.mark.parametrize("N,C,L,K,bn,res,training", [
        (4, 8, 32, 5, True, True, True),
        (32, 128, 192, 19, True, True, True),
        (32, 128, 192, 19, True, False, True),
        (16, 64, 100, 9, False, True, True),
        (32, 128, 192, 19, True, True, False),
        (512, 128, 192, 19, True, True, True),  # Z*B collapsed batch
    ])
    def test_conv_block_1d_fwd_bwd(self, N, C, L, K, bn, res, training):
        import torch.nn as nn

        from npf.ops.functional import _conv_block_ref

        g = torch.Generator(device="cuda").manual_seed(0)
        x = torch.randn(N, C, L, device="cuda", generator=g, requires_grad=True)
        conv = nn.Conv1d(C, C, K, padding=K // 2, groups=C).cuda()
        norm = nn.BatchNorm1d(C).cuda() if bn else None
        if norm is not None:
            with torch.no_grad():
                norm.weight.uniform_(0.5, 1.5, generator=g)
                norm.bias.uniform_(-0.2, 0.2, generator=g)
                norm.running_mean.uniform_(-0.1, 0.1, generator=g)
                norm.running_var.uniform_(0.5, 1.5, generator=g)
            norm.train(training)
        conv.train(training)
        residual = (
            torch.randn(N, C, L, device="cuda", generator=g, requires_grad=True)
            if res else None
        )

        # fp32 oracle on CPU with cloned modules/buffers
        conv0 = nn.Conv1d(C, C, K, padding=K // 2, groups=C)
        conv0.load_state_dict({k: v.cpu() for k, v in conv.state_dict().items()})
        norm0 = None
        if norm is not None:
            norm0 = nn.BatchNorm1d(C)
            norm0.load_state_dict({k: v.cpu() for k, v in norm.state_dict().items()})
            norm0.train(training)
        x0 = x.detach().cpu().requires_grad_(True)
        r0 = residual.detach().cpu().requires_grad_(True) if res else None

        out = F_ops.conv_block_1d(x, conv, bn=norm, residual=residual)
        ref = _conv_block_ref(x0, conv0.weight, conv0.bias, norm0, r0, training)
        assert torch.allclose(out.cpu(), ref, atol=2e-4), (
            (out.cpu() - ref).abs().max()
        )

        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu())
        pairs = [(x.grad, x0.grad), (conv.weight.grad, conv0.weight.grad),
                 (conv.bias.grad, conv0.bias.grad)]
        if res:
            pairs.append((residual.grad, r0.grad))
        if bn:
            pairs += [(norm.weight.grad, norm0.weight.grad),
                      (norm.bias.grad, norm0.bias.grad)]
        for a, b in pairs:
            assert torch.allclose(a.cpu(), b, atol=3e-3), (a.cpu() - b).abs().max()
        if bn and training:
            # running-stat update parity (in-place on the GPU module)
            assert torch.allclose(
                norm.running_mean.cpu(), norm0.running_mean, atol=1e-4
            )
            assert torch.allclose(
                norm.running_var.cpu(), norm0.running_var, atol=1e-4
            )

    @pytest.mark.parametrize("N,C,H,W,K,bn,res,training", [
        (2, 8, 16, 16, 5, True, True, True),
        (16, 128, 32, 32, 9, True, True, True),
        (8, 64, 33, 31, 9, True, False, True),  # odd sizes
        (8, 64, 32, 32, 9, False, True, True),
        (16, 128, 32, 32, 9, True, True, False),
    ])
    def test_conv_block_2d_fwd_bwd(self, N, C, H, W, K, bn, res, training):
        import torch.nn as nn

        from npf.ops.functional import _conv_block2d_ref

        g = torch.Generator(device="cuda").manual_seed(0)
        x = torch.randn(N, C, H, W, device="cuda", generator=g, requires_grad=True)
        conv = nn.Conv2d(C, C, K, padding=K // 2, groups=C).cuda()
        norm = nn.BatchNorm2d(C).cuda() if bn else None
        if norm is not None:
            with torch.no_grad():
                norm.weight.uniform_(0.5, 1.5, generator=g)
                norm.bias.uniform_(-0.2, 0.2, generator=g)
                norm.running_mean.uniform_(-0.1, 0.1, generator=g)
                norm.running_var.uniform_(0.5, 1.5, generator=g)
            norm.train(training)
        conv.train(training)
        residual = (
            torch.randn(N, C, H, W, device="cuda", generator=g, requires_grad=True)
            if res else None
        )

        conv0 = nn.Conv2d(C, C, K, padding=K // 2, groups=C)
        conv0.load_state_dict({k: v.cpu() for k, v in conv.state_dict().items()})
        norm0 = None
        if norm is not None:
            norm0 = nn.BatchNorm2d(C)
            norm0.load_state_dict({k: v.cpu() for k, v in norm.state_dict().items()})
            norm0.train(training)
        x0 = x.detach().cpu().requires_grad_(True)
        r0 = residual.detach().cpu().requires_grad_(True) if res else None

        out = F_ops.conv_block_2d(x, conv, bn=norm, residual=residual)
        ref = _conv_block2d_ref(x0, conv0.weight, conv0.bias, norm0, r0, training)
        assert torch.allclose(out.cpu(), ref, atol=3e-4), (
            (out.cpu() - ref).abs().max()
        )

        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu())
        pairs = [(x.grad, x0.grad), (conv.weight.grad, conv0.weight.grad),
                 (conv.bias.grad, conv0.bias.grad)]
        if res:
            pairs.append((residual.grad, r0.grad))
        if bn:
            pairs += [(norm.weight.grad, norm0.weight.grad),
                      (norm.bias.grad, norm0.bias.grad)]
        for a, b in pairs:
            assert torch.allclose(a.cpu(), b, atol=5e-3), (a.cpu() - b).abs().max()

    def test_resconvblock_module_fused_matches_cpu(self):
        """Module-level: the fused GPU ResConvBlock forward+backward matches
        the composed CPU module bit-for-policy (fp32)."""
        import torch.nn as nn

        from npf.architectures import ResConvBlock

        torch.manual_seed(0)
        blk = ResConvBlock(
            64, 64, nn.Conv1d, kernel_size=9, Normalization=nn.BatchNorm1d,
            n_conv_layers=2,
        )
        blk0 = ResConvBlock(
            64, 64, nn.Conv1d, kernel_size=9, Normalization=nn.BatchNorm1d,
            n_conv_layers=2,
        )
        blk0.load_state_dict(blk.state_dict())
        blk = blk.cuda().train()
        blk0 = blk0.train()

        x = torch.randn(8, 64, 100, device="cuda", requires_grad=True)
        x0 = x.detach().cpu().requires_grad_(True)
        out = blk(x)
        ref = blk0(x0)
        assert torch.allclose(out.cpu(), ref, atol=1e-3), (
            (out.cpu() - ref).abs().max()
        )
        out.sum().backward()
        ref.sum().backward()
        assert torch.allclose(x.grad.cpu(), x0.grad, atol=1e-3)
        for (n, p), (_, p0) in zip(blk.named_parameters(), blk0.named_parameters()):
            assert torch.allclose(p.grad.cpu(), p0.grad, atol=1e-2), n

    @pytest.mark.parametrize("B,C,H,W,K", [
        (2, 1, 16, 16, 5), (8, 3, 32, 32, 11), (4, 3, 33, 31, 11),
    ])
    def test_grid_density_fwd_bwd(self, B, C, H, W, K):
        from npf.ops.functional import _grid_density_ref

        g = torch.Generator(device="cuda").manual_seed(0)
        x = torch.rand(B, C, H, W, device="cuda", generator=g, requires_grad=True)
        mask = (torch.rand(B, C, H, W, device="cuda", generator=g) < 0.3).float()
        w = torch.randn(C, 1, K, K, device="cuda", generator=g, requires_grad=True)

        x0 = x.detach().cpu().requires_grad_(True)
        w0 = w.detach().cpu().requires_grad_(True)
        out = F_ops.grid_density(x, mask, w)
        ref = _grid_density_ref(x0, mask.cpu(), w0)
        assert torch.allclose(out.cpu(), ref, atol=2e-4), (
            (out.cpu() - ref).abs().max()
        )
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu())
        assert torch.allclose(x.grad.cpu(), x0.grad, atol=1e-3), (
            (x.grad.cpu() - x0.grad).abs().max()
        )
        assert torch.allclose(w.grad.cpu(), w0.grad, atol=1e-2), (
            (w.grad.cpu() - w0.grad).abs().max()
        )

    def test_gridconvcnp_module_fused_matches_cpu(self):
        """GridConvCNP encode path (fused density + fused 2D blocks) matches
        the CPU composed module."""
        import sys as _sys

        _sys.path.insert(0, "tests")
        from model_zoo import gridconvcnp_2d

        torch.manual_seed(0)
        m = gridconvcnp_2d(y_dim=3)
        m0 = gridconvcnp_2d(y_dim=3)
        m0.load_state_dict(m.state_dict())
        m = m.cuda().train()
        m0.train()

        B, H, W = 4, 32, 32
        g = torch.Generator().manual_seed(1)
        Y = torch.rand(B, H, W, 3, generator=g)
        mc = (torch.rand(B, H, W, 1, generator=g) < 0.3)
        mt = torch.ones(B, H, W, 1, dtype=torch.bool)
        out = m(mc.cuda(), Y.cuda(), mt.cuda())
        ref = m0(mc, Y, mt)
        a, b = out[0].base_dist.loc.cpu(), ref[0].base_dist.loc
        assert torch.allclose(a, b, atol=5e-3), (a - b).abs().max()

    @pytest.mark.parametrize("B,M,Z", [(4, 7, 3), (32, 1, 128), (16, 192, 16)])
    def test_gaussian_kl_fwd_bwd(self, B, M, Z):
        from npf.ops.functional import _kl_ref

        g = torch.Generator(device="cuda").manual_seed(0)
        mk = lambda: torch.randn(B, M, Z, device="cuda", generator=g, requires_grad=True)
        sk = lambda: (torch.rand(B, M, Z, device="cuda", generator=g) + 0.1
                      ).requires_grad_(True)
        mq, sq, mp, sp = mk(), sk(), mk(), sk()
        cpu = [t.detach().cpu().requires_grad_(True) for t in (mq, sq, mp, sp)]
        out = F_ops.gaussian_kl_sum(mq, sq, mp, sp)
        ref = _kl_ref(*cpu)
        assert torch.allclose(out.cpu(), ref, atol=1e-3), (out.cpu() - ref).abs().max()
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu())
        for a, b in zip((mq, sq, mp, sp), cpu):
            assert torch.allclose(a.grad.cpu(), b.grad, atol=1e-3), (
                (a.grad.cpu() - b.grad).abs().max()
            )

    @pytest.mark.parametrize("R,dims", [
        (100, [128, 128, 128]),            # x_encoder shape
        (4096, [128, 128, 128, 128, 128, 2]),  # decoder shape (4 hidden + out)
        (37, [1, 128, 128]),               # K=1 first layer (x_dim=1)
        (5696, [128, 128, 2]),
    ])
    def test_mlp_chain_fwd_bwd(self, R, dims):
        import torch.nn as nn

        g = torch.Generator(device="cuda").manual_seed(0)
        L = len(dims) - 1
        ws = [
            (torch.randn(dims[i + 1], dims[i], device="cuda", generator=g)
             * (1.0 / dims[i] ** 0.5)).requires_grad_(True)
            for i in range(L)
        ]
        bs = [
            torch.randn(dims[i + 1], device="cuda", generator=g).mul(0.1)
            .requires_grad_(True)
            for i in range(L)
        ]
        x = torch.randn(R, dims[0], device="cuda", generator=g,
                        dtype=torch.bfloat16, requires_grad=True)

        out = F_ops.mlp_chain(x, ws, bs)

        # bf16-eager oracle (same rounding points: bf16 GEMM in, bf16 out,
        # weights rounded to bf16 exactly like the kernel's LDS staging);
        # detached weight copies so its backward doesn't pollute ws/bs grads
        wd = [w.detach().to(torch.bfloat16).float() for w in ws]
        bd = [b.detach() for b in bs]

        def ref(xb):
            h = xb
            for i in range(L):
                z = (h.float() @ wd[i].t() + bd[i]).to(torch.bfloat16)
                h = torch.relu(z) if i < L - 1 else z
            return h

        x0 = x.detach().clone().requires_grad_(True)
        r = ref(x0)
        assert out.shape == r.shape
        # bf16 rounding differences compound per layer; compare loosely
        diff = (out.float() - r.float()).abs()
        assert diff.max() < 0.1 and diff.mean() < 0.01, (
            float(diff.max()), float(diff.mean())
        )

        dout = torch.randn_like(out, dtype=torch.bfloat16)
        out.backward(dout)
        r.backward(dout)
        assert torch.allclose(x.grad.float(), x0.grad.float(), atol=0.1), (
            (x.grad.float() - x0.grad.float()).abs().max()
        )
        # weight-grad correctness: autograd of a chain with the kernel's
        # rounding points (bf16 weights/activations — a pure-fp32 oracle's
        # ReLU masks flip near zero and its dW can differ structurally)
        ws2 = [w.detach().clone().requires_grad_(True) for w in ws]
        bs2 = [b.detach().clone().requires_grad_(True) for b in bs]
        h = x.detach().float()
        for i in range(L):
            z = h @ ws2[i].to(torch.bfloat16).float().t() + bs2[i]
            h = (
                torch.relu(z).to(torch.bfloat16).float() if i < L - 1 else z
            )
        h.backward(dout.float())
        for i in range(L):
            dw_rel = (ws[i].grad - ws2[i].grad).abs().max() / (
                ws2[i].grad.abs().max() + 1e-6
            )
            db_rel = (bs[i].grad - bs2[i].grad).abs().max() / (
                bs2[i].grad.abs().max() + 1e-6
            )
            assert dw_rel < 0.06, (i, float(dw_rel))
            assert db_rel < 0.06, (i, float(db_rel))

    def test_conv_block_2d_bf16_io(self):
        """bf16 I/O path (autocast regime): fp32-accumulated kernel vs the
        bf16-rounded composed oracle."""
        import torch.nn as nn

        from npf.ops.functional import _conv_block2d_ref

        g = torch.Generator(device="cuda").manual_seed(0)
        N, C, H, W, K = 8, 64, 32, 32, 9
        x = torch.randn(N, C, H, W, device="cuda", generator=g,
                        dtype=torch.bfloat16, requires_grad=True)
        conv = nn.Conv2d(C, C, K, padding=K // 2, groups=C).cuda()
        norm = nn.BatchNorm2d(C).cuda().train()
        res = torch.randn_like(x, requires_grad=True)

        out = F_ops.conv_block_2d(x, conv, bn=norm, residual=res)
        assert out.dtype == torch.bfloat16

        conv0 = nn.Conv2d(C, C, K, padding=K // 2, groups=C)
        conv0.load_state_dict({k: v.cpu() for k, v in conv.state_dict().items()})
        norm0 = nn.BatchNorm2d(C)
        norm0.train()
        x0 = x.detach().cpu().float().requires_grad_(True)
        r0 = res.detach().cpu().float().requires_grad_(True)
        ref = _conv_block2d_ref(x0, conv0.weight, conv0.bias, norm0, r0, True)
        assert torch.allclose(out.cpu().float(), ref, atol=0.15), (
            (out.cpu().float() - ref).abs().max()
        )
        dout = torch.randn_like(out)
        out.backward(dout)
        ref.backward(dout.cpu().float())
        assert torch.allclose(x.grad.cpu().float(), x0.grad, atol=0.2), (
            (x.grad.cpu().float() - x0.grad).abs().max()
        )


class TestAdviceGuards:
    """Round-2 guards from ADVICE.md (dispatch/LDS/zero-row edge cases)."""

    def test_padded_conv_rejected_by_fused_dispatch(self):
        # the zsmms CircularPad variant's PaddedConv (native padding=0 +
        # separate circular padder) must not pass the fused-kernel guard:
        # the fused path would zero same-pad and drop circular padding
        from npf.architectures.cnn import ResConvBlock
        from npf.utils.helpers import CircularPad2d, make_padded_conv

        Conv = make_padded_conv(torch.nn.Conv2d, CircularPad2d)
        blk = ResConvBlock(8, 8, Conv, kernel_size=5, n_conv_layers=2)
        assert not blk._is_plain_same_pad(blk.conv2_depthwise, 2)
        assert not blk._is_plain_same_pad(blk.conv1.depthwise, 2)
        # a plain same-padded conv passes
        blk2 = ResConvBlock(8, 8, torch.nn.Conv2d, kernel_size=5)
        assert blk2._is_plain_same_pad(blk2.conv2_depthwise, 2)

    def test_lds_guards(self):
        from npf.ops.functional import _cb1d_lds_ok, _cb2d_lds_ok, _gde_lds_ok

        ok = torch.zeros(1, 8, 128)
        big = torch.zeros(1, 8, 4096)
        assert _cb1d_lds_ok(ok, 19) and not _cb1d_lds_ok(big, 19)
        ok2 = torch.zeros(1, 8, 32, 32)
        big2 = torch.zeros(1, 8, 32, 2048)
        assert _cb2d_lds_ok(ok2, 9) and not _cb2d_lds_ok(big2, 9)
        assert _gde_lds_ok(ok2, 11) and not _gde_lds_ok(big2, 11)

    def test_mlp_chain_zero_rows_still_flows_grads(self):
        # an all-empty batch must still produce (zero) grads for every
        # parameter, or DDP flat-buffer reduction desynchronizes ranks
        from npf.ops.functional import mlp_chain

        ws = [torch.randn(16, 8, requires_grad=True),
              torch.randn(4, 16, requires_grad=True)]
        bs = [torch.randn(16, requires_grad=True),
              torch.randn(4, requires_grad=True)]
        x = torch.zeros(0, 8, requires_grad=True)
        out = mlp_chain(x, ws, bs)
        assert out.shape == (0, 4)
        out.sum().backward()
        for t in ws + bs:
            assert t.grad is not None
            assert torch.all(t.grad == 0)


@pytest.mark.gpu
class TestZsmmsFusedParity:
    def test_zsmms_model_gpu_matches_cpu(self):
        """The circular-padded zsmms GridConvCNP must produce the same
        forward on GPU (where fused dispatch is reachable) as the CPU
        composed path — i.e. the fused kernels must NOT engage for
        PaddedConv blocks (ADVICE.md high finding)."""
        from npf.zoo import gridconvcnp_zsmms

        torch.manual_seed(0)
        m = gridconvcnp_zsmms(y_dim=1)
        m0 = gridconvcnp_zsmms(y_dim=1)
        m0.load_state_dict(m.state_dict())
        m = m.cuda().eval()
        m0.eval()

        B, H, W = 2, 24, 24
        g = torch.Generator().manual_seed(3)
        mask_c = torch.rand(B, H, W, 1, generator=g) < 0.3
        mask_t = torch.ones(B, H, W, 1, dtype=torch.bool)
        Y = torch.rand(B, H, W, 1, generator=g) * 2 - 1
        out0 = m0(mask_c, Y, mask_t)[0].base_dist.loc
        out = m(mask_c.cuda(), Y.cuda(), mask_t.cuda())[0].base_dist.loc
        assert torch.allclose(out.cpu(), out0, atol=5e-4), (
            (out.cpu() - out0).abs().max()
        )


class TestNLLLogMeanExp:
    """Fused NPML objective (gaussian_nll_logmeanexp) vs composed torch."""

    def test_cpu_reference_matches_composed(self):
        import math

        from npf.ops import gaussian_nll_logmeanexp

        torch.manual_seed(0)
        Z, B, T, Y = 8, 6, 17, 2
        loc = torch.randn(Z, B, T, Y, requires_grad=True)
        scale = torch.rand(Z, B, T, Y).add(0.3).requires_grad_()
        y = torch.randn(B, T, Y)
        out = gaussian_nll_logmeanexp(loc, scale, y)

        import torch.distributions as D

        p = D.Independent(D.Normal(loc, scale), 1)
        w = p.log_prob(y.unsqueeze(0).expand(Z, B, T, Y)).sum(-1)
        ref = torch.logsumexp(w, 0) - math.log(Z)
        assert torch.allclose(out, ref, atol=1e-5)
        out.sum().backward()
        g1, g2 = loc.grad.clone(), scale.grad.clone()
        loc.grad = scale.grad = None
        ref.sum().backward()
        assert torch.allclose(g1, loc.grad, atol=1e-5)
        assert torch.allclose(g2, scale.grad, atol=1e-5)

    def test_nll_loss_dispatch_equals_manual(self):
        import math

        import torch.distributions as D

        from npf import NLLLossLNPF

        torch.manual_seed(1)
        Z, B, T, Y = 4, 5, 9, 1
        loc = torch.randn(Z, B, T, Y)
        scale = torch.rand(Z, B, T, Y).add(0.3)
        yt = torch.randn(B, T, Y)
        p = D.Independent(D.Normal(loc, scale), 1)
        crit = NLLLossLNPF(reduction=None)
        crit.eval()
        loss = crit((p, None, None, None), yt)
        w = p.log_prob(yt.unsqueeze(0).expand(Z, B, T, Y)).sum(-1)
        ref = -(torch.logsumexp(w, 0) - math.log(Z))
        assert torch.allclose(loss, ref, atol=1e-5)


@pytest.mark.gpu
class TestNLLLogMeanExpGPU:
    def test_fused_matches_fp32_reference(self):
        import math

        from npf.ops import gaussian_nll_logmeanexp

        torch.manual_seed(0)
        Z, B, T, Y = 16, 32, 128, 1
        loc = torch.randn(Z, B, T, Y, device="cuda", requires_grad=True)
        scale = torch.rand(Z, B, T, Y, device="cuda").add(0.3).requires_grad_()
        y = torch.randn(B, T, Y, device="cuda")
        out = gaussian_nll_logmeanexp(loc, scale, y)
        out.sum().backward()

        loc0 = loc.detach().cpu().requires_grad_()
        scale0 = scale.detach().cpu().requires_grad_()
        out0 = gaussian_nll_logmeanexp(loc0, scale0, y.cpu())
        out0.sum().backward()
        assert torch.allclose(out.cpu(), out0, atol=1e-3), (
            (out.cpu() - out0).abs().max()
        )
        assert torch.allclose(loc.grad.cpu(), loc0.grad, atol=1e-3)
        assert torch.allclose(scale.grad.cpu(), scale0.grad, atol=1e-3)


@pytest.mark.gpu
class TestFusedTransformerAttender:
    """Fused QKV+head-split / add+LN block vs the CPU fp32 module."""

    def _mk(self):
        from npf.architectures.attention import TransformerAttender

        torch.manual_seed(0)
        m = TransformerAttender(128, 128, 128)
        m0 = TransformerAttender(128, 128, 128)
        m0.load_state_dict(m.state_dict())
        return m.cuda(), m0

    @pytest.mark.parametrize("B,K,Q", [(4, 11, 64), (2, 50, 128), (3, 307, 1024)])
    def test_cross_attention_parity(self, B, K, Q):
        m, m0 = self._mk()
        g = torch.Generator().manual_seed(1)
        keys = torch.randn(B, K, 128, generator=g)
        queries = torch.randn(B, Q, 128, generator=g)
        values = torch.randn(B, K, 128, generator=g)
        with torch.no_grad():
            out0 = m0(keys, queries, values)
            out = m(keys.cuda(), queries.cuda(), values.cuda())
        # bf16 compute vs fp32 oracle: values are O(1); LN output O(1)
        assert (out.float().cpu() - out0).abs().max() < 0.1, (
            (out.float().cpu() - out0).abs().max()
        )
        assert (out.float().cpu() - out0).abs().mean() < 0.02

    def test_self_attention_parity(self):
        m, m0 = self._mk()
        g = torch.Generator().manual_seed(2)
        x = torch.randn(3, 40, 128, generator=g)
        with torch.no_grad():
            out0 = m0(x, x, x)
            xc = x.cuda()
            out = m(xc, xc, xc)
        assert (out.float().cpu() - out0).abs().max() < 0.1

    def test_gradients_flow_and_match(self):
        m, m0 = self._mk()
        g = torch.Generator().manual_seed(3)
        keys = torch.randn(2, 13, 128, generator=g)
        queries = torch.randn(2, 37, 128, generator=g)
        values = torch.randn(2, 13, 128, generator=g)
        # NOT square().sum(): that loss is numerically invariant to a
        # LayerNorm input (sum xhat^2 == D per row), so true input grads
        # are ~0 and bf16 noise dominates any relative comparison
        w = torch.randn(2, 37, 128, generator=g)

        kc = keys.cuda().requires_grad_()
        qc = queries.cuda().requires_grad_()
        vc = values.cuda().requires_grad_()
        (m(kc, qc, vc) * w.cuda()).sum().backward()

        k0 = keys.clone().requires_grad_()
        q0 = queries.clone().requires_grad_()
        v0 = values.clone().requires_grad_()
        (m0(k0, q0, v0) * w).sum().backward()

        for a, b, name in [
            (kc.grad, k0.grad, "dk"), (qc.grad, q0.grad, "dq"),
            (vc.grad, v0.grad, "dv"),
        ]:
            rel = (a.float().cpu() - b).abs().max() / (b.abs().max() + 1e-6)
            assert rel < 0.15, (name, rel)
        # every parameter receives a gradient
        for (n1, p1), (n0, p0) in zip(
            m.named_parameters(), m0.named_parameters()
        ):
            assert p1.grad is not None, n1
            denom = p0.grad.abs().max() + 1e-5
            rel = (p1.grad.float().cpu() - p0.grad).abs().max() / denom
            assert rel < 0.2, (n1, float(rel))
