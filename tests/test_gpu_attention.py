"""Fused MHA kernels vs the fp32 torch reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.mark.parametrize("b,t,d,h", [(32, 51, 32, 4), (17, 20, 64, 8),
                                     (8, 100, 32, 2), (4, 7, 16, 1)])
def test_mha_fwd_bwd_matches_torch(b, t, d, h):
    from deeprec_amd.ops.fused_attention import _torch_mha, fused_mha

    torch.manual_seed(b * 100 + t)
    q = torch.randn(b, t, d)
    k = torch.randn(b, t, d)
    v = torch.randn(b, t, d)
    pad = torch.rand(b, t) < 0.4
    pad[:, 0] = False  # at least one live key per sample
    scale = 1.0 / (d // h) ** 0.5

    qg = q.to(DEV).to(torch.bfloat16).requires_grad_(True)
    kg = k.to(DEV).to(torch.bfloat16).requires_grad_(True)
    vg = v.to(DEV).to(torch.bfloat16).requires_grad_(True)
    out_g = fused_mha(qg, kg, vg, pad.to(DEV), h, scale)

    qr = q.requires_grad_(True)
    kr = k.requires_grad_(True)
    vr = v.requires_grad_(True)
    out_r = _torch_mha(qr, kr, vr, pad, h, scale)
    torch.testing.assert_close(out_g.float().cpu(), out_r,
                               rtol=3e-2, atol=3e-2)

    g = torch.randn(b, t, d)
    out_g.backward(g.to(DEV).to(torch.bfloat16))
    out_r.backward(g)
    for got, ref in ((qg.grad, qr.grad), (kg.grad, kr.grad),
                     (vg.grad, vr.grad)):
        torch.testing.assert_close(got.float().cpu(), ref,
                                   rtol=5e-2, atol=5e-2)


def test_fused_transformer_layer_matches_cpu():
    from deeprec_amd.ops.fused_attention import FusedTransformerLayer

    torch.manual_seed(5)
    layer_c = FusedTransformerLayer(32, 4, 128)
    layer_g = FusedTransformerLayer(32, 4, 128)
    layer_g.load_state_dict(layer_c.state_dict())
    layer_g.to(DEV)
    x = torch.randn(16, 51, 32)
    pad = torch.rand(16, 51) < 0.3
    pad[:, 0] = False
    out_c = layer_c(x, pad)
    out_g = layer_g(x.to(DEV), pad.to(DEV))
    torch.testing.assert_close(out_g.float().cpu(), out_c.float(),
                               rtol=5e-2, atol=5e-2)


def test_bst_gpu_trains():
    from deeprec_amd.models.sequence import BST
    from deeprec_amd.optimizers import AdamAsyncOptimizer

    torch.manual_seed(0)
    m = BST(device=DEV, bf16=True, num_sparse=4)
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables())
    b, t = 256, 50
    for step in range(3):
        dense = torch.randn(b, 13, device=DEV)
        sparse = torch.randint(0, 1000, (b, 4), device=DEV)
        seq = torch.randint(0, 5000, (b, t), device=DEV)
        seq[:, t // 2:] = 0  # padding tail
        tgt = torch.randint(1, 5000, (b,), device=DEV)
        labels = torch.rand(b, device=DEV).round()
        logits = m(dense, sparse, seq, tgt)
        loss = m.loss_fn(logits, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss)


def test_fused_residual_ln_matches_torch():
    from deeprec_amd.ops.fused_attention import FusedResidualLN

    torch.manual_seed(2)
    for m, n in [(100, 32), (4096, 64), (77, 16)]:
        ln = FusedResidualLN(n).to(DEV)
        with torch.no_grad():
            ln.weight.mul_(1.5).add_(0.1)
            ln.bias.add_(0.05)
        x = (torch.randn(m, n) * 2).to(DEV).to(torch.bfloat16)
        a = torch.randn(m, n, device=DEV).to(torch.bfloat16)
        y = ln(x, a)
        # fp32 reference on the SAME bf16 inputs
        z = x.float() + a.float()
        ref = torch.nn.functional.layer_norm(
            z, (n,), ln.weight.detach(), ln.bias.detach(), ln.eps)
        torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
        g = torch.randn(m, n, device=DEV)
        xg = x.float().detach().requires_grad_(True)
        ag = a.float().detach().requires_grad_(True)
        wg = ln.weight.detach().clone().requires_grad_(True)
        bg = ln.bias.detach().clone().requires_grad_(True)
        ref2 = torch.nn.functional.layer_norm(xg + ag, (n,), wg, bg, ln.eps)
        ref2.backward(g)
        x2 = x.detach().requires_grad_(True)
        a2 = a.detach().requires_grad_(True)
        y2 = ln(x2, a2)
        y2.backward(g.to(torch.bfloat16))
        torch.testing.assert_close(x2.grad.float(), xg.grad,
                                   rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(a2.grad.float(), ag.grad,
                                   rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(ln.weight.grad, wg.grad,
                                   rtol=5e-2, atol=5e-1)
        torch.testing.assert_close(ln.bias.grad, bg.grad,
                                   rtol=5e-2, atol=5e-1)
        ln.weight.grad = None
        ln.bias.grad = None


def test_din_att_features_matches_torch():
    from deeprec_amd.ops.fused_attention import din_att_features

    torch.manual_seed(4)
    b, t, d = 33, 17, 32
    seq = torch.randn(b, t, d, device=DEV, requires_grad=True)
    tgt = torch.randn(b, d, device=DEV, requires_grad=True)
    out = din_att_features(seq, tgt)
    sr = seq.detach().clone().requires_grad_(True)
    tr = tgt.detach().clone().requires_grad_(True)
    te = tr.unsqueeze(1).expand(b, t, d)
    ref = torch.cat([sr, te, sr - te, sr * te], dim=2).reshape(b * t,
                                                               4 * d)
    torch.testing.assert_close(out.float(), ref, rtol=1e-2, atol=1e-2)
    g = torch.randn(b * t, 4 * d, device=DEV)
    out.backward(g.to(torch.bfloat16))
    ref.backward(g)
    torch.testing.assert_close(seq.grad, sr.grad, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(tgt.grad, tr.grad, rtol=2e-2, atol=2e-1)
