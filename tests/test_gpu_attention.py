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


def test_masked_softmax_pool_matches_torch():
    from deeprec_amd.ops.fused_attention import masked_softmax_pool

    torch.manual_seed(5)
    b, t, d = 257, 50, 32
    scores = torch.randn(b, t, device=DEV, requires_grad=True)
    seq = torch.randn(b, t, d, device=DEV, requires_grad=True)
    mask = torch.rand(b, t, device=DEV) > 0.3
    mask[0] = False       # fully masked row -> zero pooled output
    mask[:, 0] = True     # every other row has >=1 valid position
    mask[0, :] = False

    out = masked_softmax_pool(scores, seq, mask)
    sr = scores.detach().clone().requires_grad_(True)
    qr = seq.detach().clone().requires_grad_(True)
    mf = mask.float()
    e = torch.exp(sr - sr.amax(dim=1, keepdim=True)) * mf
    w = e / (e.sum(dim=1, keepdim=True) + 1e-20)
    ref = (w.unsqueeze(2) * qr).sum(1)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
    assert bool((out[0] == 0).all())

    g = torch.randn(b, d, device=DEV)
    out.backward(g)
    ref.backward(g)
    torch.testing.assert_close(scores.grad, sr.grad, rtol=1e-3, atol=1e-5)
    torch.testing.assert_close(seq.grad, qr.grad, rtol=1e-4, atol=1e-6)
    assert bool((scores.grad[0] == 0).all())


def test_din_trains_with_fused_attend_tail():
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.sequence import DIN
    from deeprec_amd.optimizers import AdamAsyncOptimizer
    torch.manual_seed(6)
    m = DIN(device=DEV, bf16=True)
    ds = CriteoSyntheticDataset(batch_size=512, seed=7, device=DEV)
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables(),
                             learning_rate=0.01)
    first = None
    for i in range(4):
        dense, ids, seq, target, labels = ds.next_seq_batch(seq_len=50)
        loss = m.loss_fn(m(dense, ids[:, :m.num_sparse], seq, target),
                         labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        if first is None:
            first = float(loss)
    assert torch.isfinite(loss) and float(loss) < first + 0.5
