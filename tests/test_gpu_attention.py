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
