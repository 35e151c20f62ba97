"""GPU fp8 path: the e4m3 row quantizer kernel and the
mfma_f32_16x16x32_fp8_fp8 linear against torch references
(fp8_kernels.hip). Layout check uses asymmetric random inputs — a
symmetric probe cannot catch a row/col-swapped fragment mapping."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from deeprec_amd.ops.fp8 import (Fp8Linear, _quant_rows_torch,
                                 convert_mlp_to_fp8, dequantize_fp8_rows)


@pytest.fixture(scope="module")
def ext():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from deeprec_amd.ops.build_ext import require_extension
    return require_extension()


def test_quant_kernel_matches_torch(ext):
    torch.manual_seed(0)
    for dtype in (torch.float32, torch.bfloat16):
        x = (torch.randn(257, 96) * 7).to(dtype).cuda()
        q, s = ext.quant_rows_e4m3(x)
        qr, sr = _quant_rows_torch(x.cpu())
        torch.testing.assert_close(s.cpu(), sr, rtol=1e-6, atol=0)
        # both sides are RNE+clamp e4m3 of the same scaled values: the
        # only legal disagreement is a rounding tie on the bf16 path
        deq_g = dequantize_fp8_rows(q.cpu(), s.cpu())
        deq_c = dequantize_fp8_rows(qr, sr)
        mismatch = (q.cpu() != qr).float().mean().item()
        assert mismatch < 0.001, f"{mismatch:.4%} of bytes differ"
        # any residual disagreement is a 1-ulp e4m3 tie at that
        # element's magnitude
        lim = (torch.maximum(deq_g.abs(), deq_c.abs()) * 2 ** -3
               + sr.unsqueeze(1) * 2 ** -8)
        assert bool(((deq_g - deq_c).abs() <= lim).all())


def test_quant_kernel_zero_rows(ext):
    x = torch.zeros(5, 32, device="cuda")
    q, s = ext.quant_rows_e4m3(x)
    assert bool((q == 0).all()) and bool((s == 1).all())


@pytest.mark.parametrize("m,n,k", [(128, 64, 64), (100, 37, 53),
                                   (8192, 479, 32), (16, 16, 480)])
def test_fp8_mfma_linear_exact_layout(ext, m, n, k):
    """Quantize on GPU, rebuild the EXACT expected result on CPU from
    the quantized bytes (products of e4m3 values are exact in fp32), and
    compare — this pins the MFMA fragment layout, scale application and
    partial-tile bounds, not just 'roughly close'."""
    torch.manual_seed(m + n + k)
    x = torch.randn(m, k, device="cuda") * 3
    w = torch.randn(n, k, device="cuda")
    bias = torch.randn(n, device="cuda")
    qx, sx = ext.quant_rows_e4m3(x)
    qw, sw = ext.quant_rows_e4m3(w)
    out = ext.linear_fwd_fp8(qx, sx, qw, sw, bias, 0)
    a = qx.cpu().view(torch.float8_e4m3fn).float()
    b = qw.cpu().view(torch.float8_e4m3fn).float()
    ref = (a @ b.t()) * sx.cpu().unsqueeze(1) * sw.cpu().unsqueeze(0) \
        + bias.cpu()
    scale = ref.abs().max().clamp(min=1.0)
    err = (out.float().cpu() - ref).abs().max() / scale
    assert err < 2 ** -7, f"fp8 GEMM layout err {err:.2e}"  # bf16 out ulp


def test_fp8_mfma_linear_relu(ext):
    torch.manual_seed(7)
    x = torch.randn(64, 32, device="cuda")
    w = torch.randn(24, 32, device="cuda")
    qx, sx = ext.quant_rows_e4m3(x)
    qw, sw = ext.quant_rows_e4m3(w)
    out = ext.linear_fwd_fp8(qx, sx, qw, sw, None, 1)
    assert bool((out.float() >= 0).all())


def test_fp8_linear_module_vs_fp32(ext):
    torch.manual_seed(8)
    lin = torch.nn.Linear(480, 1024).cuda()
    x = torch.randn(512, 480, device="cuda")
    ref = lin(x)
    y = Fp8Linear.from_linear(lin)(x).float()
    rel = (y - ref).norm() / ref.norm()
    assert rel < 0.05, f"fp8 vs fp32 rel Frobenius {rel:.4f}"


def test_fp8_serving_mlp_end_to_end(ext):
    """Post-training conversion of a trained-shape DLRM top MLP; fp8
    serving output tracks the bf16 fused MLP."""
    from deeprec_amd.models.common import make_mlp
    torch.manual_seed(9)
    mlp = make_mlp([1024, 512, 256], 479, "cuda", bf16=True,
                   final_activation=True).cuda()
    x = torch.randn(256, 479, device="cuda")
    ref = mlp(x.to(torch.bfloat16)).float()
    n = convert_mlp_to_fp8(mlp)
    assert n == 3
    y = mlp(x).float()
    rel = (y - ref).norm() / (ref.norm() + 1e-9)
    assert rel < 0.1, f"fp8 serving MLP rel err {rel:.4f}"
