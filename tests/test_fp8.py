"""OCP fp8 (e4m3fn) inference path — CPU numerics of the quantizer and
the Fp8Linear emulation (reference capability:
tools/low_precision_optimize post-training quantization; the GPU MFMA
kernel is validated against the same arithmetic in test_gpu_fp8.py)."""
import pytest
import torch

from deeprec_amd.ops.fp8 import (Fp8Linear, _quant_rows_torch,
                                 convert_mlp_to_fp8, dequantize_fp8_rows,
                                 quantize_fp8_rows)


def test_quant_roundtrip_error_bound():
    torch.manual_seed(0)
    x = torch.randn(64, 96) * torch.logspace(-2, 2, 64).unsqueeze(1)
    q, s = quantize_fp8_rows(x)
    assert q.dtype == torch.uint8 and s.shape == (64,)
    deq = dequantize_fp8_rows(q, s)
    # e4m3 RNE: relative error <= 2^-4 per element plus the subnormal
    # quantum of the scaled range
    tol = x.abs() * 2 ** -4 + s.unsqueeze(1) * 2 ** -9 * 1.01
    assert bool(((deq - x).abs() <= tol + 1e-12).all())


def test_quant_zero_and_extreme_rows():
    x = torch.zeros(3, 8)
    x[1] = 1e-30  # tiny: scale floors at amax/448, no inf/nan
    x[2] = 1e30   # huge: clamp keeps the convert finite
    q, s = quantize_fp8_rows(x)
    deq = dequantize_fp8_rows(q, s)
    assert torch.isfinite(deq).all()
    assert bool((deq[0] == 0).all())
    # the huge row quantizes to +-448 * scale = amax exactly
    torch.testing.assert_close(deq[2], x[2], rtol=2 ** -4, atol=0)


def test_quant_matches_torch_float8_cast():
    """The quantizer's convert step IS torch.float8_e4m3fn RNE — the
    reference and the kernel share the same arithmetic contract."""
    torch.manual_seed(1)
    x = torch.randn(16, 32)
    q, s = _quant_rows_torch(x)
    manual = (x / s.unsqueeze(1)).clamp(-448, 448).to(torch.float8_e4m3fn)
    assert bool((q == manual.view(torch.uint8)).all())


def test_fp8_linear_close_to_fp32():
    torch.manual_seed(2)
    lin = torch.nn.Linear(64, 48)
    x = torch.randn(128, 64)
    ref = lin(x)
    y = Fp8Linear.from_linear(lin)(x).float()
    rel = (y - ref).norm() / ref.norm()
    assert rel < 0.05, f"fp8 linear rel Frobenius err {rel:.4f}"
    cos = torch.nn.functional.cosine_similarity(
        y.reshape(-1), ref.reshape(-1), dim=0)
    assert cos > 0.995


def test_fp8_linear_activation_and_shape():
    torch.manual_seed(3)
    lin = torch.nn.Linear(32, 16)
    f8 = Fp8Linear.from_linear(lin, act="relu")
    x = torch.randn(4, 7, 32)
    y = f8(x)
    # output keeps the caller's dtype (compute is e4m3 + fp32 accum,
    # rounded through bf16)
    assert y.shape == (4, 7, 16) and y.dtype == x.dtype
    assert bool((y >= 0).all())
    ref = torch.relu(lin(x))
    rel = (y.float() - ref).norm() / (ref.norm() + 1e-9)
    assert rel < 0.08


def test_convert_mlp_to_fp8():
    torch.manual_seed(4)
    mlp = torch.nn.Sequential(
        torch.nn.Linear(24, 32), torch.nn.ReLU(), torch.nn.Linear(32, 8))
    x = torch.randn(64, 24)
    ref = mlp(x)
    n = convert_mlp_to_fp8(mlp)
    assert n == 2
    assert isinstance(mlp[0], Fp8Linear) and isinstance(mlp[2], Fp8Linear)
    y = mlp(x).float()
    rel = (y - ref).norm() / ref.norm()
    assert rel < 0.12, f"2-layer fp8 MLP rel err {rel:.4f}"


def test_fp8_predictor_full_update_cycle(tmp_path):
    """fp8 serving survives the online-update cycle: the converter
    reverts around a full model update so restored fp32 weights are
    re-quantized (reference: FullModelUpdate + low-precision tool)."""
    from deeprec_amd.checkpoint.saver import Saver
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.dlrm import DLRM
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.ops.fp8 import Fp8Linear
    from deeprec_amd.serving.predictor import Predictor
    from deeprec_amd.embedding.variable import GLOBAL_STEP

    m = DLRM(device="cpu", bf16=False)
    ds = CriteoSyntheticDataset(batch_size=16, seed=3, matrix_format=True)
    opt = AdagradOptimizer(params=m.parameters(),
                           embedding_variables=m.embedding_variables(),
                           learning_rate=0.1)
    saver = Saver(module=m, embedding_variables=m.embedding_variables(),
                  optimizer=opt)

    def train(n):
        for _ in range(n):
            dense, ids, labels = ds.next_batch()
            loss = m.loss_fn(m(dense, ids), labels)
            opt.zero_grad()
            loss.backward()
            opt.step()

    train(2)
    saver.save(str(tmp_path), GLOBAL_STEP.value)

    m2 = DLRM(device="cpu", bf16=False, name_prefix="dlrm_fp8")
    m2.collection.name = m.collection.name
    m2.collection.storage.default_values.copy_(
        m.collection.storage.default_values)
    pred = Predictor(m2, str(tmp_path), num_sessions=1, fp8_mlp=True)
    assert any(isinstance(mm, Fp8Linear) for mm in m2.modules())
    dense, ids, _ = ds.next_batch()
    p1 = pred.predict(dense, ids)
    ref = torch.sigmoid(m(dense, ids, train=False))
    assert (p1.float() - ref).abs().max() < 0.05  # e4m3 MLP tolerance

    # dense weights move; a NEW full checkpoint must re-quantize them
    train(3)
    saver.save(str(tmp_path), GLOBAL_STEP.value)
    assert pred.poll_updates() >= 1
    p2 = pred.predict(dense, ids)
    ref2 = torch.sigmoid(m(dense, ids, train=False))
    assert (p2.float() - ref2).abs().max() < 0.05
    assert not torch.allclose(p1.float(), p2.float())  # update landed


def test_fp8_gather_output_halves_bytes():
    """The fp8 EV gather-output story: quantized rows carry half the
    bytes of bf16 and reconstruct within e4m3 tolerance."""
    rows = torch.randn(50, 16)
    q, s = quantize_fp8_rows(rows)
    payload = q.numel() * q.element_size() + s.numel() * s.element_size()
    bf16_payload = rows.numel() * 2
    assert payload < bf16_payload
    deq = dequantize_fp8_rows(q, s)
    assert (deq - rows).abs().max() < rows.abs().max() * 0.1


def test_masked_softmax_pool_cpu_fallback():
    """CPU fallback of the fused DIN attend tail matches the plain torch
    chain (and zeroes fully-masked rows)."""
    from deeprec_amd.ops.fused_attention import masked_softmax_pool
    torch.manual_seed(5)
    b, t, d = 8, 11, 4
    scores = torch.randn(b, t)
    seq = torch.randn(b, t, d)
    mask = torch.rand(b, t) > 0.4
    mask[:, 0] = True
    mask[3] = False
    out = masked_softmax_pool(scores, seq, mask)
    w = torch.softmax(scores.masked_fill(~mask, -1e9), 1)
    ref = (w.unsqueeze(2) * seq).sum(1)
    live = mask.any(1)
    torch.testing.assert_close(out[live], ref[live], rtol=1e-5, atol=1e-6)
    assert bool((out[3] == 0).all())


def test_convert_mlp_to_fp8_across_zoo_models():
    """The fp8 converter walks every zoo model's module tree (Sequential
    MLPs, cross nets, towers, attention blocks) and the converted model
    still forwards close to the original."""
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models import MODEL_REGISTRY, SEQUENCE_MODELS

    torch.manual_seed(0)
    ds = CriteoSyntheticDataset(batch_size=32, seed=1, matrix_format=True)
    dense, ids, _ = ds.next_batch()
    for name in ("wdl", "dcn", "mmoe", "masknet"):
        m = MODEL_REGISTRY[name](device="cpu", bf16=False)
        with torch.no_grad():
            ref = m(dense, ids, train=False)
        n = convert_mlp_to_fp8(m)
        assert n > 0, name
        with torch.no_grad():
            y = m(dense, ids, train=False)
        ref0 = ref[0] if isinstance(ref, (list, tuple)) else ref
        y0 = y[0] if isinstance(y, (list, tuple)) else y
        rel = (y0.float() - ref0).norm() / (ref0.norm() + 1e-9)
        assert rel < 0.25, f"{name}: fp8 rel err {rel:.3f}"
