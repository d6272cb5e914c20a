"""CPU reference-path tests for the fused ops (the same references are the
GPU numerics oracle in test_ops_gpu.py)."""
import torch
import torch.nn.functional as F

from prime_amd import ops
from prime_amd.ops import reference as ref


def test_rmsnorm_matches_manual():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y = ops.rmsnorm(x, w, eps=1e-5)
    rstd = torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5)
    torch.testing.assert_close(y, x * rstd * w, atol=1e-5, rtol=1e-5)


def test_rmsnorm_autograd_cpu():
    x = torch.randn(3, 32, requires_grad=True)
    w = torch.randn(32, requires_grad=True)
    y = ops.rmsnorm(x, w)
    y.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert w.grad is not None and torch.isfinite(w.grad).all()


def test_rope_inverse():
    cos, sin = ref.rope_tables(64, 128)
    x = torch.randn(2, 16, 4, 64)
    y = ops.apply_rope(x, cos, sin)
    back = ref.apply_rope(y, cos, -sin)
    torch.testing.assert_close(back, x, atol=1e-5, rtol=1e-5)


def test_rope_preserves_norm():
    cos, sin = ref.rope_tables(64, 128)
    x = torch.randn(1, 8, 2, 64)
    y = ops.apply_rope(x, cos, sin)
    torch.testing.assert_close(
        x.pow(2).sum(-1), y.pow(2).sum(-1), atol=1e-4, rtol=1e-4
    )


def test_swiglu_matches_manual():
    gu = torch.randn(5, 32)
    g, u = gu.chunk(2, -1)
    torch.testing.assert_close(ops.swiglu(gu), F.silu(g) * u, atol=1e-5, rtol=1e-5)


def test_attention_matches_sdpa():
    q = torch.randn(2, 32, 4, 16)
    k = torch.randn(2, 32, 2, 16)
    v = torch.randn(2, 32, 2, 16)
    out = ops.flash_attention(q, k, v, causal=True)
    kk = k.repeat_interleave(2, dim=2)
    vv = v.repeat_interleave(2, dim=2)
    want = F.scaled_dot_product_attention(
        q.transpose(1, 2), kk.transpose(1, 2), vv.transpose(1, 2), is_causal=True
    ).transpose(1, 2)
    torch.testing.assert_close(out, want, atol=1e-4, rtol=1e-4)


def test_cross_entropy_matches_torch():
    logits = torch.randn(10, 64)
    tgt = torch.randint(0, 64, (10,))
    tgt[3] = -100
    torch.testing.assert_close(
        ops.cross_entropy(logits, tgt),
        F.cross_entropy(logits, tgt, ignore_index=-100),
        atol=1e-5, rtol=1e-5,
    )


def test_quant_int8_roundtrip_error():
    x = torch.randn(8192) * 0.01
    q, s = ref.quant_int8_blockwise(x)
    back = ref.dequant_int8_blockwise(q, s)
    # blockwise int8: relative error bounded by scale/2 = amax/254 per block
    err = (back - x).abs().max()
    assert err <= x.abs().max() / 127 + 1e-8


def test_quant_int8_zero_block():
    x = torch.zeros(2048)
    q, s = ref.quant_int8_blockwise(x)
    assert q.abs().max() == 0
    back = ref.dequant_int8_blockwise(q, s)
    assert back.abs().max() == 0


def test_adamw_reference_decreases_simple_loss():
    p32 = torch.ones(64)
    m = torch.zeros(64)
    v = torch.zeros(64)
    for step in range(1, 50):
        g = 2 * p32  # d/dp of p^2
        ref.adamw_step(p32, g, m, v, lr=0.05, beta1=0.9, beta2=0.95,
                       eps=1e-8, wd=0.0, step=step)
    assert p32.abs().max() < 0.5


def test_fused_add_rmsnorm_cpu():
    x = torch.randn(4, 64, requires_grad=True)
    res = torch.randn(4, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y, s = ops.fused_add_rmsnorm(x, res, w)
    torch.testing.assert_close(s, x + res)
    torch.testing.assert_close(y, ops.rmsnorm((x + res).detach(), w.detach()))
    (y.sum() + 2 * s.sum()).backward()
    # analytic check vs separate ops
    x2 = x.detach().clone().requires_grad_(True)
    r2 = res.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    s2 = x2 + r2
    y2 = ops.rmsnorm(s2, w2)
    (y2.sum() + 2 * s2.sum()).backward()
    torch.testing.assert_close(x.grad, x2.grad, atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(res.grad, r2.grad, atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(w.grad, w2.grad, atol=1e-5, rtol=1e-4)


def test_fp8_rowwise_env_gate(monkeypatch):
    """Per-row fp8 scaling is opt-in: default off (delayed scaling wins
    the 10B A/B by 1.5%), PRIME_AMD_FP8_ROWWISE=1 turns it on."""
    from prime_amd.ops.functional import _fp8_rowwise_on

    monkeypatch.delenv("PRIME_AMD_FP8_ROWWISE", raising=False)
    assert not _fp8_rowwise_on()
    monkeypatch.setenv("PRIME_AMD_FP8_ROWWISE", "1")
    assert _fp8_rowwise_on()
    monkeypatch.setenv("PRIME_AMD_FP8_ROWWISE", "0")
    assert not _fp8_rowwise_on()
