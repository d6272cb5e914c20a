import torch

from prime_amd.models import build_model, get_config
from prime_amd.parallel.flat import FlatParamSpace, FusedAdamW


def test_config_param_counts():
    c10 = get_config("intellect_10b")
    n = c10.n_params()
    assert 9.5e9 < n < 11e9, n
    c150 = get_config("llama_150m")
    assert 1.0e8 < c150.n_params() < 2.5e8


def test_forward_loss_backward():
    m = build_model("llama_test")
    x = torch.randint(0, 256, (2, 64))
    y = torch.randint(0, 256, (2, 64))
    loss = m.loss(x, y)
    assert torch.isfinite(loss)
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_activation_checkpointing_same_loss():
    torch.manual_seed(0)
    m1 = build_model("llama_test")
    torch.manual_seed(0)
    m2 = build_model("llama_test", activation_checkpointing=True)
    m2.train()
    x = torch.randint(0, 256, (2, 64))
    y = torch.randint(0, 256, (2, 64))
    l1 = m1.loss(x, y)
    l2 = m2.loss(x, y)
    torch.testing.assert_close(l1, l2, atol=1e-5, rtol=1e-5)
    # backward through the checkpointed graph (catches saved_tensors re-access)
    l1.backward()
    l2.backward()
    g1 = m1.layers[0].attn.wqkv.weight.grad
    g2 = m2.layers[0].attn.wqkv.weight.grad
    torch.testing.assert_close(g1, g2, atol=1e-5, rtol=1e-4)


def test_flat_param_space_views():
    m = build_model("llama_test")
    flat = FlatParamSpace(m)
    # params are views into flat_w
    for n, p in flat.params:
        o, k, shp = flat.offsets[n]
        assert p.data_ptr() == flat.flat_w[o : o + k].data_ptr()
        assert o % 64 == 0
    # writing flat_w changes the module weights
    flat.flat_w.zero_()
    assert m.tok_embeddings.weight.abs().max() == 0


def test_flat_grad_accumulation():
    m = build_model("llama_test")
    flat = FlatParamSpace(m)
    x = torch.randint(0, 256, (2, 32))
    y = torch.randint(0, 256, (2, 32))
    flat.zero_grad()
    m.loss(x, y).backward()
    g1 = flat.flat_grad.clone()
    m.loss(x, y).backward()  # accumulates
    torch.testing.assert_close(flat.flat_grad, 2 * g1, atol=1e-5, rtol=1e-4)


def test_fused_adamw_trains():
    torch.manual_seed(1)
    m = build_model("llama_test")
    flat = FlatParamSpace(m)
    opt = FusedAdamW(flat, lr=1e-3)
    x = torch.randint(0, 256, (4, 64))
    y = x.clone()  # learn identity-ish mapping
    losses = []
    for _ in range(10):
        flat.zero_grad()
        loss = m.loss(x, y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0]


def test_grad_clip():
    m = build_model("llama_test")
    flat = FlatParamSpace(m)
    flat.flat_grad.fill_(100.0)
    flat.clip_grad_norm_(1.0)
    assert flat.flat_grad.float().norm() <= 1.01


def test_generate_kv_cache_matches_full_forward():
    from prime_amd.models.generate import generate

    torch.manual_seed(3)
    m = build_model("llama_test")
    m.eval()
    B, L0, new = 2, 10, 6
    prompt = torch.randint(0, 256, (B, L0))
    out = generate(m, prompt, max_new_tokens=new, temperature=0.0)
    assert out.shape == (B, L0 + new)
    # reference: greedy with full recompute (no cache)
    cur = prompt.clone()
    for _ in range(new):
        pad = (cur.shape[1] + 63) // 64 * 64
        padded = torch.zeros(B, pad, dtype=cur.dtype)
        padded[:, : cur.shape[1]] = cur
        with torch.no_grad():
            h = m(padded)
        nxt = m.lm_head(h[:, cur.shape[1] - 1]).argmax(-1, keepdim=True)
        cur = torch.cat([cur, nxt], dim=1)
    assert out.tolist() == cur.tolist()


def test_generate_sampling_reproducible():
    from prime_amd.models.generate import generate

    torch.manual_seed(4)
    m = build_model("llama_test")
    prompt = torch.randint(0, 256, (1, 8))
    a = generate(m, prompt, 5, temperature=0.8, top_k=10, seed=7)
    b = generate(m, prompt, 5, temperature=0.8, top_k=10, seed=7)
    assert a.tolist() == b.tolist()


def test_large_configs_build_on_meta():
    """Shape-validate the big presets without allocating (meta device)."""
    from prime_amd.models import build_model

    for name, lo, hi in [("llama_8b", 7.5e9, 8.6e9),
                         ("intellect_10b", 9.5e9, 11e9),
                         ("llama_70b", 68e9, 73e9)]:
        with torch.device("meta"):
            m = build_model(name)
        n = sum(p.numel() for p in m.parameters())
        assert lo < n < hi, (name, n)


def test_evaluate_perplexity_token_file(tmp_path):
    import numpy as np

    from prime_amd.data import DataConfig
    from prime_amd.models.evaluate import evaluate_perplexity

    toks = np.random.default_rng(0).integers(0, 256, 5000).astype(np.uint16)
    p = tmp_path / "t.bin"
    toks.tofile(p)
    m = build_model("llama_test")
    res = evaluate_perplexity(
        m, DataConfig(kind="token_file", path=str(p), seq_len=64,
                      micro_batch_size=2),
        n_batches=3,
    )
    assert res["tokens"] == 3 * 2 * 64
    assert 1 < res["perplexity"] < 1e4


def _build_big_cheap(name, **over):
    """Construct a big-dim model without paying two full random-init
    passes (kaiming + normal over billions of params): meta-device
    construction + constant fill. Layout/shape tests only."""
    import torch

    from prime_amd.models import build_model

    with torch.device("meta"):
        m = build_model(name, **over)
    m = m.to_empty(device="cpu")
    with torch.no_grad():
        for prm in m.parameters():
            prm.fill_(0.01)
    m.reset_rope(torch.device("cpu"))
    return m


def test_70b_dims_block_shapes_cpu():
    """Full-dim Llama-70B block (dim 8192, 64 heads, GQA 8, inter 28672)
    at reduced layer count: fwd/bwd/optimizer shapes must hold. The full
    80-layer config is exercised on GPU via tools/sizing_70b.py."""
    import torch

    from prime_amd.models import build_model
    from prime_amd.parallel.flat import FlatParamSpace, FusedAdamW

    m = _build_big_cheap("llama_70b", n_layers=1, vocab_size=512, max_seq=128)
    assert m.cfg.dim == 8192 and m.cfg.intermediate == 28672
    assert m.cfg.n_heads == 64 and m.cfg.n_kv_heads == 8
    flat = FlatParamSpace(m)
    x = torch.randint(0, 512, (1, 32))
    y = torch.randint(0, 512, (1, 32))
    loss = m.loss(x, y)
    flat.zero_grad()
    loss.backward()
    assert float(loss) > 0
    assert float(flat.flat_grad.abs().sum()) > 0
    # optimizer pass over the full-dim flat space, on a shard-sized
    # slice (a full 1.8B-elem CPU reference AdamW takes minutes)
    from prime_amd import ops

    k = 1 << 20
    ops.reference.adamw_step(flat.master32[:k], flat.flat_grad[:k].float(),
                             torch.zeros(k), torch.zeros(k),
                             1e-4, 0.9, 0.95, 1e-8, 0.1, 1)


def _fsdp70_worker(rank, world):
    from prime_amd.models import build_model
    from prime_amd.parallel.fsdp import ShardedParamSpace
    from prime_amd.parallel.mesh import ElasticDeviceMesh, MeshConfig

    import torch

    torch.manual_seed(0)
    mesh = ElasticDeviceMesh(MeshConfig(worker_size=world))
    m = _build_big_cheap("llama_70b", n_layers=2, vocab_size=512, max_seq=128,
                          activation_checkpointing=True)
    flat = ShardedParamSpace(m, mesh)
    # shard layout: unit sizes match the 70B block param count
    blk_params = 8192 * (64 + 2 * 8) * 128 + 8192 * 8192 + 3 * 8192 * 28672 + 2 * 8192
    for u in flat.units:
        assert u.numel >= blk_params  # >=: per-param 64-elem alignment
        assert u.shard_len * world == u.numel_padded
    # layout-only: a single 70B-unit all_gather is ~7.3 GB over gloo
    # loopback and blows the CPU-lane budget; the gather path itself is
    # covered at test dims by test_fsdp.py and on GPU by the sizing run
    val = float(flat.flat_w[0])
    total = flat.numel_padded
    assert total * world * 2 / 1e9 > 3.0  # two 70B blocks of bf16 sharded
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()
    return {"val": val, "shard_len": flat.units[0].shard_len}


def test_70b_dims_fsdp_shard_layout_cpu():
    from tests.conftest import run_distributed

    outs = run_distributed(_fsdp70_worker, 2, timeout=600)
    assert abs(outs[0]["val"] - 0.01) < 1e-3  # shard holds the fill value
    assert outs[0]["shard_len"] == outs[1]["shard_len"]
