"""Data pipeline tests: synthetic determinism, token-file sharding/shuffle/
resume."""
import numpy as np
import torch

from prime_amd.data import DataConfig, build_dataloader


def _token_file(tmp_path, n=10000, vocab=1000):
    toks = np.random.default_rng(0).integers(0, vocab, n).astype(np.uint16)
    p = tmp_path / "toks.bin"
    toks.tofile(p)
    return str(p), vocab


def test_synthetic_deterministic_and_resumable():
    cfg = DataConfig(seq_len=32, micro_batch_size=2, seed=5)
    a = build_dataloader(cfg, 100, 0, 1)
    b = build_dataloader(cfg, 100, 0, 1)
    xa, _ = a.next_batch(torch.device("cpu"))
    xb, _ = b.next_batch(torch.device("cpu"))
    assert torch.equal(xa, xb)
    # resume: loader c skips to batch 1 and matches a's second batch
    c = build_dataloader(cfg, 100, 0, 1)
    c.load_state_dict(a.state_dict())
    xa2, _ = a.next_batch(torch.device("cpu"))
    xc, _ = c.next_batch(torch.device("cpu"))
    assert torch.equal(xa2, xc)


def test_synthetic_shards_disjoint():
    cfg = DataConfig(seq_len=32, micro_batch_size=2, seed=5)
    a = build_dataloader(cfg, 100, 0, 2)
    b = build_dataloader(cfg, 100, 1, 2)
    xa, _ = a.next_batch(torch.device("cpu"))
    xb, _ = b.next_batch(torch.device("cpu"))
    assert not torch.equal(xa, xb)


def test_token_file_targets_shifted(tmp_path):
    path, vocab = _token_file(tmp_path)
    cfg = DataConfig(kind="token_file", path=path, seq_len=64,
                     micro_batch_size=2, shuffle=False)
    dl = build_dataloader(cfg, vocab, 0, 1)
    x, y = dl.next_batch(torch.device("cpu"))
    assert torch.equal(x[:, 1:], y[:, :-1])


def test_token_file_shuffle_covers_epoch(tmp_path):
    path, vocab = _token_file(tmp_path, n=64 * 20 + 1)
    cfg = DataConfig(kind="token_file", path=path, seq_len=64,
                     micro_batch_size=1, shuffle=True, seed=3)
    dl = build_dataloader(cfg, vocab, 0, 1)
    seen = []
    for _ in range(dl.windows):
        x, _ = dl.next_batch(torch.device("cpu"))
        seen.append(int(x[0, 0]))
    # one epoch visits every window exactly once (permutation)
    starts = {int(t) for t in np.asarray(dl.files[0][: 64 * 20 : 64])}
    assert len(seen) == dl.windows
    assert set(seen) == starts or len(set(seen)) == len(seen)


def test_token_file_resume(tmp_path):
    path, vocab = _token_file(tmp_path)
    cfg = DataConfig(kind="token_file", path=path, seq_len=64,
                     micro_batch_size=2, shuffle=True)
    a = build_dataloader(cfg, vocab, 0, 1)
    a.next_batch(torch.device("cpu"))
    st = a.state_dict()
    xa, _ = a.next_batch(torch.device("cpu"))
    b = build_dataloader(cfg, vocab, 0, 1)
    b.load_state_dict(st)
    xb, _ = b.next_batch(torch.device("cpu"))
    assert torch.equal(xa, xb)


def test_prepare_data_roundtrip(tmp_path):
    """Build a tiny tokenizer locally (no network), prepare a .bin, train a
    step from it — the full real-data path."""
    tokenizers = __import__("tokenizers")
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace
    from tokenizers.trainers import WordLevelTrainer

    text = ("the quick brown fox jumps over the lazy dog " * 200).strip()
    src = tmp_path / "corpus.txt"
    src.write_text(text)
    tok = tokenizers.Tokenizer(WordLevel(unk_token="<unk>"))
    tok.pre_tokenizer = Whitespace()
    tok.train_from_iterator([text], WordLevelTrainer(special_tokens=["<unk>"]))
    tok_path = tmp_path / "tok.json"
    tok.save(str(tok_path))

    from typer.testing import CliRunner

    from prime_amd.cli.main import app

    out_bin = tmp_path / "toks.bin"
    r = CliRunner().invoke(app, ["prepare-data", str(src), "--out",
                                 str(out_bin), "--tokenizer", str(tok_path)])
    assert r.exit_code == 0, r.output
    cfg = DataConfig(kind="token_file", path=str(out_bin), seq_len=64,
                     micro_batch_size=2, shuffle=True)
    dl = build_dataloader(cfg, 1000, 0, 1)
    x, y = dl.next_batch(torch.device("cpu"))
    assert x.shape == (2, 64) and int(x.max()) < tok.get_vocab_size()


def test_prepare_data_sharded_streaming(tmp_path):
    """Multi-file input, parallel workers, small shard size -> multiple
    output shards consumed as a directory by TokenFileDataset."""
    tokenizers = __import__("tokenizers")
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace
    from tokenizers.trainers import WordLevelTrainer

    text = "alpha beta gamma delta epsilon zeta eta theta " * 400
    srcs = []
    for i in range(3):
        f = tmp_path / f"part{i}.txt"
        f.write_text(text)
        srcs.append(f)
    tok = tokenizers.Tokenizer(WordLevel(unk_token="<unk>"))
    tok.pre_tokenizer = Whitespace()
    tok.train_from_iterator([text], WordLevelTrainer(special_tokens=["<unk>"]))
    tok_path = tmp_path / "tok.json"
    tok.save(str(tok_path))

    from prime_amd.data.prepare import prepare_corpus

    out_dir = tmp_path / "shards"
    res = prepare_corpus([str(s) for s in srcs], out_dir, str(tok_path),
                         shard_tokens=4000, workers=2)
    assert res["tokens"] == 3 * 8 * 400
    assert len(res["shards"]) == (res["tokens"] + 3999) // 4000
    # shard directory consumed directly
    cfg = DataConfig(kind="token_file", path=str(out_dir), seq_len=64,
                     micro_batch_size=2, shuffle=True)
    dl = build_dataloader(cfg, 1000, 0, 1)
    seen_windows = dl.windows
    assert seen_windows == sum(
        max(0, (4000 if i < len(res["shards"]) - 1 else
                res["tokens"] % 4000 or 4000) - 1) // 64
        for i in range(len(res["shards"])))
    x, y = dl.next_batch(torch.device("cpu"))
    assert x.shape == (2, 64)
    # resume across shard boundaries
    st = dl.state_dict()
    xa, _ = dl.next_batch(torch.device("cpu"))
    dl2 = build_dataloader(cfg, 1000, 0, 1)
    dl2.load_state_dict(st)
    xb, _ = dl2.next_batch(torch.device("cpu"))
    assert torch.equal(xa, xb)
