"""GNMT workload: model, revert op, data pipeline, inference, BLEU."""

import torch

from ddlbench_amd.data.translation import (BucketingSampler,
                                           SyntheticTranslationDataset,
                                           collate_translation)
from ddlbench_amd.models.gnmt import (EOS, GNMT, LabelSmoothingLoss, PAD,
                                      revert_varlen, varlen_mask)
from ddlbench_amd.translation import Translator, bleu


def _tiny_gnmt():
    torch.manual_seed(0)
    return GNMT(vocab_size=64, hidden_size=32, num_layers=4, dropout=0.0)


def test_revert_varlen_self_inverse_and_grad():
    torch.manual_seed(0)
    x = torch.randn(6, 3, 4, requires_grad=True)
    lengths = torch.tensor([6, 4, 1])
    y = revert_varlen(x, lengths)
    # valid prefix reversed, padding zeroed
    torch.testing.assert_close(y[:4, 1], x[:4, 1].flip(0))
    assert torch.all(y[4:, 1] == 0)
    # self-inverse on the valid region
    z = revert_varlen(y, lengths)
    torch.testing.assert_close(z[:4, 1], x[:4, 1])
    # gradient flows (backward is the same op)
    y.sum().backward()
    assert torch.all(x.grad[:6, 0] == 1)
    assert torch.all(x.grad[4:, 1] == 0)


def test_varlen_mask():
    m = varlen_mask(torch.tensor([3, 1]), 4)
    assert m.shape == (4, 2)
    assert m[:, 0].tolist() == [True, True, True, False]
    assert m[:, 1].tolist() == [True, False, False, False]


def test_gnmt_forward_backward():
    model = _tiny_gnmt()
    src = torch.randint(3, 64, (7, 2))
    src_len = torch.tensor([7, 5])
    tgt = torch.randint(3, 64, (6, 2))
    logits = model(src, src_len, tgt)
    assert logits.shape == (6, 2, 64)
    loss = LabelSmoothingLoss()(logits, tgt)
    loss.backward()
    assert torch.isfinite(loss)


def test_label_smoothing_ignores_pad():
    lf = LabelSmoothingLoss(0.1)
    logits = torch.randn(4, 2, 8)
    tgt = torch.randint(3, 8, (4, 2))
    tgt_pad = tgt.clone()
    tgt_pad[2:, 1] = PAD
    # changing logits at padded positions must not change the loss
    l1 = lf(logits, tgt_pad)
    logits2 = logits.clone()
    logits2[2:, 1] += 100.0
    l2 = lf(logits2, tgt_pad)
    torch.testing.assert_close(l1, l2)


def test_greedy_and_beam_decode():
    model = _tiny_gnmt().eval()
    src = torch.randint(3, 64, (5, 3))
    src_len = torch.tensor([5, 4, 2])
    tr = Translator(model, max_len=7, beam_size=3)
    g = tr.greedy(src, src_len)
    assert g.dim() == 2 and g.size(1) == 3 and g.size(0) <= 7
    b = tr.beam(src, src_len)
    assert b.dim() == 2 and b.size(1) == 3


def test_bleu_sanity():
    ref = [[5, 6, 7, 8, 9, 10]]
    assert bleu(ref, ref) == 100.0
    assert bleu([[11, 12, 13, 14, 15, 16]], ref) == 0.0
    partial = bleu([[5, 6, 7, 8, 20, 21]], ref)
    assert 0.0 < partial < 100.0


def test_bucketing_sampler_shards_and_buckets():
    ds = SyntheticTranslationDataset(200, vocab_size=50, seed=1)
    s0 = BucketingSampler(ds, 8, world_size=2, rank=0, seed=3)
    s1 = BucketingSampler(ds, 8, world_size=2, rank=1, seed=3)
    b0 = list(iter(s0))
    b1 = list(iter(s1))
    assert len(b0) == len(b1)
    flat0 = {i for b in b0 for i in b}
    flat1 = {i for b in b1 for i in b}
    assert flat0.isdisjoint(flat1)
    # bucketing: within-batch length spread far below global spread
    lens = [ds.src_len(i) for i in range(len(ds))]
    spread = [max(ds.src_len(i) for i in b) - min(ds.src_len(i) for i in b)
              for b in b0]
    assert sum(spread) / len(spread) < (max(lens) - min(lens)) / 2


def test_collate_shapes():
    ds = SyntheticTranslationDataset(8, vocab_size=30, seed=2)
    src, src_len, tgt_in, tgt_out = collate_translation(
        [ds[i] for i in range(4)])
    assert src.size(1) == 4 and src.size(0) == int(src_len.max())
    assert tgt_in.shape == tgt_out.shape
    # teacher forcing alignment: tgt_out is tgt_in shifted by one
    ds0 = ds[0][1]
    torch.testing.assert_close(tgt_in[1:len(ds0) - 1, 0],
                               tgt_out[:len(ds0) - 2, 0])


def test_gnmt_runner_smoke():
    from ddlbench_amd.gnmt_runner import run_gnmt
    res = run_gnmt(epochs=1, batch_size=4, dataset_size=16, vocab=48,
                   hidden=16, layers=4, device="cpu", log_interval=0,
                   max_len=12, bleu_batches=1)
    assert res["samples_per_sec"] > 0
    assert torch.isfinite(torch.tensor(res["valid_loss"]))


def test_gnmt_pipeline_units_match_full_model():
    """The flattened pipeline units compute the same logits as GNMT."""
    from ddlbench_amd.models.gnmt import gnmt_pipeline_units
    torch.manual_seed(0)
    m = GNMT(vocab_size=64, hidden_size=32, num_layers=4,
             dropout=0.0).eval()
    units = gnmt_pipeline_units(m)
    src = torch.randint(3, 64, (7, 2))
    src_len = torch.tensor([7, 5])
    tgt = torch.randint(3, 64, (6, 2))
    with torch.no_grad():
        ref = m(src, src_len, tgt)
        xs = (src, src_len, tgt)
        for u in units:
            out = u(*xs)
            xs = (out,) if torch.is_tensor(out) else out
    torch.testing.assert_close(xs[0], ref, rtol=1e-5, atol=1e-5)


def test_gnmt_edge_specs_shapes():
    from ddlbench_amd.models.gnmt import (gnmt_edge_specs,
                                          gnmt_pipeline_units)
    m = GNMT(vocab_size=64, hidden_size=32, num_layers=4, dropout=0.0)
    units = gnmt_pipeline_units(m)
    specs = gnmt_edge_specs(m, len(units), 7, 6, 2, torch.float32)
    assert len(specs) == len(units)
    # run through the chain, check shapes against specs
    xs = (torch.randint(3, 64, (7, 2)), torch.tensor([7, 5]),
          torch.randint(3, 64, (6, 2)))
    with torch.no_grad():
        for u, sp in zip(units, specs):
            out = u(*xs)
            xs = (out,) if torch.is_tensor(out) else out
            assert len(xs) == len(sp)
            for t, spec in zip(xs, sp):
                assert tuple(t.shape) == tuple(spec.shape), (t.shape, spec)


def test_tokenizer_roundtrip(tmp_path):
    """Reference seq2seq/data/tokenizer.py position: specials, vocab
    build, encode/decode, save/load."""
    from ddlbench_amd.data.tokenizer import (BOS, EOS, PAD, Tokenizer,
                                             UNK)
    lines = ["the cat sat", "the dog ran", "a cat ran fast"]
    tok = Tokenizer.build(lines)
    ids = tok.encode("the cat flew")
    assert ids[0] == BOS and ids[-1] == EOS
    assert UNK in ids  # 'flew' unseen
    assert tok.decode(ids) == "the cat <unk>"
    p = tmp_path / "vocab.txt"
    tok.save(str(p))
    tok2 = Tokenizer.load(str(p))
    assert tok2.encode("the cat flew") == ids
    assert tok.stoi["<pad>"] == PAD


def test_text_translation_dataset(tmp_path):
    """Reference seq2seq/data/dataset.py position: padded parallel
    corpus with true lengths, same item contract as the synthetic
    stream (works with BucketingSampler)."""
    from ddlbench_amd.data.tokenizer import PAD, TextTranslationDataset
    from ddlbench_amd.data.translation import BucketingSampler
    (tmp_path / "train.src").write_text(
        "the cat sat on the mat\nhello world\na b c d e\n")
    (tmp_path / "train.tgt").write_text(
        "die katze sass auf der matte\nhallo welt\nf g h i j\n")
    ds = TextTranslationDataset(str(tmp_path), "train")
    assert len(ds) == 3
    src, tgt = ds[0]
    assert src.dtype == torch.long and tgt.dtype == torch.long
    assert ds.src_len(0) == 8  # 6 words + BOS + EOS
    assert len(src) == 8 and (src != PAD).all()
    # bucketing sampler consumes the same src_len contract
    bs = BucketingSampler(ds, batch_size=2, seed=0)
    batches = list(iter(bs))
    assert sum(len(b) for b in batches) >= 2


def test_text_translation_dataset_missing(tmp_path):
    import pytest as _pytest
    from ddlbench_amd.data.tokenizer import TextTranslationDataset
    with _pytest.raises(FileNotFoundError):
        TextTranslationDataset(str(tmp_path), "train")


def test_run_gnmt_real_corpus(tmp_path):
    """run_gnmt trains from an on-disk parallel corpus (data_dir)."""
    from ddlbench_amd.gnmt_runner import run_gnmt
    src_lines = ["a b c", "b c d e", "c d", "a a b b", "e d c b a",
                 "a c e", "b d", "c c c c"]
    tgt_lines = ["x y", "y z w", "z x", "x x y", "w z y x",
                 "x z w", "y w", "z z z"]
    (tmp_path / "train.src").write_text("\n".join(src_lines) + "\n")
    (tmp_path / "train.tgt").write_text("\n".join(tgt_lines) + "\n")
    res = run_gnmt(epochs=1, batch_size=4, hidden=16, layers=2,
                   device="cpu", log_interval=0, bleu_batches=1,
                   data_dir=str(tmp_path))
    assert res["samples_per_sec"] > 0


def test_static_distributed_sampler():
    from ddlbench_amd.data.translation import (StaticDistributedSampler,
                                               SyntheticTranslationDataset)
    ds = SyntheticTranslationDataset(10, 100, max_len=8)
    shards = [list(StaticDistributedSampler(ds, 2, 3, r))
              for r in range(3)]
    flat = sorted(i for shard in shards for b in shard for i in b)
    assert flat == list(range(10))  # exact cover, no dupes
    # contiguous, deterministic
    assert shards[0][0] == [0, 1]
    assert len(StaticDistributedSampler(ds, 4, 1, 0)) == 3


def test_gnmt_runner_adam():
    """optimizer="adam" — the reference GNMT optimizer — trains."""
    from ddlbench_amd.gnmt_runner import run_gnmt
    res = run_gnmt(epochs=1, batch_size=4, dataset_size=8, vocab=48,
                   hidden=16, layers=2, device="cpu", log_interval=0,
                   bleu_batches=1, optimizer="adam")
    assert res["samples_per_sec"] > 0
