"""T5 model family tests (CPU; GPU variants marked gpu).

Covers the model behaviors the reference workshop depends on: training
forward+loss (cell-43), greedy generate with KV cache (cell-49/predictor.py:102),
HF save_pretrained checkpoint layout."""
import os

import pytest
import torch

from mi355.models.t5.configuration import T5Config, t5_config
from mi355.models.t5.modeling import T5ForConditionalGeneration

TINY = dict(num_layers=2, num_decoder_layers=2, d_model=64, d_ff=128,
            num_heads=4, d_kv=16, vocab_size=512)


def tiny_model(**over):
    kw = dict(TINY)
    kw.update(over)
    cfg = t5_config("flan-t5-small", **kw)
    torch.manual_seed(0)
    return T5ForConditionalGeneration(cfg)


def test_forward_loss_backward():
    m = tiny_model()
    ids = torch.randint(0, 512, (2, 16))
    out = m(input_ids=ids, labels=ids.clone())
    assert out.logits.shape == (2, 16, 512)
    assert out.loss.item() > 0
    out.loss.backward()
    assert m.shared.weight.grad is not None
    assert m.encoder.block[0].layer[0].SelfAttention.relative_attention_bias.weight.grad is not None


def test_label_padding_ignored():
    m = tiny_model().eval()
    ids = torch.randint(0, 512, (1, 8))
    labels = ids.clone()
    with torch.no_grad():
        l1 = m(input_ids=ids, labels=labels).loss
        labels2 = torch.cat([labels, torch.full((1, 4), -100)], dim=1)
        dec2 = m._shift_right(labels2)
        l2 = m(input_ids=ids, decoder_input_ids=dec2, labels=labels2).loss
    # ignored positions change nothing except via decoder inputs; the first
    # 8 positions dominate — losses must be close
    assert l1.item() == pytest.approx(l2.item(), rel=0.25)


def test_generate_matches_stepwise_scoring():
    """KV-cache greedy decode == full-sequence scoring argmax."""
    m = tiny_model().eval()
    ids = torch.randint(0, 512, (2, 12))
    gen = m.generate(input_ids=ids, max_new_tokens=6, min_new_tokens=6)
    # re-score without cache: feed generated prefix through full forward
    dec_in = torch.cat([torch.zeros(2, 1, dtype=torch.long), gen[:, :-1]], dim=1)
    with torch.no_grad():
        logits = m(input_ids=ids, decoder_input_ids=dec_in).logits
    rescored = logits.argmax(-1)
    # every position where generation hadn't finished must match
    assert (rescored == gen).float().mean().item() > 0.95


def test_generate_respects_attention_mask():
    m = tiny_model().eval()
    ids = torch.randint(1, 512, (1, 8))
    padded = torch.cat([ids, torch.zeros(1, 4, dtype=torch.long)], dim=1)
    mask = torch.cat([torch.ones(1, 8), torch.zeros(1, 4)], dim=1)
    g1 = m.generate(input_ids=ids, max_new_tokens=4, min_new_tokens=4)
    g2 = m.generate(input_ids=padded, attention_mask=mask, max_new_tokens=4,
                    min_new_tokens=4)
    assert torch.equal(g1, g2), "padding with mask must not change generation"


def test_save_pretrained_layout(tmp_path):
    m = tiny_model()
    m.save_pretrained(str(tmp_path))
    assert os.path.exists(tmp_path / "config.json")
    assert os.path.exists(tmp_path / "model.safetensors")
    cfg = T5Config.from_pretrained(str(tmp_path))
    assert cfg.d_model == 64
    m2 = T5ForConditionalGeneration.from_pretrained(str(tmp_path))
    for (n1, p1), (n2, p2) in zip(m.named_parameters(), m2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2), n1


def test_tied_embeddings_variant(tmp_path):
    m = tiny_model(tie_word_embeddings=True)
    assert m.lm_head.weight is m.shared.weight
    m.save_pretrained(str(tmp_path))
    from safetensors.torch import load_file
    st = load_file(str(tmp_path / "model.safetensors"))
    assert "lm_head.weight" not in st
    m2 = T5ForConditionalGeneration.from_pretrained(str(tmp_path))
    assert m2.lm_head.weight is m2.shared.weight


def test_presets_match_reference_shapes():
    base = t5_config("flan-t5-base")
    assert (base.d_model, base.d_ff, base.num_layers, base.num_heads) == (768, 2048, 12, 12)
    assert base.vocab_size == 32128 and base.d_kv == 64
    small = t5_config("google/flan-t5-small")
    assert small.d_model == 512


@pytest.mark.gpu
def test_model_gpu_train_step_bf16():
    m = tiny_model(d_model=128, d_kv=64, num_heads=2, d_ff=256).to("cuda", torch.bfloat16)
    ids = torch.randint(0, 512, (2, 64), device="cuda")
    out = m(input_ids=ids, labels=ids.clone())
    out.loss.backward()
    torch.cuda.synchronize()
    assert out.loss.item() > 0


@pytest.mark.gpu
def test_model_gpu_generate_matches_cpu():
    torch.manual_seed(0)
    m = tiny_model(d_model=128, d_kv=64, num_heads=2, d_ff=256, dropout_rate=0.0).eval()
    ids = torch.randint(0, 512, (2, 12))
    cpu_gen = m.generate(input_ids=ids, max_new_tokens=5, min_new_tokens=5)
    mg = m.to("cuda", torch.bfloat16)
    gpu_gen = mg.generate(input_ids=ids.cuda(), max_new_tokens=5, min_new_tokens=5)
    # argmax over near-equal logits can differ on a few tokens; demand majority
    agree = (cpu_gen == gpu_gen.cpu()).float().mean().item()
    assert agree >= 0.6, f"CPU/GPU generation diverged badly: {agree}"


def test_xl_presets_build_on_meta():
    """flan-t5-xl / -xxl presets construct with the published parameter
    counts (meta device — no memory; the xxl fine-tune state fits one
    MI355X's 288 GB HBM)."""
    import torch

    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    expect = {"flan-t5-xl": (2.7e9, 3.1e9), "flan-t5-xxl": (10.5e9, 11.5e9)}
    for name, (lo, hi) in expect.items():
        with torch.device("meta"):
            m = T5ForConditionalGeneration(t5_config(name))
        n = sum(p.numel() for p in m.parameters())
        assert lo < n < hi, (name, n)
