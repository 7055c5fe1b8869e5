import torch

from pdnlp_amd.config import BertConfig
from pdnlp_amd.models import (BertForSequenceClassification,
                              RobertaForSequenceClassification)


def _batch(cfg, B=2, S=16):
    g = torch.Generator().manual_seed(0)
    ids = torch.randint(0, cfg.vocab_size, (B, S), generator=g)
    mask = torch.ones(B, S, dtype=torch.long)
    mask[0, S // 2:] = 0
    type_ids = torch.zeros(B, S, dtype=torch.long)
    labels = torch.randint(0, cfg.num_labels, (B,), generator=g)
    return ids, mask, type_ids, labels


def test_forward_shapes(tiny_cfg):
    model = BertForSequenceClassification(tiny_cfg)
    ids, mask, type_ids, labels = _batch(tiny_cfg)
    out = model(ids, mask, type_ids, labels)
    assert out.logits.shape == (2, tiny_cfg.num_labels)
    assert out.loss.dim() == 0
    assert out[1] is out.logits  # reference indexes output[1]


def test_state_dict_matches_hf_layout(tiny_cfg):
    model = BertForSequenceClassification(tiny_cfg)
    keys = set(model.state_dict().keys())
    expected = {
        "bert.embeddings.word_embeddings.weight",
        "bert.embeddings.position_embeddings.weight",
        "bert.embeddings.token_type_embeddings.weight",
        "bert.embeddings.LayerNorm.weight",
        "bert.embeddings.LayerNorm.bias",
        "bert.encoder.layer.0.attention.self.query.weight",
        "bert.encoder.layer.0.attention.self.key.bias",
        "bert.encoder.layer.0.attention.self.value.weight",
        "bert.encoder.layer.0.attention.output.dense.weight",
        "bert.encoder.layer.0.attention.output.LayerNorm.weight",
        "bert.encoder.layer.0.intermediate.dense.weight",
        "bert.encoder.layer.0.output.dense.weight",
        "bert.encoder.layer.0.output.LayerNorm.bias",
        "bert.pooler.dense.weight",
        "classifier.weight",
        "classifier.bias",
    }
    assert expected <= keys
    assert not any(k.startswith("module.") for k in keys)


def test_matches_hf_transformers_numerics(tiny_cfg):
    """Our model loaded with an HF model's weights produces the same logits
    (validates the op pipeline against transformers' BERT)."""
    transformers = __import__("transformers")
    hf_cfg = transformers.BertConfig(
        vocab_size=tiny_cfg.vocab_size, hidden_size=tiny_cfg.hidden_size,
        num_hidden_layers=tiny_cfg.num_hidden_layers,
        num_attention_heads=tiny_cfg.num_attention_heads,
        intermediate_size=tiny_cfg.intermediate_size,
        max_position_embeddings=tiny_cfg.max_position_embeddings,
        num_labels=tiny_cfg.num_labels,
        hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
        attn_implementation="eager",
    )
    hf = transformers.BertForSequenceClassification(hf_cfg).eval()
    ours = BertForSequenceClassification(tiny_cfg).eval()
    missing, unexpected = ours.load_state_dict(hf.state_dict(), strict=False)
    assert not missing, missing
    ids, mask, type_ids, labels = _batch(tiny_cfg)
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=mask, token_type_ids=type_ids,
                 labels=labels)
        got = ours(ids, mask, type_ids, labels)
    torch.testing.assert_close(got.logits, ref.logits, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(got.loss, ref.loss, rtol=1e-4, atol=1e-4)


def test_backward_produces_grads(tiny_cfg):
    model = BertForSequenceClassification(tiny_cfg)
    ids, mask, type_ids, labels = _batch(tiny_cfg)
    out = model(ids, mask, type_ids, labels)
    out.loss.backward()
    for n, p in model.named_parameters():
        assert p.grad is not None, n
        assert torch.isfinite(p.grad).all(), n


def test_overfit_tiny(tiny_cfg):
    """Loss decreases on a fixed batch — end-to-end learning signal."""
    from pdnlp_amd.ops.adamw import build_optimizer
    model = BertForSequenceClassification(tiny_cfg)
    opt = build_optimizer(model, lr=1e-3)
    ids, mask, type_ids, labels = _batch(tiny_cfg, B=8)
    first = None
    for _ in range(20):
        out = model(ids, mask, type_ids, labels)
        opt.zero_grad(set_to_none=False)
        out.loss.backward()
        opt.step()
        if first is None:
            first = out.loss.item()
    assert out.loss.item() < first * 0.6, (first, out.loss.item())


def test_roberta_layout():
    cfg = BertConfig.roberta_base()
    cfg.num_hidden_layers = 1
    cfg.hidden_size = 64
    cfg.num_attention_heads = 4
    cfg.intermediate_size = 128
    cfg.vocab_size = 512
    cfg.max_position_embeddings = 66
    model = RobertaForSequenceClassification(cfg)
    keys = set(model.state_dict().keys())
    assert "roberta.embeddings.word_embeddings.weight" in keys
    assert "classifier.out_proj.weight" in keys
    assert "roberta.pooler.dense.weight" not in str(keys)
    ids = torch.randint(2, 500, (2, 12))
    out = model(ids, torch.ones(2, 12, dtype=torch.long),
                labels=torch.tensor([0, 1]))
    assert out.logits.shape == (2, cfg.num_labels)
    out.loss.backward()


def test_gradient_checkpointing(tiny_cfg):
    model = BertForSequenceClassification(tiny_cfg)
    model.gradient_checkpointing_enable()
    ids, mask, type_ids, labels = _batch(tiny_cfg)
    out = model(ids, mask, type_ids, labels)
    out.loss.backward()
    assert all(p.grad is not None for p in model.parameters())
