"""Model shape/state-dict/loss tests (CPU, eager reference ops)."""

import numpy as np
import torch

from bert_pytorch_amd.models import (
    BertForMaskedLM,
    BertForPreTraining,
    BertForQuestionAnswering,
    BertForSequenceClassification,
    BertForTokenClassification,
    BertPretrainingCriterion,
)


def _batch(tiny_config, bsz=3, seq=16, n_masked=2, seed=0):
    g = torch.Generator().manual_seed(seed)
    input_ids = torch.randint(0, tiny_config.vocab_size, (bsz, seq), generator=g)
    token_type = torch.zeros(bsz, seq, dtype=torch.long)
    mask = torch.ones(bsz, seq, dtype=torch.long)
    mask[:, seq - 4 :] = 0  # pad tail
    labels = torch.full((bsz, seq), -1, dtype=torch.long)
    for b in range(bsz):
        pos = torch.randperm(seq - 4, generator=g)[:n_masked]
        labels[b, pos] = torch.randint(
            0, tiny_config.vocab_size, (n_masked,), generator=g
        )
    nsp = torch.randint(0, 2, (bsz,), generator=g)
    return input_ids, token_type, mask, labels, nsp


def test_pretraining_forward_shapes(tiny_config):
    model = BertForPreTraining(tiny_config).eval()
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    scores, seq_rel = model(ids, tt, mask)
    assert scores.shape == (3, 16, tiny_config.vocab_size)
    assert seq_rel.shape == (3, 2)


def test_masked_gather_path_matches_full(tiny_config):
    """Gathered-MLM fast path gives the same loss as the full-scores path."""
    model = BertForPreTraining(tiny_config).eval()
    criterion = BertPretrainingCriterion(tiny_config.vocab_size)
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    with torch.no_grad():
        full_scores, full_rel = model(ids, tt, mask)
        loss_full = criterion(full_scores, full_rel, labels, nsp)
        g_scores, g_rel, g_labels = model(ids, tt, mask, masked_lm_labels=labels)
        loss_gather = criterion(g_scores, g_rel, g_labels, nsp)
    torch.testing.assert_close(loss_full, loss_gather, rtol=1e-5, atol=1e-5)


def test_state_dict_reference_names(tiny_config):
    model = BertForPreTraining(tiny_config)
    keys = set(model.state_dict().keys())
    expected = [
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
        "bert.encoder.layer.0.intermediate.dense_act.weight",
        "bert.encoder.layer.0.output.dense.weight",
        "bert.encoder.layer.0.output.LayerNorm.bias",
        "bert.pooler.dense_act.weight",
        "cls.predictions.transform.dense_act.weight",
        "cls.predictions.transform.LayerNorm.weight",
        "cls.predictions.decoder.weight",
        "cls.predictions.bias",
        "cls.seq_relationship.weight",
    ]
    for key in expected:
        assert key in keys, f"missing reference-style key {key}"
    assert not any("qkv_weight" in k for k in keys)


def test_state_dict_roundtrip_and_qkv_pack(tiny_config):
    m1 = BertForPreTraining(tiny_config)
    m2 = BertForPreTraining(tiny_config)
    m2.load_state_dict(m1.state_dict())
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    m1.eval(), m2.eval()
    with torch.no_grad():
        s1, r1 = m1(ids, tt, mask)
        s2, r2 = m2(ids, tt, mask)
    torch.testing.assert_close(s1, s2)
    torch.testing.assert_close(r1, r2)
    # packed qkv really took the split values
    layer = m1.bert.encoder.layer[0].attention.self
    sd = m1.state_dict()
    h = tiny_config.hidden_size
    torch.testing.assert_close(
        layer.qkv_weight[:h],
        sd["bert.encoder.layer.0.attention.self.query.weight"],
    )


def test_decoder_tied_to_embeddings(tiny_config):
    model = BertForPreTraining(tiny_config)
    assert (
        model.cls.predictions.decoder.weight.data_ptr()
        == model.bert.embeddings.word_embeddings.weight.data_ptr()
    )


def test_backward_produces_grads(tiny_config):
    model = BertForPreTraining(tiny_config).train()
    criterion = BertPretrainingCriterion(tiny_config.vocab_size)
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    scores, rel, glabels = model(ids, tt, mask, masked_lm_labels=labels)
    loss = criterion(scores, rel, glabels, nsp)
    loss.backward()
    for name, p in model.named_parameters():
        assert p.grad is not None, f"no grad for {name}"
        assert torch.isfinite(p.grad).all(), f"non-finite grad for {name}"


def test_task_heads_forward(tiny_config):
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    qa = BertForQuestionAnswering(tiny_config).eval()
    start, end = qa(ids, tt, mask)
    assert start.shape == (3, 16) and end.shape == (3, 16)
    tc = BertForTokenClassification(tiny_config, num_labels=5).eval()
    assert tc(ids, tt, mask).shape == (3, 16, 5)
    sc = BertForSequenceClassification(tiny_config, num_labels=3).eval()
    assert sc(ids, tt, mask).shape == (3, 3)
    mlm = BertForMaskedLM(tiny_config).eval()
    assert mlm(ids, tt, mask).shape == (3, 16, tiny_config.vocab_size)


def test_roberta_mode_no_nsp(tiny_config):
    tiny_config.next_sentence = False
    model = BertForPreTraining(tiny_config).eval()
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    scores, rel = model(ids, None, mask)
    assert rel is None
    assert model.bert.pooler is None
    assert model.bert.embeddings.token_type_embeddings is None


def test_activation_checkpointing_matches(tiny_config):
    model = BertForPreTraining(tiny_config).train()
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    criterion = BertPretrainingCriterion(tiny_config.vocab_size)
    torch.manual_seed(0)
    model_dropoutless = model.eval()  # avoid dropout nondeterminism
    with torch.no_grad():
        s1, _ = model_dropoutless(ids, tt, mask)
    model.checkpoint_activations(True)
    with torch.no_grad():
        s2, _ = model_dropoutless(ids, tt, mask)
    torch.testing.assert_close(s1, s2)
