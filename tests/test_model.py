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


def test_padded_gather_matches_nonzero_gather(tiny_config):
    """Sync-free padded max-pred gather = variable nonzero gather, in loss
    AND gradients (padding rows carry label -1 -> zero CE gradient)."""
    criterion = BertPretrainingCriterion(tiny_config.vocab_size)
    ids, tt, mask, labels, nsp = _batch(tiny_config)

    torch.manual_seed(7)
    m1 = BertForPreTraining(tiny_config).train()
    m2 = BertForPreTraining(tiny_config).train()
    m2.load_state_dict(m1.state_dict())

    torch.manual_seed(11)  # identical dropout draws for both forwards
    s1, r1, l1 = m1(ids, tt, mask, masked_lm_labels=labels)
    loss1 = criterion(s1, r1, l1, nsp)
    loss1.backward()

    torch.manual_seed(11)
    s2, r2, l2 = m2(
        ids, tt, mask, masked_lm_labels=labels, max_predictions_per_seq=5
    )
    # fixed row budget: bsz * max_pred
    assert s2.shape[0] == ids.shape[0] * 5
    loss2 = criterion(s2, r2, l2, nsp)
    loss2.backward()

    torch.testing.assert_close(loss1, loss2, rtol=1e-5, atol=1e-6)
    for (n1, p1), (n2, p2) in zip(
        m1.named_parameters(), m2.named_parameters()
    ):
        assert n1 == n2
        if p1.grad is None:
            assert p2.grad is None or torch.all(p2.grad == 0)
            continue
        torch.testing.assert_close(p1.grad, p2.grad, rtol=1e-4, atol=1e-6,
                                   msg=lambda m, n=n1: f"{n}: {m}")


def test_fused_mlm_loss_path_matches_scores_path(tiny_config):
    """compute_mlm_loss=True returns the scalar MLM loss in place of the
    scores and the criterion's 0-dim branch yields the same total loss
    (on CPU the fused op runs its eager fp32 oracle)."""
    model = BertForPreTraining(tiny_config).eval()
    criterion = BertPretrainingCriterion(tiny_config.vocab_size)
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    with torch.no_grad():
        scores, rel, glabels = model(ids, tt, mask, masked_lm_labels=labels)
        loss_scores = criterion(scores, rel, glabels, nsp)
        fused, rel2, gl2 = model(
            ids, tt, mask, masked_lm_labels=labels, compute_mlm_loss=True
        )
        assert fused.dim() == 0
        loss_fused = criterion(fused, rel2, gl2, nsp)
    torch.testing.assert_close(loss_scores, loss_fused, rtol=1e-5, atol=1e-5)


def test_fused_mlm_loss_backward_cpu(tiny_config):
    """Gradients flow through the fused-loss path (tied decoder weight,
    bias, and the encoder) on the CPU oracle."""
    model = BertForPreTraining(tiny_config)
    criterion = BertPretrainingCriterion(tiny_config.vocab_size)
    ids, tt, mask, labels, nsp = _batch(tiny_config)
    loss, rel, gl = model(
        ids, tt, mask, masked_lm_labels=labels, compute_mlm_loss=True,
        max_predictions_per_seq=4,
    )
    criterion(loss, rel, gl, nsp).backward()
    emb = model.bert.embeddings.word_embeddings.weight
    assert emb.grad is not None and torch.isfinite(emb.grad).all()
    assert model.cls.predictions.bias.grad is not None
