"""MI355X-native BERT encoder and task heads.

A from-scratch re-design of the reference model zoo
(/root/reference/src/modeling.py — classes mapped in SURVEY.md §2.3) for
CDNA4: the residual junctions, embeddings, GELU and attention run as
fused HIP kernels (bert_pytorch_amd.ops); GEMMs go through hipBLASLt via
``F.linear`` in bf16 autocast; QKV is one packed [3H, H] GEMM.

State-dict compatibility: parameter names match the reference exactly
(``bert.encoder.layer.N.attention.self.query.weight`` …) so checkpoints
flow between pretraining and the finetune runners the same way
(reference: run_squad.py:961, run_ner.py:225-227). The packed QKV weight
is split/merged in custom state-dict hooks.
"""

from __future__ import annotations

import logging
import math
from typing import Optional

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops
from ..config import BertConfig

logger = logging.getLogger(__name__)


def _autocast_dtype(ref: torch.Tensor) -> torch.dtype:
    if ref.is_cuda and torch.is_autocast_enabled():
        return torch.get_autocast_gpu_dtype()
    return ref.dtype


class LinearActivation(nn.Module):
    """Linear + fused bias-activation (reference: src/modeling.py:141-185).

    For GELU the bias add and activation run as one HIP kernel on the
    GEMM output; for tanh (pooler) the eager op is negligible.
    """

    def __init__(self, in_features: int, out_features: int, act: str = "gelu"):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.act = act
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features))
        self.reset_parameters()

    def reset_parameters(self) -> None:
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        bound = 1 / math.sqrt(self.in_features)
        nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # activation names per the reference ACT2FN (src/modeling.py:139):
        # gelu / bias_gelu are the fused HIP path; tanh/relu/swish eager.
        if self.act in ("gelu", "bias_gelu"):
            y = ops.linear_nobias(x, self.weight)
            return ops.fused_bias_gelu(y, self.bias)
        y = F.linear(x, self.weight, self.bias)
        if self.act in ("tanh", "bias_tanh"):
            return torch.tanh(y)
        if self.act == "relu":
            return F.relu(y)
        if self.act == "swish":
            return y * torch.sigmoid(y)
        raise ValueError(f"unsupported activation {self.act!r}")


class BertEmbeddings(nn.Module):
    """word + position (+ token-type) embeddings -> LN -> dropout, fused.

    Reference: src/modeling.py:338-373 (token-type table skipped for
    RoBERTa via ``next_sentence=False``, :345-348).
    """

    def __init__(self, config: BertConfig):
        super().__init__()
        self.word_embeddings = nn.Embedding(config.vocab_size, config.hidden_size)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, config.hidden_size
        )
        if config.next_sentence:
            self.token_type_embeddings = nn.Embedding(
                config.type_vocab_size, config.hidden_size
            )
        else:
            self.token_type_embeddings = None
        self.LayerNorm = ops.FusedLayerNorm(config.hidden_size, eps=1e-12)
        self.dropout_prob = config.hidden_dropout_prob

    def forward(
        self, input_ids: torch.Tensor, token_type_ids: Optional[torch.Tensor]
    ) -> torch.Tensor:
        tok_w = (
            self.token_type_embeddings.weight
            if self.token_type_embeddings is not None
            else None
        )
        return ops.fused_embedding_ln_dropout(
            input_ids,
            token_type_ids if tok_w is not None else None,
            self.word_embeddings.weight,
            self.position_embeddings.weight,
            tok_w,
            self.LayerNorm.weight,
            self.LayerNorm.bias,
            self.dropout_prob,
            self.training,
        )


class _PackedQKV(nn.Module):
    """Packed [3H, H] QKV projection exposed as query/key/value in the
    state dict (reference names: src/modeling.py:387-389)."""

    def __init__(self, hidden_size: int):
        super().__init__()
        self.hidden_size = hidden_size
        self.qkv_weight = nn.Parameter(torch.empty(3 * hidden_size, hidden_size))
        self.qkv_bias = nn.Parameter(torch.zeros(3 * hidden_size))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.fused_linear(x, self.qkv_weight, self.qkv_bias)

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        h = self.hidden_size
        w = self.qkv_weight if keep_vars else self.qkv_weight.detach()
        b = self.qkv_bias if keep_vars else self.qkv_bias.detach()
        for i, name in enumerate(("query", "key", "value")):
            destination[f"{prefix}{name}.weight"] = w[i * h : (i + 1) * h]
            destination[f"{prefix}{name}.bias"] = b[i * h : (i + 1) * h]

    def _load_from_state_dict(
        self, state_dict, prefix, local_metadata, strict,
        missing_keys, unexpected_keys, error_msgs,
    ):
        h = self.hidden_size
        parts_w, parts_b = [], []
        for name in ("query", "key", "value"):
            wk, bk = f"{prefix}{name}.weight", f"{prefix}{name}.bias"
            if wk in state_dict:
                parts_w.append(state_dict.pop(wk))
            elif strict:
                missing_keys.append(wk)
            if bk in state_dict:
                parts_b.append(state_dict.pop(bk))
            elif strict:
                missing_keys.append(bk)
        with torch.no_grad():
            if len(parts_w) == 3:
                self.qkv_weight.copy_(torch.cat([p.reshape(h, h) for p in parts_w], 0))
            if len(parts_b) == 3:
                self.qkv_bias.copy_(torch.cat([p.reshape(h) for p in parts_b], 0))


class BertSelfOutput(nn.Module):
    """Attention out-projection + fused bias/dropout/residual/LN
    (reference: src/modeling.py:432-443)."""

    def __init__(self, config: BertConfig):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.LayerNorm = ops.FusedLayerNorm(config.hidden_size, eps=1e-12)
        self.dropout_prob = config.hidden_dropout_prob

    def forward(self, hidden: torch.Tensor, residual: torch.Tensor) -> torch.Tensor:
        y = ops.linear_nobias(hidden, self.dense.weight)  # bias in fusion
        return ops.fused_bias_dropout_residual_ln(
            y, self.dense.bias, residual,
            self.LayerNorm.weight, self.LayerNorm.bias,
            self.dropout_prob, self.training,
        )


class BertAttention(nn.Module):
    """Packed-QKV fused attention block (reference: src/modeling.py:376-455)."""

    def __init__(self, config: BertConfig):
        super().__init__()
        if config.hidden_size % config.num_attention_heads != 0:
            raise ValueError(
                f"hidden_size {config.hidden_size} not divisible by "
                f"num_attention_heads {config.num_attention_heads}"
            )
        self.num_heads = config.num_attention_heads
        self.self = _PackedQKV(config.hidden_size)
        self.output = BertSelfOutput(config)
        self.dropout_prob = config.attention_probs_dropout_prob

    def forward(self, x: torch.Tensor, seqlens: torch.Tensor) -> torch.Tensor:
        qkv = self.self(x)
        ctx = ops.fused_attention(
            qkv, seqlens, self.num_heads, self.dropout_prob, self.training
        )
        return self.output(ctx, x)


class BertIntermediate(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.dense_act = LinearActivation(
            config.hidden_size, config.intermediate_size, act=config.hidden_act
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dense_act(x)


class BertOutput(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.dense = nn.Linear(config.intermediate_size, config.hidden_size)
        self.LayerNorm = ops.FusedLayerNorm(config.hidden_size, eps=1e-12)
        self.dropout_prob = config.hidden_dropout_prob

    def forward(self, hidden: torch.Tensor, residual: torch.Tensor) -> torch.Tensor:
        y = ops.linear_nobias(hidden, self.dense.weight)
        return ops.fused_bias_dropout_residual_ln(
            y, self.dense.bias, residual,
            self.LayerNorm.weight, self.LayerNorm.bias,
            self.dropout_prob, self.training,
        )


class BertLayer(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.attention = BertAttention(config)
        self.intermediate = BertIntermediate(config)
        self.output = BertOutput(config)
        self._ffn_gelu = config.hidden_act in ("gelu", "bias_gelu")

    def forward(self, x: torch.Tensor, seqlens: torch.Tensor) -> torch.Tensor:
        attn = self.attention(x, seqlens)
        if self._ffn_gelu and ops.ffn_supported(attn):
            # GELU folded into the FFN GEMM epilogues (ops/ffn.py);
            # FFN2's bias stays fused in the bdrl kernel below
            y = ops.fused_ffn(
                attn,
                self.intermediate.dense_act.weight,
                self.intermediate.dense_act.bias,
                self.output.dense.weight,
            )
            return ops.fused_bias_dropout_residual_ln(
                y, self.output.dense.bias, attn,
                self.output.LayerNorm.weight, self.output.LayerNorm.bias,
                self.output.dropout_prob, self.training,
            )
        inter = self.intermediate(attn)
        return self.output(inter, attn)


class BertEncoder(nn.Module):
    """24-layer stack with optional sqrt(N)-chunked activation
    checkpointing (reference: src/modeling.py:495-536)."""

    def __init__(self, config: BertConfig):
        super().__init__()
        self.layer = nn.ModuleList(
            BertLayer(config) for _ in range(config.num_hidden_layers)
        )
        self.output_all_encoded_layers = config.output_all_encoded_layers
        self._checkpoint_activations = False

    def _checkpointed_forward(self, x, seqlens):
        num_layers = len(self.layer)
        chunk = int(math.sqrt(num_layers)) or 1

        def run(start, end):
            def custom(hidden, lens):
                for layer in self.layer[start:end]:
                    hidden = layer(hidden, lens)
                return hidden

            return custom

        for start in range(0, num_layers, chunk):
            end = min(start + chunk, num_layers)
            x = torch.utils.checkpoint.checkpoint(
                run(start, end), x, seqlens, use_reentrant=False
            )
        return x

    def forward(self, x: torch.Tensor, seqlens: torch.Tensor):
        all_layers = []
        if self._checkpoint_activations:
            x = self._checkpointed_forward(x, seqlens)
        else:
            for layer in self.layer:
                x = layer(x, seqlens)
                if self.output_all_encoded_layers:
                    all_layers.append(x)
        if not self.output_all_encoded_layers or self._checkpoint_activations:
            all_layers.append(x)
        return all_layers


class BertPooler(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.dense_act = LinearActivation(
            config.hidden_size, config.hidden_size, act="tanh"
        )

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        return self.dense_act(hidden[:, 0])


class BertPredictionHeadTransform(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.dense_act = LinearActivation(
            config.hidden_size, config.hidden_size, act=config.hidden_act
        )
        self.LayerNorm = ops.FusedLayerNorm(config.hidden_size, eps=1e-12)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.LayerNorm(self.dense_act(x))


class BertLMPredictionHead(nn.Module):
    """MLM head; decoder weight tied to the word embeddings
    (reference: src/modeling.py:563-579)."""

    def __init__(self, config: BertConfig, embedding_weights: torch.Tensor):
        super().__init__()
        self.transform = BertPredictionHeadTransform(config)
        self.decoder = nn.Linear(
            embedding_weights.size(1), embedding_weights.size(0), bias=False
        )
        self.decoder.weight = embedding_weights
        self.bias = nn.Parameter(torch.zeros(embedding_weights.size(0)))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.transform(x)
        return F.linear(x, self.decoder.weight, self.bias)


class BertPreTrainingHeads(nn.Module):
    def __init__(self, config: BertConfig, embedding_weights: torch.Tensor):
        super().__init__()
        self.predictions = BertLMPredictionHead(config, embedding_weights)
        if config.next_sentence:
            self.seq_relationship = nn.Linear(config.hidden_size, 2)
        else:
            self.seq_relationship = None

    def forward(self, sequence_output, pooled_output):
        prediction_scores = self.predictions(sequence_output)
        if self.seq_relationship is not None and pooled_output is not None:
            return prediction_scores, self.seq_relationship(pooled_output)
        return prediction_scores, None


class BertOnlyMLMHead(nn.Module):
    def __init__(self, config: BertConfig, embedding_weights: torch.Tensor):
        super().__init__()
        self.predictions = BertLMPredictionHead(config, embedding_weights)

    def forward(self, sequence_output):
        return self.predictions(sequence_output)


class BertOnlyNSPHead(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.seq_relationship = nn.Linear(config.hidden_size, 2)

    def forward(self, pooled_output):
        return self.seq_relationship(pooled_output)


class BertPreTrainedModel(nn.Module):
    """Base: weight init, activation-checkpoint toggle, local
    from_pretrained (reference: src/modeling.py:620-799)."""

    def __init__(self, config: BertConfig):
        super().__init__()
        if not isinstance(config, BertConfig):
            raise ValueError("config must be a BertConfig")
        self.config = config

    def init_bert_weights(self, module: nn.Module) -> None:
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(mean=0.0, std=self.config.initializer_range)
        elif isinstance(module, (LinearActivation,)):
            module.weight.data.normal_(mean=0.0, std=self.config.initializer_range)
        elif isinstance(module, _PackedQKV):
            module.qkv_weight.data.normal_(
                mean=0.0, std=self.config.initializer_range
            )
        elif isinstance(module, ops.FusedLayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()
        if isinstance(module, (nn.Linear, LinearActivation)) and module.bias is not None:
            module.bias.data.zero_()

    def apply_init(self) -> None:
        self.apply(self.init_bert_weights)

    def checkpoint_activations(self, flag: bool) -> None:
        for module in self.modules():
            if isinstance(module, BertEncoder):
                module._checkpoint_activations = flag

    @classmethod
    def from_pretrained(
        cls,
        path: str,
        config: Optional[BertConfig] = None,
        from_tf: bool = False,
        **kw,
    ):
        """Load from a local checkpoint: a ``ckpt_*.pt`` dict with a
        'model' key, a bare state-dict file, or — with ``from_tf=True``
        or when ``path`` points at a TF bundle prefix (``*.ckpt`` with
        an ``.index`` next to it) — a TensorFlow checkpoint parsed by
        the in-repo bundle reader (reference: src/modeling.py:58-116,
        no TensorFlow needed). The reference's URL/S3 download cache
        needs network access, absent here."""
        import os

        if config is None:
            cfg_file = os.path.join(path, "config.json")
            if os.path.isfile(cfg_file):
                config = BertConfig.from_json_file(cfg_file)
            else:
                raise ValueError("config required when path is not a model dir")
        model = cls(config, **kw)
        if from_tf or os.path.isfile(path + ".index"):
            from .tf_import import load_tf_weights  # noqa: PLC0415

            return load_tf_weights(model, path, strict=False)
        sd_file = path
        if os.path.isdir(path):
            sd_file = os.path.join(path, "pytorch_model.bin")
        state = torch.load(sd_file, map_location="cpu", weights_only=False)
        if isinstance(state, dict) and "model" in state:
            state = state["model"]
        state = {k.removeprefix("module."): v for k, v in state.items()}
        missing, unexpected = model.load_state_dict(state, strict=False)
        if missing:
            logger.info("from_pretrained: missing keys %s", missing)
        if unexpected:
            logger.info("from_pretrained: unexpected keys %s", unexpected)
        return model


class BertModel(BertPreTrainedModel):
    """Embeddings -> encoder -> pooler (reference: src/modeling.py:802-883)."""

    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.embeddings = BertEmbeddings(config)
        self.encoder = BertEncoder(config)
        self.pooler = BertPooler(config) if config.next_sentence else None
        self.output_all_encoded_layers = config.output_all_encoded_layers
        self.apply_init()

    def forward(
        self,
        input_ids: torch.Tensor,
        token_type_ids: Optional[torch.Tensor] = None,
        attention_mask: Optional[torch.Tensor] = None,
    ):
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        if token_type_ids is None and self.embeddings.token_type_embeddings is not None:
            token_type_ids = torch.zeros_like(input_ids)
        seqlens = attention_mask.sum(dim=-1, dtype=torch.int32)

        emb = self.embeddings(input_ids, token_type_ids)
        encoded_layers = self.encoder(emb, seqlens)
        sequence_output = encoded_layers[-1]
        pooled = self.pooler(sequence_output) if self.pooler is not None else None
        if not self.output_all_encoded_layers:
            encoded_layers = encoded_layers[-1]
        return encoded_layers, pooled


class BertForPreTraining(BertPreTrainedModel):
    """MLM (+NSP) pretraining model (reference: src/modeling.py:886-947).

    When ``masked_lm_labels`` is given, the MLM head runs only on the
    masked positions (exact same loss/gradients as the reference's full
    [B,S,V] head — unmasked rows contribute zero — at ~1/6 of the
    decoder-GEMM FLOPs for phase-1 shapes) and the gathered labels are
    returned alongside the scores.

    With ``compute_mlm_loss=True`` (the runner/bench path) the first
    return element is the scalar MLM loss instead of the [P, V] scores:
    the decoder GEMM + bias + cross-entropy run through
    ``ops.mlm_decoder_loss`` (library GEMM + fused-CE kernel by
    default; the in-repo fused MFMA kernel with BPA_FUSED_MLM=1), and
    ``BertPretrainingCriterion`` detects the 0-dim tensor and adds only
    the NSP term.
    """

    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.cls = BertPreTrainingHeads(
            config, self.bert.embeddings.word_embeddings.weight
        )
        for m in (self.cls,):
            m.apply(self.init_bert_weights)
        # re-tie after init (init overwrote the shared tensor in place; the
        # tie itself survives, but keep decoder pointing at embeddings)
        self.cls.predictions.decoder.weight = (
            self.bert.embeddings.word_embeddings.weight
        )

    def forward(
        self,
        input_ids,
        token_type_ids=None,
        attention_mask=None,
        masked_lm_labels=None,
        max_predictions_per_seq=None,
        compute_mlm_loss=False,
    ):
        encoded, pooled = self.bert(input_ids, token_type_ids, attention_mask)
        sequence_output = encoded[-1] if isinstance(encoded, list) else encoded
        if masked_lm_labels is None:
            return self.cls(sequence_output, pooled)
        flat_labels = masked_lm_labels.reshape(-1)
        if max_predictions_per_seq is not None:
            # Sync-free padded gather: a fixed [B*max_pred] row budget so no
            # torch.nonzero device->host count sync per micro-batch. Padding
            # slots point at row 0 with label -1, which the CE loss ignores
            # (zero gradient), so loss/grads match the variable-size gather
            # and the reference's full [B,S,V] head exactly.
            num = flat_labels.numel()
            cap = masked_lm_labels.shape[0] * max_predictions_per_seq
            mask = flat_labels != -1
            slots = torch.cumsum(mask.to(torch.long), 0) - 1
            slots = torch.where(
                mask, slots.clamp_(max=cap), torch.full_like(slots, cap)
            )
            positions = torch.zeros(
                cap + 1, dtype=torch.long, device=flat_labels.device
            )
            positions.scatter_(
                0, slots, torch.arange(num, device=flat_labels.device)
            )
            gathered_labels = torch.full(
                (cap + 1,), -1, dtype=flat_labels.dtype,
                device=flat_labels.device,
            )
            gathered_labels.scatter_(0, slots, flat_labels)
            positions = positions[:cap]
            gathered_labels = gathered_labels[:cap]
        else:
            positions = torch.nonzero(
                flat_labels != -1, as_tuple=False
            ).squeeze(-1)
            gathered_labels = flat_labels.index_select(0, positions)
        hidden = sequence_output.reshape(-1, sequence_output.shape[-1])
        masked_hidden = hidden.index_select(0, positions)
        if compute_mlm_loss:
            # fused decoder-GEMM + bias + CE path (ops/mlm.py): the
            # first return element is the scalar MLM loss instead of
            # the [P, V] scores; BertPretrainingCriterion detects the
            # 0-dim tensor and adds only the NSP term
            head = self.cls.predictions
            scores = ops.mlm_decoder_loss(
                head.transform(masked_hidden),
                head.decoder.weight,
                head.bias,
                gathered_labels,
            )
        else:
            scores = self.cls.predictions(masked_hidden)
        seq_rel = (
            self.cls.seq_relationship(pooled)
            if self.cls.seq_relationship is not None and pooled is not None
            else None
        )
        return scores, seq_rel, gathered_labels


class BertForMaskedLM(BertPreTrainedModel):
    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.cls = BertOnlyMLMHead(
            config, self.bert.embeddings.word_embeddings.weight
        )
        self.cls.apply(self.init_bert_weights)
        self.cls.predictions.decoder.weight = (
            self.bert.embeddings.word_embeddings.weight
        )

    def forward(
        self, input_ids, token_type_ids=None, attention_mask=None,
        masked_lm_labels=None,
    ):
        encoded, _ = self.bert(input_ids, token_type_ids, attention_mask)
        seq_out = encoded[-1] if isinstance(encoded, list) else encoded
        scores = self.cls(seq_out)
        if masked_lm_labels is not None:
            return ops.fused_cross_entropy(
                scores.view(-1, scores.shape[-1]), masked_lm_labels.view(-1), -1
            )
        return scores


class BertForNextSentencePrediction(BertPreTrainedModel):
    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.cls = BertOnlyNSPHead(config)
        self.cls.apply(self.init_bert_weights)

    def forward(
        self, input_ids, token_type_ids=None, attention_mask=None,
        next_sentence_label=None,
    ):
        _, pooled = self.bert(input_ids, token_type_ids, attention_mask)
        score = self.cls(pooled)
        if next_sentence_label is not None:
            return F.cross_entropy(
                score.view(-1, 2), next_sentence_label.view(-1), ignore_index=-1
            )
        return score


class BertForSequenceClassification(BertPreTrainedModel):
    def __init__(self, config: BertConfig, num_labels: int = 2):
        super().__init__(config)
        self.num_labels = num_labels
        self.bert = BertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, num_labels)
        self.classifier.apply(self.init_bert_weights)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None, labels=None):
        _, pooled = self.bert(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits.view(-1, self.num_labels), labels.view(-1))
        return logits


class BertForMultipleChoice(BertPreTrainedModel):
    def __init__(self, config: BertConfig, num_choices: int = 2):
        super().__init__(config)
        self.num_choices = num_choices
        self.bert = BertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, 1)
        self.classifier.apply(self.init_bert_weights)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None, labels=None):
        flat = lambda t: t.view(-1, t.size(-1)) if t is not None else None  # noqa: E731
        _, pooled = self.bert(flat(input_ids), flat(token_type_ids), flat(attention_mask))
        logits = self.classifier(self.dropout(pooled)).view(-1, self.num_choices)
        if labels is not None:
            return F.cross_entropy(logits, labels)
        return logits


class BertForTokenClassification(BertPreTrainedModel):
    """Per-token classifier (NER) with active-loss masking
    (reference: src/modeling.py:1200-1271)."""

    def __init__(self, config: BertConfig, num_labels: int = 2):
        super().__init__(config)
        self.num_labels = num_labels
        self.bert = BertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, num_labels)
        self.classifier.apply(self.init_bert_weights)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None, labels=None):
        encoded, _ = self.bert(input_ids, token_type_ids, attention_mask)
        seq_out = encoded[-1] if isinstance(encoded, list) else encoded
        logits = self.classifier(self.dropout(seq_out))
        if labels is not None:
            if attention_mask is not None:
                active = attention_mask.view(-1) == 1
                return F.cross_entropy(
                    logits.view(-1, self.num_labels)[active],
                    labels.view(-1)[active],
                )
            return F.cross_entropy(logits.view(-1, self.num_labels), labels.view(-1))
        return logits


class BertForQuestionAnswering(BertPreTrainedModel):
    """SQuAD span head (reference: src/modeling.py:1274-1327)."""

    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.qa_outputs = nn.Linear(config.hidden_size, 2)
        self.qa_outputs.apply(self.init_bert_weights)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        encoded, _ = self.bert(input_ids, token_type_ids, attention_mask)
        seq_out = encoded[-1] if isinstance(encoded, list) else encoded
        logits = self.qa_outputs(seq_out)
        start_logits, end_logits = logits.split(1, dim=-1)
        return start_logits.squeeze(-1), end_logits.squeeze(-1)
