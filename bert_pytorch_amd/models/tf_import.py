"""Import Google's published TF BERT checkpoints — without TensorFlow.

The reference's ``load_tf_weights_in_bert`` (src/modeling.py:58-116)
walks the module tree with ``tf.train.load_variable``; here the
checkpoint is parsed by the in-repo bundle reader
(``data/tf_bundle.py``) and mapped onto this package's
reference-compatible state-dict names. Behavior matches the reference:
``kernel`` weights are transposed (TF stores [in, out]), ``gamma`` /
``beta`` map to LayerNorm weight/bias, ``output_bias`` /
``output_weights`` map to the head bias/weight, and optimizer slots
(``adam_m`` / ``adam_v`` / ``global_step``) are skipped. One
deliberate divergence: TF's ``.../dense`` inside pooler / transform /
intermediate maps to this model's fused ``dense_act`` modules (the
reference's own attribute walk raises AttributeError on those — its
modules are ``dense_act`` too).
"""

from __future__ import annotations

import re
from typing import Dict, Optional, Tuple

import torch

from ..data.tf_bundle import TFBundleReader

_SKIP = {"adam_m", "adam_v", "global_step", "good_steps",
         "current_loss_scale", "beta1_power", "beta2_power"}

# parents whose TF "dense" child is a fused LinearActivation here
_DENSE_ACT_PARENTS = {"pooler", "transform", "intermediate"}


def tf_name_to_state_key(name: str) -> Optional[Tuple[str, bool]]:
    """Map a TF variable name to (state_dict key, transpose?).

    Returns None for variables that have no model equivalent
    (optimizer slots etc.).
    """
    parts = name.split("/")
    if any(p in _SKIP for p in parts):
        return None
    out = []
    transpose = False
    for i, p in enumerate(parts):
        if p == "kernel":
            out.append("weight")
            transpose = True
        elif p == "gamma" or p == "output_weights":
            out.append("weight")
        elif p == "beta" or p == "output_bias":
            out.append("bias")
        elif re.fullmatch(r"layer_\d+", p):
            out.append("layer." + p.split("_")[1])
        elif p == "dense" and i > 0 and parts[i - 1] in _DENSE_ACT_PARENTS:
            out.append("dense_act")
        else:
            out.append(p)
    key = ".".join(out)
    if parts[-1].endswith("_embeddings"):
        key += ".weight"
    return key, transpose


def load_tf_weights(model: torch.nn.Module, ckpt_prefix: str,
                    strict: bool = True) -> torch.nn.Module:
    """Load a TF checkpoint (``.../bert_model.ckpt`` prefix) into
    ``model`` (BertModel / BertForPreTraining / any task head whose
    ``bert.*`` names match)."""
    reader = TFBundleReader(ckpt_prefix)
    state: Dict[str, torch.Tensor] = {}
    model_keys = set(model.state_dict().keys())
    unmapped = []
    for name, _shape in reader.list_variables():
        mapped = tf_name_to_state_key(name)
        if mapped is None:
            continue
        key, transpose = mapped
        if key not in model_keys:
            unmapped.append(name)
            continue
        arr = reader.load_variable(name)
        t = torch.from_numpy(arr.copy())
        if transpose:
            t = t.t().contiguous()
        state[key] = t
    if strict and unmapped:
        raise KeyError(
            f"TF variables with no model equivalent: {unmapped[:8]}"
            + ("..." if len(unmapped) > 8 else "")
        )
    missing, unexpected = model.load_state_dict(state, strict=False)
    # the tied decoder weight comes from word_embeddings; everything
    # else the checkpoint does not cover must be explicitly acceptable
    tied_ok = {"cls.predictions.decoder.weight"}
    hard_missing = [m for m in missing if m not in tied_ok]
    if strict and (hard_missing or unexpected):
        raise KeyError(
            f"TF import mismatch: missing={hard_missing[:8]} "
            f"unexpected={list(unexpected)[:8]}"
        )
    return model
