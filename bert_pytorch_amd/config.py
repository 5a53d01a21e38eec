"""Model + training configuration.

API-compatible with the reference's two-level config system
(reference: src/modeling.py:188-280 ``BertConfig`` and
run_pretraining.py:75-177 CLI>JSON>defaults merge): the same JSON model
config files and the same training-config JSON keys are accepted.
"""

from __future__ import annotations

import argparse
import copy
import json
import sys
from typing import Any, Dict, Optional


class BertConfig:
    """BERT model hyperparameters.

    Accepts the same JSON files as the reference (e.g.
    config/bert_large_uncased_config.json). Extra keys that the runners
    consume (vocab_file, lowercase, tokenizer, model_name) are stored as
    attributes too.
    """

    def __init__(
        self,
        vocab_size_or_config_json_file: Any = 30522,
        hidden_size: int = 768,
        num_hidden_layers: int = 12,
        num_attention_heads: int = 12,
        intermediate_size: int = 3072,
        hidden_act: str = "gelu",
        hidden_dropout_prob: float = 0.1,
        attention_probs_dropout_prob: float = 0.1,
        max_position_embeddings: int = 512,
        type_vocab_size: int = 2,
        initializer_range: float = 0.02,
        next_sentence: bool = True,
        output_all_encoded_layers: bool = False,
        **extra: Any,
    ) -> None:
        if isinstance(vocab_size_or_config_json_file, str):
            with open(vocab_size_or_config_json_file, "r", encoding="utf-8") as f:
                cfg = json.load(f)
            for key, value in cfg.items():
                setattr(self, key, value)
            # Fill defaults for anything the file omitted.
            defaults = dict(
                hidden_size=hidden_size,
                num_hidden_layers=num_hidden_layers,
                num_attention_heads=num_attention_heads,
                intermediate_size=intermediate_size,
                hidden_act=hidden_act,
                hidden_dropout_prob=hidden_dropout_prob,
                attention_probs_dropout_prob=attention_probs_dropout_prob,
                max_position_embeddings=max_position_embeddings,
                type_vocab_size=type_vocab_size,
                initializer_range=initializer_range,
                next_sentence=next_sentence,
                output_all_encoded_layers=output_all_encoded_layers,
            )
            for key, value in defaults.items():
                if not hasattr(self, key):
                    setattr(self, key, value)
        elif isinstance(vocab_size_or_config_json_file, int):
            self.vocab_size = vocab_size_or_config_json_file
            self.hidden_size = hidden_size
            self.num_hidden_layers = num_hidden_layers
            self.num_attention_heads = num_attention_heads
            self.intermediate_size = intermediate_size
            self.hidden_act = hidden_act
            self.hidden_dropout_prob = hidden_dropout_prob
            self.attention_probs_dropout_prob = attention_probs_dropout_prob
            self.max_position_embeddings = max_position_embeddings
            self.type_vocab_size = type_vocab_size
            self.initializer_range = initializer_range
            self.next_sentence = next_sentence
            self.output_all_encoded_layers = output_all_encoded_layers
            for key, value in extra.items():
                setattr(self, key, value)
        else:
            raise ValueError(
                "first argument must be a vocab size (int) or a config "
                "JSON path (str)"
            )

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "BertConfig":
        config = cls(vocab_size_or_config_json_file=d.get("vocab_size", 30522))
        for key, value in d.items():
            setattr(config, key, value)
        return config

    @classmethod
    def from_json_file(cls, json_file: str) -> "BertConfig":
        return cls(vocab_size_or_config_json_file=json_file)

    def to_dict(self) -> Dict[str, Any]:
        return copy.deepcopy(self.__dict__)

    def to_json_string(self) -> str:
        return json.dumps(self.to_dict(), indent=2, sort_keys=True) + "\n"

    def __repr__(self) -> str:
        return "BertConfig " + self.to_json_string()


def merge_config_and_args(
    parser: argparse.ArgumentParser,
    args: Optional[list] = None,
    config_key: str = "config_file",
) -> argparse.Namespace:
    """Merge precedence: CLI flag > training-config JSON > argparse default.

    Mirrors the reference's behavior (run_pretraining.py:159-172): a
    second parse with suppressed defaults identifies which flags were
    explicitly given on the command line; JSON values override defaults,
    explicit CLI values override JSON.
    """
    argv = sys.argv[1:] if args is None else args
    namespace = parser.parse_args(argv)

    config_path = getattr(namespace, config_key, None)
    if not config_path:
        return namespace

    with open(config_path, "r", encoding="utf-8") as f:
        json_cfg = json.load(f)

    # Which destinations were explicitly provided on the CLI?
    aux = argparse.ArgumentParser(add_help=False)
    for action in parser._actions:
        if action.dest in ("help", config_key) or not action.option_strings:
            continue
        if isinstance(action, (argparse._StoreTrueAction, argparse._StoreFalseAction)):
            aux.add_argument(
                *action.option_strings,
                dest=action.dest,
                action="store_true"
                if isinstance(action, argparse._StoreTrueAction)
                else "store_false",
                default=argparse.SUPPRESS,
            )
        else:
            aux.add_argument(
                *action.option_strings,
                dest=action.dest,
                nargs=action.nargs,
                type=action.type,
                choices=action.choices,
                default=argparse.SUPPRESS,
            )
    explicit, _ = aux.parse_known_args(argv)
    explicit_dests = set(vars(explicit).keys())

    valid = {a.dest for a in parser._actions}
    for key, value in json_cfg.items():
        if key in explicit_dests:
            continue  # CLI wins
        if key not in valid:
            # Tolerate unknown keys (forward compat) but keep them visible.
            setattr(namespace, key, value)
            continue
        setattr(namespace, key, value)
    return namespace
