"""SQuAD data pipeline: examples, sliding-window features, span decode,
and the official-metric evaluator.

Re-designed from the reference's run_squad.py data machinery
(read_squad_examples :131-206, convert_examples_to_features :209-346,
get_answers :427-506, get_final_text :570-664) with the same behavior:
doc-stride sliding windows, max-context token assignment, n-best span
decoding with heuristic text alignment back to the original passage.
The evaluator reimplements the official evaluate-v1.1 EM/F1 (the
reference subprocess-runs the downloaded script, :1197-1204 — no
network here, so the metric lives in-repo).
"""

from __future__ import annotations

import collections
import json
import math
import re
import string
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from .tokenization import BasicTokenizer


def _is_whitespace(c: str) -> bool:
    return c in " \t\r\n" or ord(c) == 0x202F


@dataclass
class SquadExample:
    qas_id: str
    question_text: str
    doc_tokens: List[str]
    orig_answer_text: Optional[str] = None
    start_position: Optional[int] = None
    end_position: Optional[int] = None
    is_impossible: bool = False


@dataclass
class InputFeatures:
    unique_id: int
    example_index: int
    doc_span_index: int
    tokens: List[str]
    token_to_orig_map: Dict[int, int]
    token_is_max_context: Dict[int, bool]
    input_ids: List[int]
    input_mask: List[int]
    segment_ids: List[int]
    start_position: Optional[int] = None
    end_position: Optional[int] = None
    is_impossible: bool = False


RawResult = collections.namedtuple(
    "RawResult", ["unique_id", "start_logits", "end_logits"]
)


def read_squad_examples(
    input_file: str, is_training: bool, version_2_with_negative: bool = False
) -> List[SquadExample]:
    with open(input_file, "r", encoding="utf-8") as f:
        input_data = json.load(f)["data"]

    examples: List[SquadExample] = []
    for entry in input_data:
        for paragraph in entry["paragraphs"]:
            text = paragraph["context"]
            doc_tokens: List[str] = []
            char_to_word: List[int] = []
            prev_ws = True
            for c in text:
                if _is_whitespace(c):
                    prev_ws = True
                else:
                    if prev_ws:
                        doc_tokens.append(c)
                    else:
                        doc_tokens[-1] += c
                    prev_ws = False
                char_to_word.append(len(doc_tokens) - 1)

            for qa in paragraph["qas"]:
                start_pos = end_pos = None
                orig_answer = None
                is_impossible = False
                if is_training:
                    if version_2_with_negative:
                        is_impossible = qa.get("is_impossible", False)
                    if not is_impossible:
                        if len(qa["answers"]) < 1:
                            continue
                        answer = qa["answers"][0]
                        orig_answer = answer["text"]
                        start_char = answer["answer_start"]
                        start_pos = char_to_word[start_char]
                        end_pos = char_to_word[
                            start_char + len(orig_answer) - 1
                        ]
                        # skip examples whose answer can't be recovered
                        actual = " ".join(doc_tokens[start_pos : end_pos + 1])
                        cleaned = " ".join(orig_answer.strip().split())
                        if actual.find(cleaned) == -1:
                            continue
                    else:
                        start_pos = end_pos = -1
                        orig_answer = ""
                examples.append(
                    SquadExample(
                        qas_id=qa["id"],
                        question_text=qa["question"],
                        doc_tokens=doc_tokens,
                        orig_answer_text=orig_answer,
                        start_position=start_pos,
                        end_position=end_pos,
                        is_impossible=is_impossible,
                    )
                )
    return examples


def _improve_answer_span(doc_tokens, start, end, tokenizer, orig_answer_text):
    """Match the tokenized answer tighter (reference :349-383)."""
    tok_answer = " ".join(
        tokenizer.encode(orig_answer_text, add_special_tokens=False).tokens
    )
    for new_start in range(start, end + 1):
        for new_end in range(end, new_start - 1, -1):
            span = " ".join(doc_tokens[new_start : new_end + 1])
            if span == tok_answer:
                return new_start, new_end
    return start, end


def _check_is_max_context(doc_spans, cur_span_index, position):
    best_score, best_idx = None, None
    for idx, (span_start, span_len) in enumerate(doc_spans):
        end = span_start + span_len - 1
        if position < span_start or position > end:
            continue
        left = position - span_start
        right = end - position
        score = min(left, right) + 0.01 * span_len
        if best_score is None or score > best_score:
            best_score, best_idx = score, idx
    return cur_span_index == best_idx


def convert_examples_to_features(
    examples: List[SquadExample],
    tokenizer,
    max_seq_length: int = 384,
    doc_stride: int = 128,
    max_query_length: int = 64,
    is_training: bool = True,
) -> List[InputFeatures]:
    features: List[InputFeatures] = []
    unique_id = 1000000000
    cls_id = tokenizer.token_to_id("[CLS]")
    sep_id = tokenizer.token_to_id("[SEP]")
    pad_id = tokenizer.token_to_id("[PAD]") or 0

    for example_index, example in enumerate(examples):
        query_tokens = tokenizer.encode(
            example.question_text, add_special_tokens=False
        ).tokens[:max_query_length]

        tok_to_orig: List[int] = []
        orig_to_tok: List[int] = []
        all_doc_tokens: List[str] = []
        for i, token in enumerate(example.doc_tokens):
            orig_to_tok.append(len(all_doc_tokens))
            for sub in tokenizer.encode(token, add_special_tokens=False).tokens:
                tok_to_orig.append(i)
                all_doc_tokens.append(sub)

        tok_start = tok_end = None
        if is_training:
            if example.is_impossible:
                tok_start = tok_end = -1
            else:
                tok_start = orig_to_tok[example.start_position]
                tok_end = (
                    orig_to_tok[example.end_position + 1] - 1
                    if example.end_position < len(example.doc_tokens) - 1
                    else len(all_doc_tokens) - 1
                )
                tok_start, tok_end = _improve_answer_span(
                    all_doc_tokens, tok_start, tok_end, tokenizer,
                    example.orig_answer_text,
                )

        max_doc = max_seq_length - len(query_tokens) - 3
        doc_spans: List[Tuple[int, int]] = []
        start_offset = 0
        while start_offset < len(all_doc_tokens):
            length = min(len(all_doc_tokens) - start_offset, max_doc)
            doc_spans.append((start_offset, length))
            if start_offset + length == len(all_doc_tokens):
                break
            start_offset += min(length, doc_stride)

        for span_index, (span_start, span_len) in enumerate(doc_spans):
            tokens = ["[CLS]"] + query_tokens + ["[SEP]"]
            segment_ids = [0] * len(tokens)
            token_to_orig_map: Dict[int, int] = {}
            token_is_max_context: Dict[int, bool] = {}
            for i in range(span_len):
                split_idx = span_start + i
                token_to_orig_map[len(tokens)] = tok_to_orig[split_idx]
                token_is_max_context[len(tokens)] = _check_is_max_context(
                    doc_spans, span_index, split_idx
                )
                tokens.append(all_doc_tokens[split_idx])
                segment_ids.append(1)
            tokens.append("[SEP]")
            segment_ids.append(1)

            input_ids = tokenizer.convert_tokens_to_ids(tokens)
            input_mask = [1] * len(input_ids)
            while len(input_ids) < max_seq_length:
                input_ids.append(pad_id)
                input_mask.append(0)
                segment_ids.append(0)

            start_position = end_position = None
            if is_training:
                if example.is_impossible or not (
                    tok_start >= span_start
                    and tok_end <= span_start + span_len - 1
                ):
                    start_position = end_position = 0
                else:
                    offset = len(query_tokens) + 2
                    start_position = tok_start - span_start + offset
                    end_position = tok_end - span_start + offset

            features.append(
                InputFeatures(
                    unique_id=unique_id,
                    example_index=example_index,
                    doc_span_index=span_index,
                    tokens=tokens,
                    token_to_orig_map=token_to_orig_map,
                    token_is_max_context=token_is_max_context,
                    input_ids=input_ids,
                    input_mask=input_mask,
                    segment_ids=segment_ids,
                    start_position=start_position,
                    end_position=end_position,
                    is_impossible=example.is_impossible,
                )
            )
            unique_id += 1
    return features


def get_final_text(pred_text: str, orig_text: str, do_lower_case: bool) -> str:
    """Project the tokenized prediction back onto the original text
    (reference :570-664)."""
    def strip_spaces(text):
        ns_chars = []
        ns_to_s = collections.OrderedDict()
        for i, c in enumerate(text):
            if c == " ":
                continue
            ns_to_s[len(ns_chars)] = i
            ns_chars.append(c)
        return "".join(ns_chars), ns_to_s

    tokenizer = BasicTokenizer(do_lower_case=do_lower_case)
    tok_text = " ".join(tokenizer.tokenize(orig_text))
    start_position = tok_text.find(pred_text)
    if start_position == -1:
        return orig_text
    end_position = start_position + len(pred_text) - 1

    orig_ns_text, orig_ns_map = strip_spaces(orig_text)
    tok_ns_text, tok_ns_map = strip_spaces(tok_text)
    if len(orig_ns_text) != len(tok_ns_text):
        return orig_text

    tok_s_to_ns = {v: k for k, v in tok_ns_map.items()}
    orig_start = orig_ns_map.get(tok_s_to_ns.get(start_position))
    orig_end = orig_ns_map.get(tok_s_to_ns.get(end_position))
    if orig_start is None or orig_end is None:
        return orig_text
    return orig_text[orig_start : orig_end + 1]


def _best_indexes(logits, n_best_size):
    return [
        i for i, _ in sorted(
            enumerate(logits), key=lambda x: x[1], reverse=True
        )[:n_best_size]
    ]


def get_answers(
    examples: List[SquadExample],
    features: List[InputFeatures],
    results: List[RawResult],
    n_best_size: int = 20,
    max_answer_length: int = 30,
    do_lower_case: bool = True,
    version_2_with_negative: bool = False,
    null_score_diff_threshold: float = 0.0,
) -> Tuple[Dict[str, str], Dict[str, list]]:
    example_to_features = collections.defaultdict(list)
    for f in features:
        example_to_features[f.example_index].append(f)
    result_by_id = {r.unique_id: r for r in results}

    predictions: Dict[str, str] = {}
    nbest_out: Dict[str, list] = {}
    Prelim = collections.namedtuple(
        "Prelim", ["feature", "start", "end", "start_logit", "end_logit"]
    )
    for example_index, example in enumerate(examples):
        prelim: List[Prelim] = []
        null_score = 1e30
        null_entry = None
        for f in example_to_features[example_index]:
            result = result_by_id.get(f.unique_id)
            if result is None:
                continue
            if version_2_with_negative:
                feature_null = result.start_logits[0] + result.end_logits[0]
                if feature_null < null_score:
                    null_score = feature_null
                    null_entry = Prelim(f, 0, 0, result.start_logits[0],
                                        result.end_logits[0])
            for s in _best_indexes(result.start_logits, n_best_size):
                for e in _best_indexes(result.end_logits, n_best_size):
                    if (
                        s >= len(f.tokens) or e >= len(f.tokens)
                        or s not in f.token_to_orig_map
                        or e not in f.token_to_orig_map
                        or not f.token_is_max_context.get(s, False)
                        or e < s or e - s + 1 > max_answer_length
                    ):
                        continue
                    prelim.append(
                        Prelim(f, s, e, result.start_logits[s],
                               result.end_logits[e])
                    )
        if version_2_with_negative and null_entry is not None:
            prelim.append(null_entry)
        prelim.sort(key=lambda p: p.start_logit + p.end_logit, reverse=True)

        seen = set()
        nbest = []
        for p in prelim[:n_best_size]:
            if p.start > 0:
                tok_tokens = p.feature.tokens[p.start : p.end + 1]
                orig_start = p.feature.token_to_orig_map[p.start]
                orig_end = p.feature.token_to_orig_map[p.end]
                orig_tokens = example.doc_tokens[orig_start : orig_end + 1]
                tok_text = " ".join(tok_tokens).replace(" ##", "").replace("##", "")
                tok_text = " ".join(tok_text.strip().split())
                orig_text = " ".join(orig_tokens)
                final = get_final_text(tok_text, orig_text, do_lower_case)
            else:
                final = ""
            if final in seen:
                continue
            seen.add(final)
            nbest.append(
                {"text": final, "start_logit": p.start_logit,
                 "end_logit": p.end_logit}
            )
        if not nbest:
            nbest.append({"text": "empty", "start_logit": 0.0, "end_logit": 0.0})

        scores = [e["start_logit"] + e["end_logit"] for e in nbest]
        max_score = max(scores)
        probs = [math.exp(s - max_score) for s in scores]
        total = sum(probs)
        for entry, prob in zip(nbest, probs):
            entry["probability"] = prob / total

        best = nbest[0]["text"]
        if version_2_with_negative:
            best_non_null = next((e for e in nbest if e["text"]), None)
            if best_non_null is None:
                best = ""
            else:
                diff = null_score - (
                    best_non_null["start_logit"] + best_non_null["end_logit"]
                )
                best = "" if diff > null_score_diff_threshold else best_non_null["text"]
        predictions[example.qas_id] = best
        nbest_out[example.qas_id] = nbest
    return predictions, nbest_out


# -- official v1.1 metric (in-repo reimplementation) ------------------------
def _normalize_answer(s: str) -> str:
    s = s.lower()
    s = "".join(c for c in s if c not in set(string.punctuation))
    s = re.sub(r"\b(a|an|the)\b", " ", s)
    return " ".join(s.split())


def _f1_score(prediction: str, ground_truth: str) -> float:
    pred_tokens = _normalize_answer(prediction).split()
    gt_tokens = _normalize_answer(ground_truth).split()
    common = collections.Counter(pred_tokens) & collections.Counter(gt_tokens)
    num_same = sum(common.values())
    if num_same == 0:
        return 0.0
    precision = num_same / len(pred_tokens)
    recall = num_same / len(gt_tokens)
    return 2 * precision * recall / (precision + recall)


def _em_score(prediction: str, ground_truth: str) -> float:
    return float(_normalize_answer(prediction) == _normalize_answer(ground_truth))


def evaluate_predictions(dataset_file: str, predictions: Dict[str, str]) -> Dict[str, float]:
    """SQuAD v1.1 exact_match / F1 over a predictions dict."""
    with open(dataset_file, "r", encoding="utf-8") as f:
        dataset = json.load(f)["data"]
    f1 = em = total = 0.0
    for article in dataset:
        for paragraph in article["paragraphs"]:
            for qa in paragraph["qas"]:
                total += 1
                if qa["id"] not in predictions:
                    continue
                ground_truths = [a["text"] for a in qa["answers"]] or [""]
                prediction = predictions[qa["id"]]
                em += max(_em_score(prediction, gt) for gt in ground_truths)
                f1 += max(_f1_score(prediction, gt) for gt in ground_truths)
    total = max(total, 1.0)
    return {"exact_match": 100.0 * em / total, "f1": 100.0 * f1 / total}
