"""Structured metric logging with four sinks.

Replaces ``loggerplus`` (reference: run_pretraining.py:191-204):
stream (rank-0), append-file, CSV, and TensorBoard sinks with the same
metric names, so runs are comparable the same way.
TensorBoard output degrades gracefully to no-op when the writer is
unavailable in this image.
"""

from __future__ import annotations

import csv
import logging
import os
import sys
from typing import Any, Dict, Optional

logger = logging.getLogger("bert_pytorch_amd")


class MetricLogger:
    def __init__(
        self,
        log_prefix: Optional[str] = None,
        tensorboard_dir: Optional[str] = None,
        verbose: bool = True,
    ):
        self.verbose = verbose
        self._stream = logging.StreamHandler(sys.stdout)
        self._stream.setFormatter(
            logging.Formatter("%(asctime)s %(levelname)s %(message)s")
        )
        logger.setLevel(logging.INFO)
        if verbose and not logger.handlers:
            logger.addHandler(self._stream)
        self._file = None
        self._csv_path = None
        self._csv_fields: list[str] = []
        if log_prefix:
            os.makedirs(os.path.dirname(log_prefix) or ".", exist_ok=True)
            self._file = open(f"{log_prefix}.txt", "a", encoding="utf-8")
            self._csv_path = f"{log_prefix}_metrics.csv"
            if os.path.exists(self._csv_path):
                # Resume: adopt the existing header so appended rows line up.
                with open(self._csv_path, newline="", encoding="utf-8") as f:
                    header = f.readline().strip()
                if header:
                    self._csv_fields = header.split(",")
        self._tb = None
        if tensorboard_dir:
            try:
                from torch.utils.tensorboard import SummaryWriter  # noqa: PLC0415

                self._tb = SummaryWriter(tensorboard_dir)
            except Exception:  # noqa: BLE001 - tensorboard optional
                logger.info("tensorboard unavailable; skipping TB sink")

    def info(self, msg: str, *args: Any) -> None:
        logger.info(msg, *args)
        if self._file:
            self._file.write((msg % args if args else msg) + "\n")
            self._file.flush()

    def log(self, tag: str, step: int, **metrics: Any) -> None:
        parts = " ".join(f"{k}={v}" for k, v in metrics.items())
        self.info("[%s] step=%d %s", tag, step, parts)
        if self._csv_path:
            row: Dict[str, Any] = {"tag": tag, "step": step, **metrics}
            new_fields = [k for k in row if k not in self._csv_fields]
            exists = os.path.exists(self._csv_path)
            if new_fields and exists and self._csv_fields:
                # Schema widened mid-run (e.g. eval metrics joining a train
                # log): rewrite the file under the union header so every row
                # has the same columns and csv.DictReader parses it whole.
                with open(self._csv_path, newline="", encoding="utf-8") as f:
                    old_rows = list(csv.DictReader(f))
                self._csv_fields += new_fields
                with open(self._csv_path, "w", newline="", encoding="utf-8") as f:
                    writer = csv.DictWriter(f, fieldnames=self._csv_fields)
                    writer.writeheader()
                    writer.writerows(old_rows)
                    writer.writerow(row)
            else:
                if new_fields:
                    self._csv_fields += new_fields
                write_header = not exists
                with open(self._csv_path, "a", newline="", encoding="utf-8") as f:
                    writer = csv.DictWriter(f, fieldnames=self._csv_fields)
                    if write_header:
                        writer.writeheader()
                    writer.writerow(row)
        if self._tb:
            for key, value in metrics.items():
                if isinstance(value, (int, float)):
                    self._tb.add_scalar(f"{tag}/{key}", value, step)

    def close(self) -> None:
        if self._file:
            self._file.close()
        if self._tb:
            self._tb.close()
