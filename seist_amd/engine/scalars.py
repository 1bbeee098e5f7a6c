"""Scalar stream writer: TensorBoard when available, JSONL fallback.

The reference requires the tensorboard package (train.py:10); this image
may not have it, so the same add_scalar/add_scalars API writes JSON lines
that tensorboard-equipped machines can ingest later.
"""

import json
import os
import time


class ScalarWriter:
    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        self._tb = None
        try:
            from torch.utils.tensorboard import SummaryWriter
            self._tb = SummaryWriter(logdir)
        except Exception:
            self._fh = open(os.path.join(logdir, "scalars.jsonl"), "a")

    def add_scalar(self, tag: str, value, step: int):
        if self._tb is not None:
            self._tb.add_scalar(tag, value, step)
        else:
            self._fh.write(json.dumps(
                {"t": time.time(), "tag": tag, "value": float(value),
                 "step": int(step)}) + "\n")

    def add_scalars(self, tag: str, values: dict, step: int):
        for k, v in values.items():
            self.add_scalar(f"{tag}/{k}", v, step)

    def close(self):
        if self._tb is not None:
            self._tb.close()
        else:
            self._fh.close()
