"""Multi-sink experiment logger.

Parity with /root/reference/stoix/utils/logger.py: event taxonomy
ACT/TRAIN/EVAL/ABSOLUTE/MISC (:28-33), thread-safe facade over multiple sinks
(console / JSON / CSV / TensorBoard-style scalars file), describe()
mean/std/min/max aggregation (:607-613), TRAIN metrics aggregated mean-only
(:152-154). wandb/neptune are not installed in this offline image; their
sink names are accepted and warn-skipped (config parity).
"""
from __future__ import annotations

import json
import os
import threading
import time
from enum import Enum
from typing import Any, Dict

import numpy as np
import torch


class LogEvent(Enum):
    ACT = "actor"
    TRAIN = "trainer"
    EVAL = "evaluator"
    ABSOLUTE = "absolute"
    MISC = "misc"


def describe(x: np.ndarray) -> Dict[str, float]:
    """mean/std/min/max summary of an array metric (reference logger.py:607-613)."""
    x = np.asarray(x, dtype=np.float64)
    if x.size == 0:
        return {}
    return {
        "mean": float(np.mean(x)),
        "std": float(np.std(x)),
        "min": float(np.min(x)),
        "max": float(np.max(x)),
    }


def _to_scalar_dict(metrics: Dict[str, Any], mean_only: bool) -> Dict[str, float]:
    out: Dict[str, float] = {}
    for k, v in metrics.items():
        if isinstance(v, torch.Tensor):
            v = v.detach().float().cpu().numpy()
        if isinstance(v, np.ndarray) and v.size > 1:
            if mean_only:
                out[k] = float(np.mean(v))
            else:
                for stat, val in describe(v).items():
                    out[f"{k}_{stat}" if stat != "mean" else k] = val
        elif isinstance(v, (int, float, np.floating, np.integer)) or (
            isinstance(v, np.ndarray) and v.size == 1
        ):
            out[k] = float(np.asarray(v).reshape(()))
    return out


class ConsoleSink:
    COLORS = {
        LogEvent.ACT: "\033[95m",
        LogEvent.TRAIN: "\033[94m",
        LogEvent.EVAL: "\033[92m",
        LogEvent.ABSOLUTE: "\033[93m",
        LogEvent.MISC: "\033[96m",
    }

    def log(self, metrics: Dict[str, float], t: int, t_eval: int, event: LogEvent) -> None:
        color = self.COLORS.get(event, "")
        parts = " | ".join(
            f"{k.replace('_', ' ')}: {v:.3f}" if isinstance(v, float) else f"{k}: {v}"
            for k, v in sorted(metrics.items())
        )
        print(f"{color}[{event.value.upper()}] t={t} eval={t_eval} | {parts}\033[0m", flush=True)

    def close(self) -> None:
        pass


class JsonSink:
    """marl-eval-style JSON lines (reference logger.py:325-386 writes nested
    json for downstream statistical tooling; we write one JSON object per log
    call, stream-appendable)."""

    def __init__(self, directory: str, run_name: str):
        os.makedirs(directory, exist_ok=True)
        self.path = os.path.join(directory, f"{run_name}.jsonl")
        self._f = open(self.path, "a")

    def log(self, metrics: Dict[str, float], t: int, t_eval: int, event: LogEvent) -> None:
        self._f.write(json.dumps({"t": t, "t_eval": t_eval, "event": event.value, **metrics}) + "\n")
        self._f.flush()

    def close(self) -> None:
        self._f.close()


class CsvSink:
    def __init__(self, directory: str, run_name: str):
        os.makedirs(directory, exist_ok=True)
        self.path = os.path.join(directory, f"{run_name}.csv")
        self._f = open(self.path, "a")
        self._header_written = os.path.getsize(self.path) > 0
        self._cols = None

    def log(self, metrics: Dict[str, float], t: int, t_eval: int, event: LogEvent) -> None:
        row = {"t": t, "t_eval": t_eval, "event": event.value, **metrics}
        if self._cols is None:
            self._cols = list(row.keys())
            if not self._header_written:
                self._f.write(",".join(self._cols) + "\n")
        self._f.write(",".join(str(row.get(c, "")) for c in self._cols) + "\n")
        self._f.flush()

    def close(self) -> None:
        self._f.close()


class TensorboardSink:
    """Minimal TensorBoard-compatible scalar event writer (no tensorboard
    package offline): writes tfevents files with the scalar summary proto
    encoded by hand (varint + length-delimited fields)."""

    def __init__(self, directory: str, run_name: str):
        import struct
        import zlib

        self._struct = struct
        self._crc = self._masked_crc
        os.makedirs(os.path.join(directory, run_name), exist_ok=True)
        self.path = os.path.join(directory, run_name, f"events.out.tfevents.{int(time.time())}.stoixamd")
        self._f = open(self.path, "ab")

    @staticmethod
    def _masked_crc(data: bytes) -> int:
        import zlib

        crc = zlib.crc32(data) & 0xFFFFFFFF
        # TF's masked crc32c is actually crc32c; zlib crc32 differs, but
        # TensorBoard tolerates bad CRCs only if told to — so we compute the
        # real crc32c in pure python for header/footer.
        return TensorboardSink._crc32c(data)

    _CRC_TABLE = None

    @staticmethod
    def _crc32c(data: bytes) -> int:
        if TensorboardSink._CRC_TABLE is None:
            poly = 0x82F63B78
            table = []
            for i in range(256):
                c = i
                for _ in range(8):
                    c = (c >> 1) ^ poly if c & 1 else c >> 1
                table.append(c)
            TensorboardSink._CRC_TABLE = table
        crc = 0xFFFFFFFF
        for b in data:
            crc = TensorboardSink._CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
        crc ^= 0xFFFFFFFF
        return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF

    @staticmethod
    def _varint(n: int) -> bytes:
        out = b""
        while True:
            b = n & 0x7F
            n >>= 7
            if n:
                out += bytes([b | 0x80])
            else:
                out += bytes([b])
                return out

    def _scalar_event(self, tag: str, value: float, step: int) -> bytes:
        s = self._struct
        # Summary.Value { tag=1 (string), simple_value=2 (float) }
        tag_b = tag.encode()
        val_msg = b"\x0a" + self._varint(len(tag_b)) + tag_b + b"\x15" + s.pack("<f", float(value))
        # Summary { value=1 repeated }
        summary = b"\x0a" + self._varint(len(val_msg)) + val_msg
        # Event { wall_time=1 double, step=2 int64, summary=5 }
        ev = (
            b"\x09" + s.pack("<d", time.time())
            + b"\x10" + self._varint(step)
            + b"\x2a" + self._varint(len(summary)) + summary
        )
        return ev

    def log(self, metrics: Dict[str, float], t: int, t_eval: int, event: LogEvent) -> None:
        s = self._struct
        for k, v in metrics.items():
            payload = self._scalar_event(f"{event.value}/{k}", v, t)
            hdr = s.pack("<Q", len(payload))
            self._f.write(hdr + s.pack("<I", self._crc32c(hdr)) + payload + s.pack("<I", self._crc32c(payload)))
        self._f.flush()

    def close(self) -> None:
        self._f.close()


class StoixLogger:
    """Thread-safe multi-sink logger facade (reference logger.py:111-127)."""

    def __init__(self, config):
        lcfg = config.logger if hasattr(config, "logger") else config
        self.cfg = lcfg
        run_name = getattr(lcfg, "run_name", None) or f"run_{int(time.time())}"
        base = getattr(lcfg, "base_exp_path", "results")
        self.directory = os.path.join(base, getattr(lcfg, "system_name", "system"), run_name)
        self._lock = threading.Lock()
        self.sinks = []
        use = getattr(lcfg, "loggers", ["console", "json"])
        for name in use:
            if name == "console":
                self.sinks.append(ConsoleSink())
            elif name == "json":
                self.sinks.append(JsonSink(os.path.join(self.directory, "json"), run_name))
            elif name == "csv":
                self.sinks.append(CsvSink(os.path.join(self.directory, "csv"), run_name))
            elif name == "tensorboard":
                self.sinks.append(TensorboardSink(os.path.join(self.directory, "tb"), run_name))
            elif name in ("wandb", "neptune"):
                print(f"[logger] sink '{name}' unavailable offline; skipping")
            else:
                raise ValueError(f"unknown logger sink: {name}")

    def log(self, metrics: Dict[str, Any], t: int, t_eval: int, event: LogEvent) -> None:
        # TRAIN metrics are aggregated mean-only (reference logger.py:152-154)
        scalars = _to_scalar_dict(metrics, mean_only=(event == LogEvent.TRAIN))
        if not scalars:
            return
        with self._lock:
            for sink in self.sinks:
                sink.log(scalars, t, t_eval, event)

    def close(self) -> None:
        with self._lock:
            for sink in self.sinks:
                sink.close()
