"""Async checkpoint saving.

Reference behavior: paddlenlp/trainer/utils/async_save.py (AsyncSaver —
background-process save of optimizer states) + plugins/shared_memory_utils.py
(state dict -> POSIX shared memory -> daemon writer).  Here the state dict is
snapshotted to pinned CPU memory synchronously (cheap next to a training
step) and a worker thread writes the files, so the training loop never waits
on filesystem IO."""
from __future__ import annotations

import queue
import threading
from typing import Dict, Optional

import torch

from ...utils.log import logger


class AsyncSaver:
    def __init__(self):
        self._queue: "queue.Queue" = queue.Queue()
        self._thread: Optional[threading.Thread] = None
        self._pending = 0
        self._lock = threading.Lock()

    def _ensure_thread(self):
        if self._thread is None or not self._thread.is_alive():
            self._thread = threading.Thread(target=self._worker, daemon=True)
            self._thread.start()

    def _worker(self):
        while True:
            item = self._queue.get()
            if item is None:
                return
            kind, payload, path = item
            try:
                if kind == "safetensors":
                    from safetensors.torch import save_file

                    save_file(payload, path, metadata={"format": "pt"})
                else:
                    torch.save(payload, path)
                logger.info(f"[async-save] wrote {path}")
            except Exception as e:  # pragma: no cover
                logger.error(f"[async-save] failed for {path}: {e}")
            finally:
                with self._lock:
                    self._pending -= 1

    def save_safetensors(self, state_dict: Dict[str, torch.Tensor], path: str):
        snapshot = {k: v.detach().cpu().clone().contiguous() for k, v in state_dict.items()}
        with self._lock:
            self._pending += 1
        self._ensure_thread()
        self._queue.put(("safetensors", snapshot, path))

    def save_torch(self, obj, path: str):
        with self._lock:
            self._pending += 1
        self._ensure_thread()
        self._queue.put(("torch", obj, path))

    def wait_all(self, timeout: float = 600.0):
        import time

        deadline = time.time() + timeout
        while time.time() < deadline:
            with self._lock:
                if self._pending == 0:
                    return True
            time.sleep(0.05)
        return False
