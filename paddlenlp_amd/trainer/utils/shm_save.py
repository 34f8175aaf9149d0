"""Shared-memory + writer-process checkpoint saving.

Reference behavior: paddlenlp/trainer/plugins/unified_checkpoint.py:159-299
with shared_memory_utils.py — the state dict is copied into POSIX shared
memory and a daemon process serializes it to disk, so after the (memcpy-
speed) snapshot the training loop never touches the file again.  The
thread-based AsyncSaver (async_save.py) still holds the GIL while
serializing; this writer runs in a separate PROCESS.
"""
from __future__ import annotations

import multiprocessing as mp
from multiprocessing import shared_memory
from typing import Dict, Optional

import numpy as np
import torch

from ...utils.log import logger

_DTYPE_TO_NP = {
    torch.float32: np.float32,
    torch.float16: np.float16,
    torch.bfloat16: np.uint16,   # bit-pattern transport; restored on load
    torch.int64: np.int64,
    torch.int32: np.int32,
    torch.int8: np.int8,
    torch.uint8: np.uint8,
    torch.bool: np.bool_,
}


def _writer_main(task_q, done_q):
    while True:
        item = task_q.get()
        if item is None:
            return
        shm_name, metas, path = item
        try:
            shm = shared_memory.SharedMemory(name=shm_name)
            tensors = {}
            for name, (np_dtype_str, torch_dtype_str, shape, off, nbytes) in metas.items():
                arr = np.frombuffer(shm.buf, dtype=np.dtype(np_dtype_str),
                                    count=nbytes // np.dtype(np_dtype_str).itemsize,
                                    offset=off).reshape(shape)
                t = torch.from_numpy(arr.copy())
                del arr   # release the exported shm pointer before close()
                if torch_dtype_str == "torch.bfloat16":
                    t = t.view(torch.bfloat16)
                tensors[name] = t
            from safetensors.torch import save_file

            save_file(tensors, path, metadata={"format": "pt"})
            shm.close()
            shm.unlink()
            done_q.put((path, None))
        except Exception as e:  # pragma: no cover
            done_q.put((path, str(e)))


class ShmAsyncSaver:
    """state dict -> shared memory -> writer process -> safetensors file."""

    def __init__(self):
        self._ctx = mp.get_context("spawn")
        self._task_q = None
        self._done_q = None
        self._proc: Optional[mp.Process] = None
        self._pending = 0

    def _ensure_proc(self):
        if self._proc is None or not self._proc.is_alive():
            self._task_q = self._ctx.Queue()
            self._done_q = self._ctx.Queue()
            self._proc = self._ctx.Process(
                target=_writer_main, args=(self._task_q, self._done_q), daemon=True)
            self._proc.start()

    def save_safetensors(self, state_dict: Dict[str, torch.Tensor], path: str):
        metas = {}
        total = 0
        cpu = {}
        for name, t in state_dict.items():
            t = t.detach().cpu().contiguous()
            cpu[name] = t
            nbytes = t.numel() * t.element_size()
            np_dtype = _DTYPE_TO_NP[t.dtype]
            metas[name] = (np.dtype(np_dtype).str, str(t.dtype), tuple(t.shape),
                           total, nbytes)
            total += (nbytes + 63) // 64 * 64
        shm = shared_memory.SharedMemory(create=True, size=max(total, 64))
        for name, t in cpu.items():
            _, _, shape, off, nbytes = metas[name]
            src = t.view(torch.uint16) if t.dtype == torch.bfloat16 else t
            dst = np.frombuffer(shm.buf, dtype=np.uint8, count=nbytes, offset=off)
            dst[:] = src.numpy().view(np.uint8).reshape(-1)
            del dst   # release the exported shm pointer before close()
        self._ensure_proc()
        self._task_q.put((shm.name, metas, path))
        shm.close()   # writer holds its own handle; unlinks when done
        self._pending += 1

    def wait_all(self, timeout: float = 600.0) -> bool:
        ok = True
        while self._pending > 0:
            try:
                path, err = self._done_q.get(timeout=timeout)
            except Exception:
                return False
            self._pending -= 1
            if err:
                logger.error(f"[shm-save] failed for {path}: {err}")
                ok = False
            else:
                logger.info(f"[shm-save] wrote {path}")
        return ok

    def shutdown(self):
        if self._proc is not None and self._proc.is_alive():
            self._task_q.put(None)
            self._proc.join(10)
