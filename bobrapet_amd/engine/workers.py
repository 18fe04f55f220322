"""GPU/CPU worker slots: the execution substrate for engram steps.

Role replacement for the reference's Job→pod materialization
(reference: steprun_controller.go:533-724): a step binds to a
(device, stream) slot; the worker thread launches the engram's HIP work on
its dedicated stream and posts completion back to the engine loop —
HIP-event-driven instead of watch/requeue-driven.
"""
from __future__ import annotations

import queue
import threading
import traceback
import typing as _t


class _Slot(threading.Thread):
    def __init__(self, name: str, device: _t.Optional[int], q: "queue.Queue"):
        super().__init__(name=name, daemon=True)
        self.device = device
        self.q = q
        self.stream = None

    def run(self) -> None:
        if self.device is not None:
            import torch

            torch.cuda.set_device(self.device)
            self.stream = torch.cuda.Stream(device=self.device)
        while True:
            item = self.q.get()
            if item is None:
                return
            fn = item
            try:
                if self.stream is not None:
                    import torch

                    with torch.cuda.stream(self.stream):
                        fn(self)
                    # engram completion = all work on the slot stream done
                    self.stream.synchronize()
                else:
                    fn(self)
            except Exception:  # worker must never die
                traceback.print_exc()


class WorkerPool:
    """Per-device worker slots + a CPU pool.

    Device workers pin a torch CUDA stream each, so concurrent steps on one
    GPU overlap via multiple HIP streams."""

    def __init__(
        self,
        device_count: _t.Optional[int] = None,
        workers_per_device: int = 2,
        cpu_workers: int = 4,
        device_ids: _t.Optional[_t.List[int]] = None,
    ):
        if device_ids is None:
            if device_count is None:
                device_count = 0
                try:
                    import torch

                    if torch.cuda.is_available():
                        device_count = torch.cuda.device_count()
                except Exception:
                    device_count = 0
            device_ids = list(range(device_count))
        self.device_ids = list(device_ids)
        self.device_count = len(self.device_ids)
        self._device_queues: _t.Dict[int, queue.Queue] = {}
        self._slots: _t.List[_Slot] = []
        for dev in self.device_ids:
            q: queue.Queue = queue.Queue()
            self._device_queues[dev] = q
            for i in range(workers_per_device):
                slot = _Slot(f"gpu{dev}-w{i}", dev, q)
                slot.start()
                self._slots.append(slot)
        self._cpu_queue: queue.Queue = queue.Queue()
        for i in range(max(cpu_workers, 1)):
            slot = _Slot(f"cpu-w{i}", None, self._cpu_queue)
            slot.start()
            self._slots.append(slot)

    def submit(self, fn: _t.Callable[[_Slot], None], device: _t.Optional[int] = None) -> None:
        if device is not None and device in self._device_queues:
            self._device_queues[device].put(fn)
        elif self._device_queues and device is not None:
            # unknown device index: route to a deterministic owned device
            dev = self.device_ids[device % len(self.device_ids)]
            self._device_queues[dev].put(fn)
        else:
            self._cpu_queue.put(fn)

    def shutdown(self) -> None:
        for q in self._device_queues.values():
            for _ in range(8):
                q.put(None)
        for _ in range(32):
            self._cpu_queue.put(None)
