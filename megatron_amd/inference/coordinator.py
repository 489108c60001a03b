"""Data-parallel inference coordinator.

Capability analog of reference megatron/core/inference/
data_parallel_inference_coordinator/ (+ async_zmq_communicator.py): a
front-door that spreads generation requests over several engine replicas
(data-parallel inference), tracks in-flight work per replica, and returns
results to the submitting caller.

MI355X-native design: the reference brokers over ZMQ sockets between
processes; here replicas are driven by worker threads around the in-process
engines (one engine per GPU process in a real deployment — the REST server
wraps this coordinator the same way it wraps a single engine).  Scheduling
is least-loaded-first with a round-robin tiebreak, which is also the
xGMI-friendly choice: replicas never talk to each other, so there is no
cross-GPU traffic to coordinate.
"""

from __future__ import annotations

import itertools
import queue
import threading
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

from megatron_amd.inference.sampling import SamplingParams


@dataclass
class _Pending:
    request_id: int
    prompt: Any
    params: SamplingParams
    done: threading.Event = field(default_factory=threading.Event)
    result: Any = None
    error: Optional[BaseException] = None


class DataParallelCoordinator:
    """Fans requests out to N engine replicas with worker threads.

    Engines must expose `generate(prompts, params) -> [GenerationResult]`
    (both StaticInferenceEngine and DynamicInferenceEngine do).
    """

    def __init__(self, engines: Sequence, max_batch_per_engine: int = 8):
        assert len(engines) > 0
        self.engines = list(engines)
        self.max_batch = max_batch_per_engine
        self._id_gen = itertools.count()
        self._queues: List[queue.Queue] = [queue.Queue() for _ in self.engines]
        self._inflight = [0] * len(self.engines)
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._workers = [
            threading.Thread(target=self._worker, args=(i,), daemon=True)
            for i in range(len(self.engines))
        ]
        for w in self._workers:
            w.start()

    # -- scheduling --------------------------------------------------------
    def _pick_engine(self) -> int:
        with self._lock:
            i = min(range(len(self.engines)), key=lambda j: self._inflight[j])
            self._inflight[i] += 1
            return i

    def submit(self, prompt, params: SamplingParams = SamplingParams()) -> _Pending:
        p = _Pending(request_id=next(self._id_gen), prompt=prompt, params=params)
        self._queues[self._pick_engine()].put(p)
        return p

    def generate(self, prompts: Sequence, params: SamplingParams = SamplingParams()) -> List:
        """Synchronous fan-out/fan-in over all replicas."""
        pending = [self.submit(p, params) for p in prompts]
        for p in pending:
            p.done.wait()
        errs = [p.error for p in pending if p.error is not None]
        if errs:
            raise errs[0]
        return [p.result for p in pending]

    # -- workers -----------------------------------------------------------
    def _worker(self, idx: int):
        eng = self.engines[idx]
        q = self._queues[idx]
        while not self._stop.is_set():
            try:
                first: _Pending = q.get(timeout=0.05)
            except queue.Empty:
                continue
            batch = [first]
            # opportunistic batching of same-params requests
            while len(batch) < self.max_batch:
                try:
                    nxt = q.get_nowait()
                except queue.Empty:
                    break
                if nxt.params != first.params:
                    q.put(nxt)
                    break
                batch.append(nxt)
            try:
                results = eng.generate([b.prompt for b in batch], first.params)
                for b, r in zip(batch, results):
                    b.result = r
            except BaseException as e:  # propagate to the submitter
                for b in batch:
                    b.error = e
            finally:
                with self._lock:
                    self._inflight[idx] -= len(batch)
                for b in batch:
                    b.done.set()

    def stats(self) -> Dict[str, Any]:
        with self._lock:
            return {"replicas": len(self.engines), "inflight": list(self._inflight)}

    def shutdown(self):
        self._stop.set()
        for w in self._workers:
            w.join(timeout=1.0)
