"""ActorPool — capability/affinity-routed pool of actor workers.

Reference parity: engine/graph/pool.py (ActorPoolConfig + capability
inference 28-33; per-worker affinity capability "worker::<name>-<idx>"
128-146; idle-queue rotation with capability match 224-250; waiter wakeup
252-273; retry loop 202-219; pool channels 164-189/334-374; cloudpickled
subtask fns with an LRU cache 315-331).

MI355X note: a pool of ``stream:N`` workers is the single-node GPU fleet —
each worker owns a HIP stream; subtasks are kernel launches and the
windowed in-flight limit (Operator) becomes stream depth.
"""
from __future__ import annotations

import asyncio
import collections
from dataclasses import dataclass
from functools import lru_cache
from typing import Any, Dict, List, Optional, Sequence, Tuple

import cloudpickle

from byzpy_amd.actor.base import ActorRef
from byzpy_amd.actor.factory import resolve_backend
from byzpy_amd.graph.subtask import SubTask


@lru_cache(maxsize=64)
def _pickle_fn(fn: Any) -> bytes:
    return cloudpickle.dumps(fn)


class _SubTaskWorker:
    """Constructed ON the actor backend; executes shipped subtask fns.

    ``torch_threads``: out-of-process workers cap their intra-op thread
    count so k worker processes don't oversubscribe the host (reference
    krum.py:425-475 did the same inside its chunk fns)."""

    def __init__(self, torch_threads: int = 0) -> None:
        self._fn_cache: Dict[bytes, Any] = {}
        if torch_threads > 0:
            import torch

            torch.set_num_threads(torch_threads)

    def execute(self, fn_blob: bytes, args: tuple, kwargs: dict) -> Any:
        fn = self._fn_cache.get(fn_blob)
        if fn is None:
            fn = cloudpickle.loads(fn_blob)
            if len(self._fn_cache) < 64:
                self._fn_cache[fn_blob] = fn
        return fn(*args, **kwargs)

    def execute_direct(self, fn: Any, args: tuple, kwargs: dict) -> Any:
        return fn(*args, **kwargs)

    def ping(self) -> str:
        return "pong"


def _looks_dead(e: BaseException) -> bool:
    if isinstance(e, (ConnectionError, EOFError, BrokenPipeError)):
        return True
    return "closed" in str(e).lower() or "pipe" in str(e).lower()


def _infer_capabilities(backend_spec: Any) -> Tuple[str, ...]:
    spec = backend_spec if isinstance(backend_spec, str) else getattr(backend_spec, "scheme", "")
    if isinstance(spec, str) and (spec.startswith(("stream", "gpu"))):
        return ("gpu",)
    return ("cpu",)


@dataclass
class ActorPoolConfig:
    backend: Any = "thread"
    count: int = 1
    capabilities: Optional[Tuple[str, ...]] = None
    name: str = "worker"

    def resolved_capabilities(self) -> Tuple[str, ...]:
        if self.capabilities is not None:
            return tuple(self.capabilities)
        return _infer_capabilities(self.backend)


class _PoolWorker:
    def __init__(
        self,
        backend: Any,
        capabilities: Tuple[str, ...],
        label: str,
        spec: Any = None,
    ) -> None:
        self.backend = backend
        self.capabilities = set(capabilities) | {label}
        self.label = label
        self.spec = spec  # original string spec: enables respawn after a crash
        self.ref = ActorRef(backend)
        self._in_process = getattr(backend, "scheme", "") in ("thread", "stream")
        self.torch_threads = 0  # set by the pool for out-of-process workers

    async def start(self) -> None:
        await self.backend.start()
        await self.backend.construct(_SubTaskWorker, self.torch_threads)

    async def run(self, subtask: SubTask) -> Any:
        if self._in_process:
            return await self.ref.execute_direct(subtask.fn, subtask.args, subtask.kwargs)
        return await self.ref.execute(
            _pickle_fn(subtask.fn), subtask.args, subtask.kwargs
        )

    async def close(self) -> None:
        await self.backend.close()

    async def respawn(self) -> bool:
        """Replace a crashed worker with a fresh backend (string specs
        only — instance backends cannot be re-created)."""
        if not isinstance(self.spec, str):
            return False
        try:
            await self.backend.close()
        except Exception:
            pass
        self.backend = resolve_backend(self.spec)
        self.ref = ActorRef(self.backend)
        await self.backend.start()
        await self.backend.construct(_SubTaskWorker, self.torch_threads)
        return True


class ActorPoolChannel:
    """A named channel bound on every pool worker."""

    def __init__(self, pool: "ActorPool", name: str) -> None:
        self.pool = pool
        self.name = name

    async def send(self, sender: int, recipient: int, payload: Any) -> None:
        target = self.pool.workers[recipient]
        endpoint = target.backend.get_endpoint()
        await self.pool.workers[sender].backend.chan_put(endpoint, self.name, payload)

    async def recv(self, worker: int) -> Any:
        from byzpy_amd.actor.ipc import unwrap_payload

        return unwrap_payload(await self.pool.workers[worker].backend.chan_get(self.name))


class ActorPool:
    def __init__(self, configs: Sequence[ActorPoolConfig] | ActorPoolConfig) -> None:
        if isinstance(configs, ActorPoolConfig):
            configs = [configs]
        self.configs = list(configs)
        self.workers: List[_PoolWorker] = []
        self._idle: collections.deque[int] = collections.deque()
        self._waiters: collections.deque[Tuple[Optional[str], asyncio.Future]] = (
            collections.deque()
        )
        self._started = False

    # -- lifecycle ---------------------------------------------------------
    async def start(self) -> None:
        if self._started:
            return
        for cfg in self.configs:
            caps = cfg.resolved_capabilities()
            for idx in range(cfg.count):
                label = f"worker::{cfg.name}-{len(self.workers)}"
                backend = resolve_backend(
                    cfg.backend if isinstance(cfg.backend, str) else cfg.backend
                )
                self.workers.append(
                    _PoolWorker(backend, caps, label, spec=cfg.backend)
                )
        import os

        n_out = sum(1 for w in self.workers if not w._in_process)
        if n_out:
            per = max(1, (os.cpu_count() or 4) // n_out)
            for w in self.workers:
                if not w._in_process:
                    w.torch_threads = per
        await asyncio.gather(*(w.start() for w in self.workers))
        self._idle.extend(range(len(self.workers)))
        self._started = True

    async def close(self) -> None:
        await asyncio.gather(*(w.close() for w in self.workers), return_exceptions=True)
        self.workers.clear()
        self._idle.clear()
        self._started = False

    # -- introspection -----------------------------------------------------
    @property
    def size(self) -> int:
        return len(self.workers) if self.workers else sum(c.count for c in self.configs)

    @property
    def worker_affinities(self) -> List[str]:
        if self.workers:
            return [w.label for w in self.workers]
        labels, i = [], 0
        for cfg in self.configs:
            for _ in range(cfg.count):
                labels.append(f"worker::{cfg.name}-{i}")
                i += 1
        return labels

    @property
    def prefers_shared_memory(self) -> bool:
        schemes = {
            (c.backend if isinstance(c.backend, str) else getattr(c.backend, "scheme", ""))
            for c in self.configs
        }
        return any(isinstance(s, str) and s.startswith(("process", "tcp")) for s in schemes)

    def _worker_matches(self, idx: int, affinity: Optional[str]) -> bool:
        if affinity is None:
            return True
        return affinity in self.workers[idx].capabilities

    def _any_worker_matches(self, affinity: Optional[str]) -> bool:
        return any(self._worker_matches(i, affinity) for i in range(len(self.workers)))

    # -- scheduling --------------------------------------------------------
    async def _acquire(self, affinity: Optional[str]) -> int:
        if not self._started:
            raise RuntimeError("pool not started")
        if affinity is not None and not self._any_worker_matches(affinity):
            raise RuntimeError(f"no pool worker matches affinity {affinity!r}")
        # rotate the idle queue looking for a capability match
        for _ in range(len(self._idle)):
            idx = self._idle.popleft()
            if self._worker_matches(idx, affinity):
                return idx
            self._idle.append(idx)
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._waiters.append((affinity, fut))
        return await fut

    def _release(self, idx: int) -> None:
        # wake the first waiter this worker satisfies
        for i, (affinity, fut) in enumerate(self._waiters):
            if not fut.done() and self._worker_matches(idx, affinity):
                del self._waiters[i]
                fut.set_result(idx)
                return
        self._idle.append(idx)

    async def run_subtask(self, subtask: SubTask) -> Any:
        attempts = max(1, 1 + int(subtask.max_retries))
        last_err: Optional[BaseException] = None
        for _ in range(attempts):
            idx = await self._acquire(subtask.affinity)
            try:
                return await self.workers[idx].run(subtask)
            except (ConnectionError, EOFError, OSError, RuntimeError) as e:
                # worker may have died (killed process, broken pipe):
                # respawn it so the pool heals before the next attempt
                last_err = e
                if _looks_dead(e):
                    try:
                        await self.workers[idx].respawn()
                    except Exception:  # noqa: BLE001 — stays broken, others serve
                        pass
            except (asyncio.CancelledError, KeyboardInterrupt, SystemExit):
                # cancellation/shutdown must propagate immediately, never be
                # recorded as a retryable worker failure (the finally clause
                # still releases the worker)
                raise
            except BaseException as e:  # noqa: BLE001
                last_err = e
            finally:
                self._release(idx)
        raise last_err  # type: ignore[misc]

    # -- channels ----------------------------------------------------------
    async def open_channel(self, name: str) -> ActorPoolChannel:
        await asyncio.gather(*(w.backend.chan_open(name) for w in self.workers))
        return ActorPoolChannel(self, name)
