"""Leader election (reference: internal/leader/election.go — LeaseLock).

Process-level analog: an fcntl file lock with a heartbeat lease file. In a
cluster deployment the same Election interface is backed by a K8s Lease
(deploy/ manifests grant the RBAC). is_leader() is the gate the autoscaler
checks every tick.
"""
from __future__ import annotations

import asyncio
import fcntl
import os
import time
from typing import Optional


class Election:
    def __init__(
        self,
        lock_path: str = "/tmp/kubeai-amd-leader.lock",
        lease_duration: float = 15.0,
        identity: Optional[str] = None,
    ):
        self.lock_path = lock_path
        self.lease_duration = lease_duration
        self.identity = identity or f"pid-{os.getpid()}"
        self._is_leader = False
        self._fh = None
        self._task: Optional[asyncio.Task] = None

    def is_leader(self) -> bool:
        return self._is_leader

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
        self._release()

    async def _loop(self) -> None:
        while True:
            if not self._is_leader:
                self._try_acquire()
            else:
                self._heartbeat()
            await asyncio.sleep(self.lease_duration / 3)

    def _try_acquire(self) -> None:
        fh = None
        try:
            fh = open(self.lock_path, "a+")
            fcntl.flock(fh, fcntl.LOCK_EX | fcntl.LOCK_NB)
            fh.seek(0)
            fh.truncate()
            fh.write(f"{self.identity} {time.time()}")
            fh.flush()
            self._fh = fh
            self._is_leader = True
        except (OSError, BlockingIOError):
            if fh is not None:
                fh.close()
            self._is_leader = False

    def _heartbeat(self) -> None:
        try:
            self._fh.seek(0)
            self._fh.truncate()
            self._fh.write(f"{self.identity} {time.time()}")
            self._fh.flush()
        except Exception:
            self._release()

    def _release(self) -> None:
        if self._fh:
            try:
                fcntl.flock(self._fh, fcntl.LOCK_UN)
                self._fh.close()
            except Exception:
                pass
            self._fh = None
        self._is_leader = False
