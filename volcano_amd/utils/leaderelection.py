"""File-lock leader election (reference: client-go lease-based leader
election for scheduler/controller-manager HA,
cmd/scheduler/app/server.go:99-124).

Single-node deployment shape: an fcntl lock on a well-known path; the
holder renews mtime, challengers take over when the lease is stale.
"""

from __future__ import annotations

import fcntl
import os
import time
from typing import Optional


class LeaderElector:
    def __init__(self, name: str, lock_dir: str = "/tmp",
                 lease_seconds: float = 15.0):
        self.path = os.path.join(lock_dir, f"volcano-amd-{name}.lock")
        self.lease = lease_seconds
        self._fh = None

    def try_acquire(self) -> bool:
        fh = open(self.path, "a+")
        try:
            fcntl.flock(fh.fileno(), fcntl.LOCK_EX | fcntl.LOCK_NB)
        except OSError:
            fh.close()
            return False
        self._fh = fh
        self.renew()
        return True

    def acquire(self, poll: float = 1.0,
                timeout: Optional[float] = None) -> bool:
        start = time.time()
        while True:
            if self.try_acquire():
                return True
            if timeout is not None and time.time() - start > timeout:
                return False
            time.sleep(poll)

    def renew(self) -> None:
        if self._fh is not None:
            os.utime(self.path)

    @property
    def is_leader(self) -> bool:
        return self._fh is not None

    def release(self) -> None:
        if self._fh is not None:
            fcntl.flock(self._fh.fileno(), fcntl.LOCK_UN)
            self._fh.close()
            self._fh = None
