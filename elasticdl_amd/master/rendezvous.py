"""Elastic rendezvous for RCCL process groups.

Capability equivalent of the reference's HorovodRendezvousServer
(elasticdl/python/master/rendezvous_server.py:34-167), rebuilt for
torch.distributed over RCCL: the master hosts one long-lived
``torch.distributed.TCPStore``; each rendezvous generation is a key
prefix ``"<rendezvous_id>/"`` inside that store, so re-forming the world
after elasticity events never reuses stale bootstrap keys.

State machine (mirrors the reference):

- ``add_worker`` / ``remove_worker`` stage the *next* host list;
- ``get_worker_host_rank`` flips to the staged world once every member of
  the current world has reported ready (polled its rank), bumping the
  monotonically increasing ``rendezvous_id``;
- a worker not in the current world receives rank -1 and keeps polling.

Workers consume this through
elasticdl_amd.collective.communicator.CommunicatorManager, which tears
down and rebuilds the RCCL communicator whenever ``rendezvous_id``
changes — the RCCL analog of the reference's hvd.shutdown()/init() cycle.
"""

import copy
import threading
import time
from datetime import timedelta
from typing import List, Optional

from elasticdl_amd.common.log_utils import default_logger as logger


class ElasticRendezvousServer:
    def __init__(self, host: str, port: int = 0):
        self._host = host
        self._requested_port = port
        self._port: Optional[int] = None
        self._store = None
        self._lock = threading.Lock()
        self._rendezvous_id = 0
        self._cur_hosts: List[str] = []
        self._next_hosts: Optional[List[str]] = None
        self._ready_hosts = set()
        self._dead_hosts = set()
        self._cur_completed = True
        # flip debounce: a staged world only flips once it has been stable
        # for this long, so racing removals of the same elasticity event
        # coalesce into one generation. Implemented as a deadline (staged_at
        # timestamp), NOT a sleep: sleeping under self._lock would serialize
        # every get_comm_rank poll AND prevent the very membership calls the
        # debounce is meant to observe from landing during the window.
        self._flip_delay_sec = 0.5
        self._staged_at: Optional[float] = None

    # ------------------------------------------------------------- lifecycle
    def start(self) -> int:
        from torch.distributed import TCPStore

        port = self._requested_port or _free_port()
        self._store = TCPStore(
            self._host if self._host not in ("", "0.0.0.0") else "127.0.0.1",
            port,
            is_master=True,
            timeout=timedelta(seconds=3600),
        )
        self._port = port
        logger.info("Rendezvous TCPStore listening on %s:%d", self._host, port)
        return port

    @property
    def port(self) -> int:
        return self._port

    @property
    def rendezvous_id(self) -> int:
        return self._rendezvous_id

    def world_size(self) -> int:
        return len(self._cur_hosts)

    # ------------------------------------------------------------ membership
    def add_worker(self, worker_host: str) -> None:
        with self._lock:
            if not worker_host:
                return
            if self._next_hosts is None:
                self._next_hosts = copy.deepcopy(self._cur_hosts)
            # NOTE: the reference refuses to resurrect an empty world
            # (rendezvous_server.py add_worker) because its master stops
            # the job when every worker dies; here relaunched workers are
            # first-class and MUST be able to re-form a world after a
            # total wipeout, so the refusal is intentionally dropped.
            if worker_host not in self._next_hosts:
                self._next_hosts.append(worker_host)
                self._staged_at = time.monotonic()
                logger.info(
                    "Rendezvous: staged add of %s (next world %s)",
                    worker_host,
                    self._next_hosts,
                )

    def remove_worker(self, worker_host: str) -> None:
        with self._lock:
            if worker_host in self._cur_hosts or (
                self._next_hosts and worker_host in self._next_hosts
            ):
                if self._next_hosts is None:
                    self._next_hosts = copy.deepcopy(self._cur_hosts)
                if worker_host in self._next_hosts:
                    self._next_hosts.remove(worker_host)
                # a dead member can never report ready — drop it from the
                # current world's readiness requirement so the flip is not
                # deadlocked (improves on the reference, which can stall if
                # a worker dies between world formation and completion)
                self._dead_hosts.add(worker_host)
                self._staged_at = time.monotonic()
                logger.info(
                    "Rendezvous: staged removal of %s (next world %s)",
                    worker_host,
                    self._next_hosts,
                )

    def force_reset(self) -> None:
        """Stage a generation bump without membership change — used when a
        worker reports a collective/bootstrap failure so the whole world
        re-forms under a FRESH key prefix (rebuilding the same generation
        would reread half-written bootstrap keys)."""
        with self._lock:
            if self._next_hosts is None:
                self._next_hosts = copy.deepcopy(self._cur_hosts)
            self._staged_at = time.monotonic()
            logger.info("Rendezvous: reset requested (next world %s)",
                        self._next_hosts)

    # ----------------------------------------------------------------- query
    def get_comm_rank(self, worker_host: str) -> dict:
        """One-stop poll for workers: rank/world/rendezvous_id/store addr."""
        with self._lock:
            self._mark_ready(worker_host)
            if (
                self._next_hosts is not None
                and self._cur_completed
                and (
                    self._staged_at is None
                    or time.monotonic() - self._staged_at
                    >= self._flip_delay_sec
                )
            ):
                self._flip()
                self._mark_ready(worker_host)
            rank = (
                self._cur_hosts.index(worker_host)
                if worker_host in self._cur_hosts
                else -1
            )
            return {
                "rank_id": rank,
                "world_size": len(self._cur_hosts),
                "rendezvous_id": self._rendezvous_id,
                "rendezvous_port": self._port or 0,
            }

    def _mark_ready(self, worker_host: str) -> None:
        """Update current-world readiness. Evaluated on EVERY poll (not
        just members'): when all remaining members of the current world
        are dead, any poller — e.g. a relaunched worker waiting to join —
        completes it so the staged world can flip."""
        if self._cur_completed:
            return
        if worker_host in self._cur_hosts:
            self._ready_hosts.add(worker_host)
        if self._ready_hosts >= set(self._cur_hosts) - self._dead_hosts:
            self._cur_completed = True
            self._ready_hosts = set()

    def _flip(self) -> None:
        self._cur_hosts = self._next_hosts
        self._next_hosts = None
        self._staged_at = None
        self._rendezvous_id += 1
        # an empty world can never report readiness — it is trivially
        # complete, so the next staged world can flip immediately
        self._cur_completed = not self._cur_hosts
        self._ready_hosts = set()
        self._dead_hosts = set()
        logger.info(
            "Rendezvous %d: world=%s", self._rendezvous_id, self._cur_hosts
        )


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]
