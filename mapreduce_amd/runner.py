"""Single-node launch helpers.

run_local(): server + N worker threads over an in-process coordinator —
the one-machine multi-worker setup of the reference's test.sh (server in the
foreground, workers detached via screen, test.sh:10-16) without processes.
Used by the unit tests and as the smallest way to run a task.
"""

from __future__ import annotations

import threading
from typing import Optional

from .parallel.coord import LocalCoordinator
from .server import Server
from .worker import Worker


def run_local(params: dict, nworkers: int = 2,
              coord: Optional[LocalCoordinator] = None) -> Server:
    """Run one MapReduce task to completion with threaded workers.

    params is Server.configure() input.  Storage defaults to "mem:<unique>"
    so concurrent tests don't collide.
    """
    coord = coord or LocalCoordinator()
    params = dict(params)
    if "storage" not in params:
        import uuid

        params["storage"] = f"mem:{uuid.uuid4().hex}"
    srv = Server(coord=coord).configure(params)
    workers = []
    threads = []
    for i in range(nworkers):
        w = Worker(coord=coord, name=f"local{i}")
        w.configure({"max_iter": 10 ** 9, "max_tasks": 10 ** 9,
                     "min_sleep": 0.002, "max_sleep": 0.05})
        workers.append(w)
        t = threading.Thread(target=w.execute, daemon=True,
                             name=f"mr-worker-{i}")
        threads.append(t)
        t.start()
    try:
        srv.loop()
    finally:
        for w in workers:
            w.stop()
        for t in threads:
            t.join(timeout=5)
    return srv
