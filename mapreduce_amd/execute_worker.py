"""Generic worker CLI (execute_worker.lua:7-11 analogue).

    python -m mapreduce_amd.execute_worker tcp://HOST:PORT DBNAME \
        [--max-iter N] [--max-sleep S] [--max-tasks N]

Connects to the server's TCPStore control plane and claims jobs until
max_tasks tasks are done (or the task finishes).
"""

from __future__ import annotations

import argparse
import sys


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("cnn")
    p.add_argument("db")
    p.add_argument("--max-iter", type=int, default=20)
    p.add_argument("--max-sleep", type=float, default=20.0)
    p.add_argument("--max-tasks", type=int, default=1)
    p.add_argument("--verbose", action="store_true")
    args = p.parse_args(argv)

    from .worker import Worker

    w = Worker(args.cnn, args.db).configure({
        "max_iter": args.max_iter,
        "max_sleep": args.max_sleep,
        "max_tasks": args.max_tasks,
        "verbose": args.verbose,
    })
    w.execute()
    return 0


if __name__ == "__main__":
    sys.exit(main())
