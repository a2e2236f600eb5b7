#!/bin/bash
# Example-worker launcher (execute_example_worker.sh parity).
CNN=${CNN:-tcp://127.0.0.1:29500}
DB=${DB:-wc}
exec python -m mapreduce_amd.execute_worker "$CNN" "$DB" --max-tasks 1000000
