#!/bin/bash
# Example-server launcher (execute_example_server.sh parity): wordcount on
# the repo's own docs, elastic workers via run_example_worker.sh.
CNN=${CNN:-tcp://127.0.0.1:29500}
DB=${DB:-wc}
M=mapreduce_amd.examples.wordcount
exec python -m mapreduce_amd.execute_server "$CNN" "$DB" $M $M $M $M $M $M \
    --storage "${STORAGE:-shared:/tmp/mr_amd_example}" \
    --init-args "{\"files\": [\"README.md\", \"PARITY.md\", \"SURVEY.md\"]}"
