#!/bin/bash
# Retry wrapper for gpurun: exit code 3 = no slot free (nothing charged),
# retry with backoff. Usage: run_gpu.sh <timeout_s> <max_tries> -- <cmd>
set -u
TIMEOUT=$1; TRIES=$2; shift 3
for i in $(seq 1 "$TRIES"); do
  /usr/local/graft/bin/gpurun --timeout "$TIMEOUT" -- "$@"
  rc=$?
  if [ $rc -ne 3 ]; then exit $rc; fi
  echo "[run_gpu] slot busy (try $i/$TRIES); sleeping 120s"
  sleep 120
done
exit 3
