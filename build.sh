#!/usr/bin/env bash
# Builder-pod entrypoint (reference build.sh:1-16): wait for the shared
# /gordo model volume, then run `gordo build` with the env-provided
# MACHINE / OUTPUT_DIR / MODEL_REGISTER_DIR.
set -e
while [ ! -d /gordo ]; do
  echo "Waiting for /gordo volume..."
  sleep 2
done
exec gordo build
