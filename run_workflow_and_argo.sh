#!/usr/bin/env bash
# Deploy entrypoint (reference run_workflow_and_argo.sh): generate the
# Argo workflow for a config and submit it.
set -e
: "${MACHINE_CONFIG:?need MACHINE_CONFIG}"
: "${PROJECT_NAME:?need PROJECT_NAME}"
gordo workflow generate \
  --machine-config "$MACHINE_CONFIG" \
  --project-name "$PROJECT_NAME" \
  --output-file /tmp/workflow.yml "$@"
argo submit /tmp/workflow.yml
