#!/usr/bin/env bash
# container contract: jupyter on 8888 honoring NOTEBOOK_TOKEN
exec jupyter lab --allow-root --ip=0.0.0.0 --port=8888 \
  --NotebookApp.token="${NOTEBOOK_TOKEN:-default}" --notebook-dir=/content
