#!/usr/bin/env bash
# Parity: the reference base image's notebook.sh
# (reference docs/container-contract.md:17-23): start Jupyter on :8888
# honoring $NOTEBOOK_TOKEN, rooted at /content.
set -e
exec jupyter lab --allow-root --ip=0.0.0.0 \
  --NotebookApp.token="${NOTEBOOK_TOKEN:-default}" --notebook-dir=/content
