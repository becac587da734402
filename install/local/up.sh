#!/usr/bin/env bash
# single-node local platform (no cluster): the embedded daemon
# (apiserver + controllers + local runtime + SCI) — `sub up`.
exec python -m substratus_amd.kube.apiserver
