#!/bin/bash
# Build the torchx_amd job image (run from the repo root):
#   bash torchx_amd/runtime/container/build.sh
set -ex
docker build -t torchx_amd -f torchx_amd/runtime/container/Dockerfile .
