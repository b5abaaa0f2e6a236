#!/bin/bash
# Build the native modules in-tree (reference install.sh analog).
set -e
python -m gpudpf._build
python -m pytest tests/ -q -m "not gpu"
