#!/bin/bash
# Soak scenario (reference benchmarks/README.md §5: 30-60 min sustained
# load).  DUR defaults to 1800 s; emits the same CSV schema plus a
# post-run health snapshot so leaks/ejections show up.
set -e
cd "$(dirname "$0")"
DUR=${DUR:-1800} CONNS=${CONNS:-100} ./run_local.sh
