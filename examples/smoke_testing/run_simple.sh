#!/usr/bin/env bash
# Counterpart of the reference's run_simple.sh: start a server, run the
# simple MLP client against it.
set -e
cd "$(dirname "$0")/../.."
python -m tepdist_amd.rpc.server --port 2233 &
SRV=$!
sleep 2
SERVER_PORT=2233 python examples/smoke_testing/simple.py
kill $SRV
