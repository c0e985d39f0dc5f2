#!/usr/bin/env bash
# Launch a TepDist server worker (the reference's launch_worker.sh
# counterpart): ./launch_worker.sh <cluster.json> <task_index>
# Cluster spec format: examples/cluster_1node_template.json
set -e
SPEC=${1:-examples/cluster_1node_template.json}
IDX=${2:-0}
PORT=$(python -c "import json,sys; c=json.load(open('$SPEC')); w=c['workers'][$IDX]; print(w['port'])")
GPUS=$(python -c "import json,sys; c=json.load(open('$SPEC')); w=c['workers'][$IDX]; print(','.join(str(g) for g in w['gpu_ids']))")
export CLUSTER_SPEC=$(cat "$SPEC")
export HIP_VISIBLE_DEVICES=$GPUS
exec python -m tepdist_amd.rpc.server --port "$PORT" --task_index "$IDX"
