#!/usr/bin/env bash
# Wide-ResNet fake-data benchmark (reference examples/wide_resnet).
set -e
cd "$(dirname "$0")/../.."
exec python examples/wide_resnet/train_imagenet.py --stop-at-step "${STOP:-10}" "$@"
