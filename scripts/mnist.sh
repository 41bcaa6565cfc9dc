#!/bin/bash
# MNIST convergence example on N local GPUs (reference mnist.sh equivalent).
N=${1:-2}
exec python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
    --nproc-per-node "$N" "$(dirname "$0")/../examples/mnist.py" "${@:2}"
