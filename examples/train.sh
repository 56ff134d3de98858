#!/bin/bash
# Launch training (reference examples/train.sh UX):
#   CONF_FILE=hf_llama3_8B_config ./train.sh [extra key=value overrides]
# TRAIN_ITERS=N bounds max_steps (smoke runs, reference orchestrator :48-58).
set -euo pipefail
SCRIPT_DIR="$(cd "$(dirname "${BASH_SOURCE[0]}")" && pwd)"
source "$SCRIPT_DIR/train_setup.sh"

: "${CONF_FILE:=hf_llama3_8B_config}"
CONF_PATH="$SCRIPT_DIR/conf/${CONF_FILE}.yaml"
[ -f "$CONF_PATH" ] || { echo "config not found: $CONF_PATH"; exit 1; }

# read devices from the YAML (reference train.sh reads `devices:`)
DEVICES=$(python - "$CONF_PATH" <<'EOF'
import sys, yaml
cfg = yaml.safe_load(open(sys.argv[1]))
print(cfg.get("trainer", {}).get("devices", 8))
EOF
)

echo "launching: $CONF_FILE on $DEVICES devices/node, $NNODES node(s)"
torchrun $DISTRIBUTED_ARGS --nproc-per-node "$DEVICES" \
    "$SCRIPT_DIR/training.py" --config "$CONF_PATH" "$@"
