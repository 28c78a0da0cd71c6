#!/bin/bash
# Start the in-tree llmq broker as a background daemon.
#
# Replaces the reference's RabbitMQ-in-Singularity launcher
# (reference utils/start_singularity_broker.sh:1-47): the MI355X build
# ships its own asyncio broker, so there is no container image to pull,
# no cookie, no 80-second warmup — just a python process with a durable
# spool directory.
#
# Usage: start_broker.sh [PORT] [DATA_DIR]

set -euo pipefail

PORT="${1:-5672}"
DATA_DIR="${2:-$HOME/.llmq/spool}"
LOG_DIR="${LLMQ_LOG_DIR:-$HOME/.llmq/logs}"
mkdir -p "$LOG_DIR" "$DATA_DIR"

echo "Starting llmq broker on :$PORT (spool: $DATA_DIR)"
nohup python -m llmq_amd broker serve --port "$PORT" --data-dir "$DATA_DIR" \
    > "$LOG_DIR/broker.log" 2>&1 &
BROKER_PID=$!
echo "$BROKER_PID" > "$LOG_DIR/broker.pid"

# Wait for the port to accept connections (the reference sleeps 80 s for
# RabbitMQ; the in-tree broker is up in well under a second).
for _ in $(seq 1 50); do
    if python - "$PORT" <<'EOF'
import socket, sys
try:
    socket.create_connection(("127.0.0.1", int(sys.argv[1])), timeout=1).close()
except OSError:
    sys.exit(1)
EOF
    then
        echo "Broker ready (pid $BROKER_PID)"
        exit 0
    fi
    sleep 0.2
done
echo "Broker did not come up; see $LOG_DIR/broker.log" >&2
exit 1
