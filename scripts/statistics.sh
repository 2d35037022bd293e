#!/bin/bash
# GPU utilization pollers (reference statistics.sh:1-4), MI355X edition.
# Starts a 500 ms rocm-smi poller writing <variant>_log.csv.
# Usage: ./scripts/statistics.sh <variant-name>
VARIANT=${1:-run}
python - "$VARIANT" <<'EOF'
import signal, sys, time
from amdtrain.utils.monitor import GpuMonitor
mon = GpuMonitor(f"{sys.argv[1]}_log.csv").start()
signal.signal(signal.SIGTERM, lambda *a: sys.exit(0))
try:
    while True:
        time.sleep(1)
finally:
    mon.stop()
EOF
