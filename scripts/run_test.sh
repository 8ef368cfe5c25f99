#!/usr/bin/env bash
# Run the test pyramid (reference scripts/run_test.sh parity).
#   ./scripts/run_test.sh          # CPU tier (no GPU required)
#   ./scripts/run_test.sh gpu      # GPU tier (on an MI355X)
#   ./scripts/run_test.sh all      # both
set -e
cd "$(dirname "$0")/.."
case "${1:-cpu}" in
  cpu) python -m pytest tests/ -q -m "not gpu" ;;
  gpu) python -m pytest tests/ -q -m gpu ;;
  all) python -m pytest tests/ -q -m "not gpu" && python -m pytest tests/ -q -m gpu ;;
  *) echo "usage: $0 [cpu|gpu|all]"; exit 2 ;;
esac
