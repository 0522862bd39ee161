#!/usr/bin/env bash
# Launch an N-process ditask topology on one host (stdlib-TCP bus).
# Usage: local_parallel.sh <module.main> <n_workers>
MAIN=${1:?usage: local_parallel.sh <module.main> <n_workers>}
N=${2:-2}
exec python -m ding.entry.cli_ditask --main "$MAIN" --parallel-workers "$N" --topology mesh
