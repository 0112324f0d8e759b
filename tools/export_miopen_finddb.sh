#!/bin/bash
# Run ON a GPU box (via gpurun) after a --benchmark bench run: exports the
# MIOpen user find-db so the tuned conv solver choices ship with the repo
# snapshot and later boxes skip the minutes-long exhaustive find.
#
#   gpurun -- 'python bench.py --benchmark --steps 5; tools/export_miopen_finddb.sh'
#
# bench runs then pick it up via MIOPEN_USER_DB_PATH (see tools/use_finddb.sh).
set -e
DEST="$(dirname "$0")/../miopen_db"
mkdir -p "$DEST"
SRC="${MIOPEN_USER_DB_PATH:-$HOME/.config/miopen}"
if [ -d "$SRC" ]; then
  cp -v "$SRC"/*.udb* "$DEST"/ 2>/dev/null || true
  cp -v "$SRC"/*.ufdb* "$DEST"/ 2>/dev/null || true
  echo "exported MIOpen user db from $SRC to $DEST"
else
  echo "no MIOpen user db at $SRC" >&2
fi
