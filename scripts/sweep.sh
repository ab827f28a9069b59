#!/usr/bin/env bash
# Attack x defense sweep (reference: scripts/cifar10.sh analog).
# Usage: scripts/sweep.sh [extra main.py args...]
set -u
for attack in signflipping labelflipping noise ipm alie; do
  for agg in mean median trimmedmean krum geomed centeredclipping clippedclustering; do
    echo "=== attack=$attack agg=$agg ==="
    python "$(dirname "$0")/main.py" \
      --attack "$attack" --agg "$agg" "$@" || exit 1
  done
done
