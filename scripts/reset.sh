#!/usr/bin/env bash
# Full state wipe (reference: scripts/reset.sh:6-11 — etcdctl del + rm merges).
# Memory/WAL backend: remove the data dir. etcd backend: delete the prefix
# via the JSON gateway (no etcdctl dependency).
set -euo pipefail

DATA_DIR="${DATA_DIR:-./.state}"
ETCD="${ETCD:-}"

if [[ -n "$ETCD" ]]; then
  key=$(printf '/gpu-docker-api/' | base64)
  end=$(printf '/gpu-docker-api0' | base64)
  curl -s -X POST "$ETCD/v3/kv/deleterange" \
    -d "{\"key\": \"$key\", \"range_end\": \"$end\"}" > /dev/null
  echo "cleared etcd prefix /gpu-docker-api/ at $ETCD"
fi

rm -rf "$DATA_DIR"
echo "removed $DATA_DIR"
