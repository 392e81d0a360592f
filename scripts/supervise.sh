#!/usr/bin/env bash
# Process supervisor for framework roles — the pm2-wrapper equivalent of the
# reference's run_miner.sh / run_validator.sh (restart-on-crash with a cap,
# min-uptime rule, version watch + restart; reference lines:
# run_miner.sh:157-228 launch, :220-221 min_uptime/max_restarts,
# :233-268 version watch loop).
#
# Usage: scripts/supervise.sh <role> [role args...]
#   role: miner | validator | averager | bench
# Env:
#   DTA_MAX_RESTARTS   (default 5)   — stop after this many fast crashes
#   DTA_MIN_UPTIME_S   (default 300) — uptime above this resets the counter
#   DTA_VERSION_WATCH  (default "")  — path to a version file; when its
#                                      content changes the role is restarted
#                                      (the reference polls GitHub for a
#                                      __version__ bump; offline equivalent)
set -u

ROLE="${1:?usage: supervise.sh <role> [args...]}"
shift || true
MAX_RESTARTS="${DTA_MAX_RESTARTS:-5}"
MIN_UPTIME="${DTA_MIN_UPTIME_S:-300}"
WATCH="${DTA_VERSION_WATCH:-}"
HERE="$(cd "$(dirname "$0")/.." && pwd)"

restarts=0
last_version=""
[ -n "$WATCH" ] && [ -f "$WATCH" ] && last_version="$(cat "$WATCH")"

while :; do
  start=$(date +%s)
  echo "[supervise] starting role=$ROLE (restart #$restarts)"
  if [ "$ROLE" = bench ]; then
    (cd "$HERE" && exec python bench.py "$@") &
  else
    (cd "$HERE" && exec python -m distributedtraining_amd.cli "$ROLE" "$@") &
  fi
  child=$!

  # watch loop: exit of child, or version change
  while kill -0 "$child" 2>/dev/null; do
    sleep 5
    if [ -n "$WATCH" ] && [ -f "$WATCH" ]; then
      v="$(cat "$WATCH")"
      if [ -n "$last_version" ] && [ "$v" != "$last_version" ]; then
        echo "[supervise] version change ($last_version -> $v): restarting"
        kill "$child" 2>/dev/null
        wait "$child" 2>/dev/null
        last_version="$v"
        restarts=0
        continue 2
      fi
      last_version="$v"
    fi
  done
  wait "$child"
  rc=$?
  uptime=$(( $(date +%s) - start ))
  if [ "$rc" -eq 0 ]; then
    echo "[supervise] role=$ROLE exited cleanly"
    exit 0
  fi
  if [ "$uptime" -ge "$MIN_UPTIME" ]; then
    restarts=0           # long uptime: reset the crash counter (min_uptime)
  else
    restarts=$((restarts + 1))
  fi
  if [ "$restarts" -gt "$MAX_RESTARTS" ]; then
    echo "[supervise] role=$ROLE crashed $restarts times fast; giving up"
    exit "$rc"
  fi
  echo "[supervise] role=$ROLE crashed (rc=$rc, uptime=${uptime}s); restarting"
  sleep 2
done
