#!/usr/bin/env bash
# miner launcher (reference: run_miner.sh pm2 wrapper) — supervised with restart
# cap + version watch; pass role flags through.
exec "$(dirname "$0")/supervise.sh" miner "$@"
