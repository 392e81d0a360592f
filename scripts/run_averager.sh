#!/usr/bin/env bash
# averager launcher (reference: run_averager.sh pm2 wrapper) — supervised with restart
# cap + version watch; pass role flags through.
exec "$(dirname "$0")/supervise.sh" averager "$@"
