#!/usr/bin/env bash
# validator launcher (reference: run_validator.sh pm2 wrapper) — supervised with restart
# cap + version watch; pass role flags through.
exec "$(dirname "$0")/supervise.sh" validator "$@"
