#!/usr/bin/env bash
# Request-persistence suite (reference `test-persistence` analog).
set -e
cd "$(dirname "$0")/../.."
exec python -m pytest tests/test_wal.py tests/test_store.py -q
