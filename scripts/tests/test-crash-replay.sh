#!/usr/bin/env bash
# Kill-and-replay integration suite (reference Makefile:41-57 `test-crash`
# named a script that did not exist; this one does).
set -e
cd "$(dirname "$0")/../.."
exec python -m pytest tests/test_crash_integration.py -q -k "kill_and_replay or kill_mid_stream"
