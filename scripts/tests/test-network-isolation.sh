#!/usr/bin/env bash
# Auth/proxy isolation suite (reference `test-network` analog).
set -e
cd "$(dirname "$0")/../.."
exec python -m pytest tests/test_crash_integration.py -q -k network_isolation
