#!/bin/sh
# Example elastic host-discovery script (reference: docs/elastic.rst).
# Print one "host:slots" line per available host; the elastic driver polls
# this every second and re-rendezvouses when the output changes.
echo "127.0.0.1:2"
