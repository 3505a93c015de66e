#!/bin/bash
# ASan+UBSan pass over the native router core (the reference
# runs every test tier under `go test -race`; this is the C++ analog).
set -e
cd "$(dirname "$0")/.."
BIN=$(mktemp /tmp/ldsr_sanitize.XXXXXX)
g++ -std=c++17 -O1 -g -fsanitize=address,undefined -fno-sanitize-recover=all \
    csrc/router/test/sanitize_main.cpp -o "$BIN"
"$BIN"
rm -f "$BIN"
