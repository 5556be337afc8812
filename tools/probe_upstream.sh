#!/usr/bin/env bash
# Probe for upstream parity-pinning material.
# The RESTATED protocol layouts can only be upgraded to
# PINNED with material produced by the reference's own toolchain:
#   - post-rs v0.7.13 sources or a libpost.so binary (Makefile-libs.Inc:49)
#   - spacemeshos/post v0.12.9 Go sources (go.mod:48)
#   - spacemeshos/api v1.55.0 protobuf descriptors (go.mod:42)
#   - a RandomX library (k2pow, activation/post_types.go:84-144)
#   - a Go toolchain (compile the cgo shim) or Rust toolchain (build post-rs)
# This script records what the current round's environment can and cannot
# reach; its committed output is the decision record for keeping the
# RESTATED stance. Re-run each round.
set -u
echo "# Upstream pinning probe — $(date -u +%Y-%m-%dT%H:%MZ 2>/dev/null || echo unknown-date)"
echo
echo '```'
echo "## toolchains"
for t in go rustc cargo javac; do
  printf '%-8s: %s\n' "$t" "$(command -v $t >/dev/null 2>&1 && $t --version 2>&1 | head -1 || echo ABSENT)"
done
echo
echo "## network egress"
timeout 5 python3 - <<'EOF' 2>&1
import socket
for host in ("proxy.golang.org", "github.com", "pypi.org"):
    try:
        socket.create_connection((host, 443), timeout=3)
        print(f"{host}: REACHABLE")
    except OSError as e:
        print(f"{host}: unreachable ({e})")
EOF
echo
echo "## pip index"
timeout 20 pip download --no-deps --dest /tmp/_probe_pip randomx 2>&1 | tail -1
timeout 20 pip download --no-deps --dest /tmp/_probe_pip spacemesh-api 2>&1 | tail -1
rm -rf /tmp/_probe_pip
echo
echo "## filesystem search (post-rs / libpost / RandomX / spacemesh protos)"
find / -xdev \( -iname "*randomx*" -o -iname "libpost*" -o -iname "post.h" \
  -o \( -iname "*.proto" -path "*spacemesh*" \) \) \
  -not -path "/proc/*" -not -path "/sys/*" -not -path "/root/repo/*" \
  -not -path "/tmp/*" 2>/dev/null | head -20 || true
echo "(paths under /root/repo excluded: those are this build's own artifacts)"
echo
echo "## reference tree: is the engine vendored?"
ls /root/reference/vendor 2>/dev/null || echo "/root/reference/vendor: ABSENT (reference fetches libpost.so prebuilt, Makefile-libs.Inc:93-120)"
grep -c "spacemeshos/post\|spacemeshos/api" /root/reference/go.sum 2>/dev/null | sed 's/^/go.sum entries naming post\/api (hashes only, no sources): /'
echo '```'
