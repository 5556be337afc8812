#!/usr/bin/env bash
# Round-2 gold-plated cfg4: 64 GiB file-mode init at MAINNET scryptN=8192
# with mid-kill resume, disk-backed prove (288 nonces), GPU verify.
# ~35-45 min of box time (labeling 4.29e9 labels at ~2.05 M labels/s).
# Run: /usr/local/graft/bin/gpurun --timeout 3000 -- 'bash tools/r2_cfg4_mainnet.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"
df -B1G /tmp | tail -1
timeout -k 30 2820 python tools/cfg4_soak.py --gib 64 --scrypt-n 8192 \
  --dir /tmp/cfg4main > "$OUT/r2_cfg4_mainnet.log" 2>&1
rc=$?
grep -E '"phase"' "$OUT/r2_cfg4_mainnet.log" || tail -15 "$OUT/r2_cfg4_mainnet.log"
echo "soak rc=$rc"
df -B1G /tmp | tail -1
