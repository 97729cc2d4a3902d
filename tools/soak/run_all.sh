#!/usr/bin/env bash
# One-command soak sweep (CPU only, ~10-15 min): every harness once
# with a seed from $1 (default 1).  Any non-zero "bad=" line fails.
set -e
cd "$(dirname "$0")/../.."
S=${1:-1}
P=$((28000 + S % 500))
fail=0
run() { out=$("$@" 2>/dev/null | tail -1); echo "$out";
        case "$out" in *"bad=0"*|*"bad = 0"*) ;; *) fail=1 ;; esac; }
run python tools/soak/dist_soak.py $S $P
run python tools/soak/lubm4_dist_soak.py $((S+1)) $((P+2)) 2
run python tools/soak/lubm4_dist_soak.py $((S+2)) $((P+4)) 4
run python tools/soak/watdiv_dist_soak.py $((S+3)) $((P+6)) 2
run python tools/soak/opt3_dist_soak.py $((S+4)) $((P+8)) 3
run python tools/soak/planned_dist_soak.py $((S+5)) $((P+10))
run python tools/soak/planned_groups_dist.py
run python tools/soak/env_combo_soak.py
exit $fail
