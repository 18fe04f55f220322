#!/bin/bash
# PMC refresh for the two flagship kernels (late round 2).
set -e
REPO=$(pwd)
mkdir -p "$REPO/gpurun_out"
cd /tmp && export TMPDIR=/tmp
cat > /tmp/counters.txt <<'CEOF'
pmc: SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT SQ_WAIT_INST_ANY SQ_WAIT_ANY
CEOF
run_pmc () {
  local tag="$1"; shift
  rm -rf /tmp/pmc_$tag
  timeout 300 rocprofv3 -i /tmp/counters.txt -d /tmp/pmc_$tag -o pmc_$tag -- \
    bash -c "cd $REPO && $*" >/dev/null 2>&1 || true
  local db=$(find /tmp/pmc_$tag -name "*.db" | head -1)
  python3 - "$db" "$tag" <<'PEOF'
import sqlite3, sys
db, tag = sys.argv[1], sys.argv[2]
try:
    con = sqlite3.connect(db)
    rows = con.execute("""
      select p.name, sum(e.value) from rocpd_pmc_event e
      join rocpd_info_pmc p on e.pmc_id = p.id group by p.name""").fetchall()
    for name, val in rows:
        print(f"{tag} {name} {val:.4g}")
except Exception as ex:
    print(tag, "ERR", ex)
PEOF
}
run_pmc attn "python tests/prefill_profile_driver.py"
run_pmc gemm "python tests/gemm256_pmc_driver.py gateup 8 b"
