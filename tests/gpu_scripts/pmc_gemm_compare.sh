#!/bin/bash
# PMC compare gemm256b (16x16x32) vs gemm256w (32x32x16) on one shape.
# Run ON THE GPU BOX from the repo root via gpurun.
set -e
REPO=$(pwd)
cd /tmp && export TMPDIR=/tmp
cat > /tmp/counters.txt <<'CEOF'
pmc: SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT SQ_WAIT_INST_ANY
CEOF
for v in b w; do
  rm -rf /tmp/pmc_$v
  timeout 240 rocprofv3 -i /tmp/counters.txt -d /tmp/pmc_$v -o pmc_$v -- \
    bash -c "cd $REPO && python tests/gemm256_pmc_driver.py gateup 8 $v" >/dev/null 2>&1 || true
  db=$(find /tmp/pmc_$v -name "*.db" | head -1)
  python3 - "$db" "$v" <<'PEOF'
import sqlite3, sys
db, v = sys.argv[1], sys.argv[2]
con = sqlite3.connect(db)
try:
    rows = con.execute("""
      select p.name, sum(e.value) from rocpd_pmc_event e
      join rocpd_info_pmc p on e.pmc_id = p.id group by p.name""").fetchall()
except Exception as ex:
    rows = []
    print(v, "ERR", ex)
for name, val in rows:
    print(f"{v} {name} {val:.4g}")
PEOF
done
