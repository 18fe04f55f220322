#!/bin/bash
# Refresh rocprof kernel-stats evidence for prefill + graphed decode.
# Run ON THE GPU BOX from the repo root via gpurun; writes gpurun_out/.
set -e
REPO=$(pwd)
mkdir -p "$REPO/gpurun_out"
cd /tmp && export TMPDIR=/tmp
rm -rf /tmp/prof_dec /tmp/prof_pre
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_dec -o dec -- \
  bash -c "cd $REPO && python tests/dec_profile_driver.py" > /tmp/dec_stats.txt 2>&1 || true
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_pre -o pre -- \
  bash -c "cd $REPO && python tests/attn_profile_driver.py prefill_story 2>/dev/null || cd $REPO && python -c '
import sys, torch; sys.path.insert(0, \".\")
from bobrapet_amd.models.llama import LlamaModel
m = LlamaModel(\"llama-3-8b\", device=\"cuda\")
ids = torch.randint(0, m.cfg.vocab_size, (4, 2048), device=\"cuda\")
for _ in range(3): m.prefill(ids)
torch.cuda.synchronize(); print(\"done\")
'" > /tmp/pre_stats.txt 2>&1 || true
for t in dec pre; do
  f=$(find /tmp/prof_$t -name "*kernel_stats*" | head -1)
  if [ -n "$f" ]; then cp "$f" "$REPO/gpurun_out/${t}_kernel_stats_r02b.csv"; fi
done
tail -3 /tmp/dec_stats.txt; tail -3 /tmp/pre_stats.txt
echo profile refresh done
