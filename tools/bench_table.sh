#!/bin/bash
# Full benchmark table (fixed-work steps), run on an MI355X via gpurun.
mkdir -p gpurun_out
run() {
  timeout 900 python bench.py --model $1 --dtype $2 --diff $3 --schur $4 --steps $5 --warmup 2 2>/dev/null | tail -1
}
{
run ladybug49 float64 auto explicit 10
run trafalgar257 float64 auto explicit 10
run trafalgar257 float64 analytical explicit 10
run venice1778 float64 auto explicit 10
run venice1778 float64 analytical explicit 10
run venice1778 float64 auto implicit 10
run venice1778 float32 auto implicit 10
run venice1778 float32 analytical explicit 10
run final13682 float32 analytical implicit 4
run final13682 float32 analytical explicit 4
run synth20k float64 auto implicit 4
run synth20k float64 auto explicit 4
} > gpurun_out/bench_table.jsonl
python - <<'PYEOF'
import json
for l in open("gpurun_out/bench_table.jsonl"):
    l = l.strip()
    if not l:
        continue
    d = json.loads(l)
    c = d["config"]
    print("%-30s %-5s %-10s %-8s %9.2f ms/step" % (
        c["model"], d["dtype"], c["diff"], c["schur"], d["ms_per_step"]))
PYEOF
