set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
SB=sboxgates_amd/sboxes/rijndael.txt

echo "=== 1. service smoke (tiny, 90s timeout) ==="
timeout 90 python - <<'PY' 2>&1 | tail -5
import sys, time
sys.path.insert(0, '.')
from sboxgates_amd import models
from sboxgates_amd.ops import make_engine, mask_for_inputs
eng = make_engine(gpu="force", seed=1)
sbox, n = models.load("rijndael")
eng.set_sbox(sbox, n)
st = eng.initial_state()
st.grow_pool_random(80, 7)
t = eng.target(0); mask = mask_for_inputs(8)
f, r, ev = eng.scan_pool(4, st, t, mask, 0, 82160, 1, True)
assert ev == 82160, ev
print("service smoke OK", ev)
PY
echo "SMOKE_RC=$?"

echo "=== 2. k4 per-call latency: service vs one-shot ==="
for MODE in svc nosvc; do
  if [ $MODE = nosvc ]; then export SBOXGATES_NO_SVC=1; else unset SBOXGATES_NO_SVC; fi
  timeout 180 python - <<'PY' 2>&1 | tail -4
import sys, time, os
sys.path.insert(0, '.')
from sboxgates_amd import models
from sboxgates_amd.ops import make_engine, mask_for_inputs
eng = make_engine(gpu="force", seed=1)
sbox, n = models.load("rijndael")
eng.set_sbox(sbox, n)
st = eng.initial_state()
st.grow_pool_random(80, 7)
t = eng.target(0); mask = mask_for_inputs(8)
mode = "nosvc" if os.environ.get("SBOXGATES_NO_SVC") else "svc"
for rng in (2048, 16384, 82160):
    for _ in range(20): eng.scan_pool(4, st, t, mask, 0, rng, 1, True)
    N = 1000
    t0 = time.perf_counter()
    for i in range(N): eng.scan_pool(4, st, t, mask, 0, rng, i, True)
    dt = (time.perf_counter() - t0) / N
    print(f"[{mode}] k4 range={rng}: {dt*1e6:.1f} us/call (python-inclusive)")
PY
done
unset SBOXGATES_NO_SVC

echo "=== 3. gate-mode AES bit 0, WITH service ==="
timeout 420 bash -c "time ./bin/sboxgates -o 0 --seed 11 --gpu -v $SB" > gpurun_out/gate_svc.log 2>&1
tail -6 gpurun_out/gate_svc.log

echo "=== 4. GPU test suite ==="
timeout 900 python -m pytest tests/test_gpu.py -x -q 2>&1 | tail -6

echo "=== 5. LUT-mode rocprof kernel breakdown (post-FM_SPLIT) ==="
mkdir -p gpurun_out/prof_r2
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_r2 -o lutbit0 -- /root/repo/bin/sboxgates -l -o 0 --seed 11 --gpu /root/repo/$SB > /root/repo/gpurun_out/lut_prof.log 2>&1
tail -4 /root/repo/gpurun_out/lut_prof.log
grep -A30 "KERNEL" /root/repo/gpurun_out/prof_r2/*lutbit0*stats* 2>/dev/null | head -20 || ls /root/repo/gpurun_out/prof_r2/

echo "=== 6. bench sanity (3 steps + gate search) ==="
cd /root/repo
timeout 420 python bench.py --steps 3 --warmup 1 2>&1 | tail -2
echo ALL_DONE
