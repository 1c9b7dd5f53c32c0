cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
export SBOXGATES_SVC_DEBUG=1
mkdir -p gpurun_out
cat > /tmp/svc_smoke.py <<'PY'
import sys, time
sys.path.insert(0, '.')
from sboxgates_amd import models
from sboxgates_amd.ops import make_engine, mask_for_inputs
print("creating engine...", flush=True)
eng = make_engine(gpu="force", seed=1)
sbox, n = models.load("rijndael")
eng.set_sbox(sbox, n)
st = eng.initial_state()
st.grow_pool_random(80, 7)
t = eng.target(0); mask = mask_for_inputs(8)
print("first scan...", flush=True)
t0 = time.perf_counter()
f, r, ev = eng.scan_pool(4, st, t, mask, 0, 82160, 1, True)
print(f"scan1 done in {(time.perf_counter()-t0)*1e3:.1f} ms, ev={ev}", flush=True)
assert ev == 82160, ev
N = 200
t0 = time.perf_counter()
for i in range(N):
    f, r, ev = eng.scan_pool(4, st, t, mask, 0, 2048, i, True)
    assert ev == 2048
dt = (time.perf_counter()-t0)/N
print(f"200 small scans: {dt*1e6:.1f} us/call", flush=True)
print("SMOKE_ALL_OK", flush=True)
PY

echo "=== A. grid=8 smoke ==="
SBOXGATES_SVC_GRID=8 timeout 60 python -u /tmp/svc_smoke.py > gpurun_out/svc_a.log 2>&1
echo "A_rc=$?"; tail -12 gpurun_out/svc_a.log

echo "=== B. grid=64 smoke ==="
SBOXGATES_SVC_GRID=64 timeout 60 python -u /tmp/svc_smoke.py > gpurun_out/svc_b.log 2>&1
echo "B_rc=$?"; tail -8 gpurun_out/svc_b.log

echo "=== C. default grid smoke ==="
timeout 90 python -u /tmp/svc_smoke.py > gpurun_out/svc_c.log 2>&1
echo "C_rc=$?"; tail -8 gpurun_out/svc_c.log

echo "=== dmesg tail ==="
dmesg 2>/dev/null | tail -15
