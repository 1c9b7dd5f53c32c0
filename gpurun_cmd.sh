cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
export SBOXGATES_SVC_DEBUG=1
mkdir -p gpurun_out
cat > /tmp/svc_smoke.py <<'PY'
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
t0 = time.perf_counter()
f, r, ev = eng.scan_pool(4, st, t, mask, 0, 82160, 1, True)
print(f"scan1 done in {(time.perf_counter()-t0)*1e3:.1f} ms, ev={ev}", flush=True)
PY
echo "=== grid=64 with stage trace ==="
SBOXGATES_SVC_GRID=64 timeout 50 python -u /tmp/svc_smoke.py 2>&1 | grep -v Warning
echo "rc=$?"
