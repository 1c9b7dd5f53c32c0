cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
echo "=== trial-17 repro ==="
timeout 120 python -u - <<'PY' 2>&1 | tail -12
import sys, struct, random
sys.path.insert(0, '.')
from sboxgates_amd import models
from sboxgates_amd.ops import make_engine, mask_for_inputs
gpu = make_engine(lut_graph=True, seed=1, gpu="force", save_states=False)
cpu = make_engine(lut_graph=True, seed=1, gpu="off", save_states=False)
sbox, n = models.load("rijndael")
gpu.set_sbox(sbox, n); cpu.set_sbox(sbox, n)
# reconstruct trial 17's exact inputs
rng = random.Random(0xF022)
for trial in range(18):
    k = rng.choice([3, 4, 5, 5, 7, 7])
    pool = rng.choice([15, 25, 40, 70])
    if k == 7 and pool > 40: pool = 40
    seedbits = rng.getrandbits(32)
    kind = rng.randrange(3)
    if kind == 0:
        words = [2**64-1]*4
        maskbits = None
    else:
        words = [0,0,0,0]
        nbits = rng.choice([4, 12, 40, 150])
        for _ in range(nbits):
            i = rng.randrange(256); words[i//64] |= 1 << (i % 64)
    tbit = rng.randrange(8)
    import math
    total = math.comb(pool, 3 if k == 4 else k)
    a = rng.randrange(total)
    b = min(total, a + rng.choice([3, 500, 30_000, 200_000]))
st = gpu.initial_state(); st.grow_pool_random(pool, seedbits)
mask = struct.pack("<4Q", *words)
target = gpu.target(tbit)
print("trial17 config:", k, pool, hex(seedbits), tbit, a, b)
f_g, r_g, ev_g = gpu.scan_pool(k, st, target, mask, a, b, 17, True)
f_c, r_c, ev_c = cpu.scan_pool(k, st, target, mask, a, b, 17, True)
print("count_all: ev_g=%d ev_c=%d want=%d f_g=%s f_c=%s" % (ev_g, ev_c, b-a, f_g, f_c))
f_g, r_g, ev_g = gpu.scan_pool(k, st, target, mask, a, b, 17)
f_c, r_c, ev_c = cpu.scan_pool(k, st, target, mask, a, b, 17)
print("early-exit: f_g=%s f_c=%s r_g=%s r_c=%s" % (f_g, f_c, r_g[:5], r_c[:5]))
PY
echo "=== full GPU suite ==="
timeout 1100 python -m pytest tests/test_gpu.py -q 2>&1 | tail -12
echo ALL_DONE
