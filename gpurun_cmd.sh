cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
SB=sboxgates_amd/sboxes/rijndael.txt
nproc
echo "=== config 2 continuation: resume + jobs 8 ==="
timeout 620 bash -c "time ./bin/sboxgates --resume-dir gpurun_out/aes_gate8 --beam 1 --jobs 8 --seed 11 -v $SB" > gpurun_out/aes_gate8_b.log 2>&1
echo "rc=$?"
grep -E "Resuming|Found|outputs\.|No solution" gpurun_out/aes_gate8_b.log | tail -10
ls gpurun_out/aes_gate8/
echo ALL_DONE
