#!/bin/bash
# Round-2 GPU session 1: regression of the GPU test tier with the new
# 2-phase/wire-contract code, plus the probes that decide this round's work:
#  - partition-switch writability on this round's pool
#  - ROC_GLOBAL_CU_MASK real enforcement measurement
#  - amdsmi event API surface for the health path
#  - GPU count (multi-GPU items feasibility)
set -x
OUT=gpurun_out/r2s1
mkdir -p "$OUT"
cd /root/repo

python -c "import torch; print('torch', torch.__version__, 'gpus', torch.cuda.device_count())" 2>&1 | tail -2 | tee "$OUT/env.txt"
rocm-smi --showhw 2>&1 | head -20 >> "$OUT/env.txt" || true

# 0. rebuild in place (fast no-op if current) so the GPU tier runs this tree's code
timeout 300 python -c "import __graft_entry__ as g; g.build()" > "$OUT/build.log" 2>&1 || { echo BUILD_FAIL; tail -20 "$OUT/build.log"; }

# 1. GPU test tier
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -15 | tee "$OUT/pytest_gpu.txt"

# 2. partition-switch writability probe (read-only checks + one guarded write attempt)
timeout 120 python - > "$OUT/partition_probe.txt" 2>&1 <<'EOF'
import glob, os, subprocess
for f in glob.glob('/sys/class/drm/card*/device/current_compute_partition'):
    print(f, open(f).read().strip(), 'writable=', os.access(f, os.W_OK))
for f in glob.glob('/sys/class/drm/card*/device/current_memory_partition'):
    print(f, open(f).read().strip(), 'writable=', os.access(f, os.W_OK))
for f in glob.glob('/sys/class/drm/card*/device/available_compute_partition'):
    print(f, open(f).read().strip())
# guarded amdsmi attempt (returns error on pool per round 1; re-verify this round)
try:
    import amdsmi
    amdsmi.amdsmi_init()
    h = amdsmi.amdsmi_get_processor_handles()[0]
    try:
        cur = amdsmi.amdsmi_get_gpu_compute_partition(h)
        print('amdsmi current partition:', cur)
    except Exception as e:
        print('amdsmi get partition failed:', e)
    try:
        amdsmi.amdsmi_set_gpu_compute_partition(h, amdsmi.AmdSmiComputePartitionType.SPX)
        print('amdsmi SET SPX->SPX: OK (writable pool!)')
    except Exception as e:
        print('amdsmi set partition failed:', type(e).__name__, e)
    amdsmi.amdsmi_shut_down()
except Exception as e:
    print('amdsmi unavailable:', e)
r = subprocess.run(['amd-smi', 'partition', '--accelerator'], capture_output=True, text=True, timeout=60)
print('amd-smi partition rc=', r.returncode)
print(r.stdout[:2000]); print(r.stderr[:500])
EOF

# 3. ROC_GLOBAL_CU_MASK enforcement: MFMA burn with 1 XCD (32 CUs) vs full chip
timeout 240 python - > "$OUT/cumask_probe.txt" 2>&1 <<'EOF'
import os, subprocess, sys
code = "from k8s_dra_driver_gpu_amd.fabric import probe; print(probe.mfma_bf16_tflops(0, 2048, 10))"
def run(env_extra):
    env = dict(os.environ); env.update(env_extra)
    r = subprocess.run([sys.executable, '-c', code], capture_output=True, text=True, timeout=120, env=env)
    return r.stdout.strip(), r.stderr[-300:]
full, err1 = run({})
print('full-chip TFLOPs:', full, err1 if not full else '')
one, err2 = run({'ROC_GLOBAL_CU_MASK': '0xffffffff'})
print('1-XCD (32 CU) TFLOPs:', one, err2 if not one else '')
two, err3 = run({'ROC_GLOBAL_CU_MASK': hex((1<<64)-1)})
print('2-XCD (64 CU) TFLOPs:', two, err3 if not two else '')
try:
    ratio = float(one) / float(full)
    print(f'ratio 1xcd/full = {ratio:.3f} (expect ~0.125 if mask enforces)')
except Exception as e:
    print('ratio unavailable:', e)
EOF

# 4. amdsmi event/RAS API surface
timeout 90 python - > "$OUT/event_api.txt" 2>&1 <<'EOF'
import amdsmi, inspect
fns = [n for n in dir(amdsmi) if any(k in n.lower() for k in ('event','ras','ecc','xgmi_error'))]
print('\n'.join(sorted(fns)))
amdsmi.amdsmi_init()
h = amdsmi.amdsmi_get_processor_handles()[0]
for name in ('amdsmi_get_gpu_total_ecc_count','amdsmi_get_gpu_ecc_count','amdsmi_get_gpu_ras_feature_info','amdsmi_get_gpu_ras_block_features_enabled'):
    fn = getattr(amdsmi, name, None)
    if fn is None: continue
    try:
        print(name, '->', str(fn(h))[:300])
    except Exception as e:
        print(name, 'ERR', type(e).__name__, str(e)[:120])
# event notification lifecycle
try:
    amdsmi.amdsmi_init_gpu_event_notification(h)
    masks = amdsmi.AmdSmiEvtNotificationType
    print('evt types:', [m.name for m in masks])
    amdsmi.amdsmi_set_gpu_event_notification_mask(h, 0xFFFFFFFF)
    evts = amdsmi.amdsmi_get_gpu_event_notification(1000)  # 1s timeout
    print('events (1s poll):', evts)
    amdsmi.amdsmi_stop_gpu_event_notification(h)
except Exception as e:
    print('event notification path:', type(e).__name__, str(e)[:200])
amdsmi.amdsmi_shut_down()
EOF

# 5. short bench regression
timeout 300 python bench.py --steps 10 --warmup 2 > "$OUT/bench.json" 2> "$OUT/bench.err" || tail -5 "$OUT/bench.err"
tail -1 "$OUT/bench.json"
echo DONE
