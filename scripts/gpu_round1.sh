#!/bin/bash
# First GPU validation pass: pytest -m gpu, bench.py, rocprofv3 kernel stats.
set -x
cd /root/repo
mkdir -p gpurun_out
python -c "import torch; print('torch', torch.__version__, 'gpus', torch.cuda.device_count(), torch.cuda.get_device_name(0))" 2>&1 | tail -2

echo "=== GPU TESTS ==="
timeout 420 python -m pytest tests/ -q -m gpu 2>&1 | tail -25

echo "=== BENCH N=1 ==="
timeout 240 python bench.py --steps 8 --warmup 2 > gpurun_out/bench_n1.json 2> gpurun_out/bench_n1.err
tail -3 gpurun_out/bench_n1.err
cat gpurun_out/bench_n1.json

echo "=== TOPOLOGY PROBE ==="
timeout 120 python -c "
from grove_amd.topology.agent import probe, discover_node, xgmi_hives
import json
info = probe()
print('backend', info.get('backend'), 'gpus', info.get('gpu_count'))
print('hives', xgmi_hives(info))
links = info.get('links') or []
print('links sample', links[:4])
n = discover_node('box')
print('node labels', n['metadata']['labels'])
json.dump(info, open('gpurun_out/topology.json','w'), indent=1, default=str)
" 2>&1 | tail -8

echo "=== ROCPROF KERNELS ==="
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o kern -- \
  python -c "
import sys; sys.path.insert(0, '/root/repo')
from grove_amd.kubelet.gpunode import load_gpuwork
e = load_gpuwork()
print('gemm 4096^3 TF/s:', e.burn_gemm(4096, 4096, 4096, 16))
print('gemm 8192^3 TF/s:', e.burn_gemm(8192, 8192, 8192, 4))
print('triad GB/s:', e.stream_triad(1 << 26, 10))
" 2>&1 | tail -6
ls -la /root/repo/gpurun_out/prof/ 2>/dev/null | head
find /root/repo/gpurun_out/prof -name "*stats*" | head -5
echo DONE
