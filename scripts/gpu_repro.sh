#!/bin/bash
# Retry gpurun until a slot frees up (rc=3 => nothing charged, retry).
cd /root/repo
for i in $(seq 1 40); do
  /usr/local/graft/bin/gpurun --timeout 900 -- '
mkdir -p gpurun_out
{
echo "== pure torch sanity (no mxnet_amd imported) =="
timeout 180 python3 -c "import torch; x=torch.randn(37,64,device=\"cuda:0\",dtype=torch.float16); torch.cuda.synchronize(); print(\"torch ok\", float(x.float().abs().mean()))"
echo "rc=$?"
echo "== exact driver pytest =="
timeout 420 python3 -m pytest tests/ -x -q -m gpu -p no:cacheprovider 2>&1 | tail -25
echo "rc=$?"
echo "== smoke =="
timeout 240 python3 -c "import sys; sys.path.insert(0,\".\"); import __graft_entry__ as e; e.smoke()" 2>&1 | tail -10
echo "rc=$?"
} > gpurun_out/repro.log 2>&1
tail -60 gpurun_out/repro.log
'
  rc=$?
  echo "[gpu_repro] attempt $i rc=$rc"
  if [ $rc -ne 3 ]; then exit $rc; fi
  sleep 150
done
exit 3
