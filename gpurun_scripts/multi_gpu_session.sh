#!/bin/bash
# Single-lease 8-GPU session: link probe -> sweeps at N=2/4/8 -> BASELINE
# configs 3-5. Run on a node with 8 visible MI355X GPUs:
#   bash gpurun_scripts/multi_gpu_session.sh
set -x
mkdir -p gpurun_out
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=0
NGPU=$(python -c "import torch; print(torch.cuda.device_count())")
echo "GPUs: $NGPU"
# 1) per-pair link probe
timeout 600 python tools/linkprobe.py > gpurun_out/linkprobe.json 2>&1
RUN="python -m torch.distributed.run --nnodes=1 --master-addr 127.0.0.1 --master-port 29317"
# 2) allreduce sweep at each N (BASELINE config 2 at N=2; curve at 4/8)
for N in 2 4 8; do
  [ "$N" -le "$NGPU" ] || continue
  timeout 900 $RUN --nproc-per-node $N bench.py --gpus $N --steps 10 --warmup 3 --sweep \
    > gpurun_out/mg_allreduce_n$N.log 2>&1
done
# 3) config 3: fp32 bcast + scatter/gather, 256 MB, 4 GPUs
for C in bcast scatter gather; do
  timeout 600 $RUN --nproc-per-node 4 bench.py --gpus 4 --steps 10 --warmup 3 \
    --collective $C --bytes $((256<<20)) > gpurun_out/mg_${C}_n4.log 2>&1
done
# 4) config 4: bf16 reduce-scatter + allgather, 8 GPUs
for C in reduce_scatter allgather; do
  timeout 600 $RUN --nproc-per-node 8 bench.py --gpus 8 --steps 10 --warmup 3 \
    --collective $C --dtype bf16 --bytes $((256<<20)) > gpurun_out/mg_${C}_bf16_n8.log 2>&1
done
# 5) config 5: allreduce overlapped with co-resident GEMM, 8 GPUs
timeout 600 $RUN --nproc-per-node 8 bench.py --gpus 8 --steps 10 --warmup 3 \
  --overlap-gemm --bytes $((256<<20)) > gpurun_out/mg_overlap_n8.log 2>&1
grep -h '"metric"' gpurun_out/mg_*.log
