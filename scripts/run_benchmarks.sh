#!/usr/bin/env bash
# Launch recipes for every BASELINE.json config (run_deepreduce.sh equivalent,
# torchrun/RCCL instead of mpirun/Horovod).  One process per GPU over RCCL
# ("nccl" backend IS RCCL on ROCm); HSA_ENABLE_IPC_MODE_LEGACY=0 required for
# dmabuf IPC on this pool.
set -euo pipefail
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}

NGPUS=${NGPUS:-8}
STEPS=${STEPS:-30}
WARMUP=${WARMUP:-10}

launch() {
  local n=$1; shift
  if [ "$n" -gt 1 ]; then
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n" \
      --master-addr 127.0.0.1 --master-port 29560 \
      bench.py --gpus "$n" --steps "$STEPS" --warmup "$WARMUP" "$@"
  else
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP" "$@"
  fi
}

case "${1:-all}" in
  config1)  # ResNet-20/CIFAR-10 top-k 1% + residual, gloo CPU ws=2 (no GPU)
    python -m pytest tests/test_distributed.py -q -k resnet20 ;;
  config2)  # ResNet-50 top-k 1% + Bloom index, N GPUs
    launch "$NGPUS" --model resnet50 --deepreduce index --index bloom ;;
  config3)  # ResNet-50 top-k 1% + polyfit value compression
    launch "$NGPUS" --model resnet50 --deepreduce value --value polyfit ;;
  config4)  # NCF embedding-heavy sparse grads
    launch "${NGPUS_NCF:-4}" --model ncf --batch 65536 --deepreduce index --index bloom ;;
  config5)  # BERT-base top-k 0.1% 'both' (bloom + polyfit)
    launch "$NGPUS" --model bert --batch 16 --compress-ratio 0.001 \
      --deepreduce both --index bloom --value polyfit ;;
  dense)    # uncompressed RCCL all-reduce baseline (the 1.0 of rel. volume)
    launch "$NGPUS" --model resnet50 --deepreduce dense ;;
  scaling)  # 1/2/4/8 scaling curve for config 2
    for n in 1 2 4 8; do launch "$n" --model resnet50 --deepreduce index; done ;;
  all)
    for c in config2 config3 config4 config5 dense; do "$0" "$c"; done ;;
  *) echo "usage: $0 {config1..config5|dense|scaling|all}"; exit 1 ;;
esac
