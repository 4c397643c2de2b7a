#!/usr/bin/env bash
# BASELINE.json config matrix on one 8-GPU MI355X node (round-2 runner).
# Each block prints one JSON result line; collect into results/.
set -x
mkdir -p results
TR="python -m torch.distributed.run --nnodes=1 --master-addr 127.0.0.1"

# config 2: Llama-2 7B TP=8 bf16 pretrain, seq 4096 (the headline)
$TR --nproc-per-node 8 bench.py --gpus 8 --steps 8 --warmup 3 \
    | tee results/llama2_7b_tp8.json

# config 3: Llama-2 7B TP=2 x PP=4, 1F1B
$TR --nproc-per-node 8 bench.py --gpus 8 --steps 8 --warmup 3 \
    --tp 2 --pp 4 | tee results/llama2_7b_tp2_pp4.json

# config 4: Llama-3 70B TP=8 + SP + ZeRO-1, seq 8192
$TR --nproc-per-node 8 bench.py --gpus 8 --steps 4 --warmup 2 \
    --model llama3-70b --seq 8192 --batch 8 --microbatch 1 \
    | tee results/llama3_70b_tp8_seq8192.json

# config 5: Llama-3 8B inference TP=8, GQA KV-cache decode, batch 32
$TR --nproc-per-node 8 bench_infer.py \
    | tee results/llama3_8b_decode_tp8.json

# config 2 again with the ring-pipelined SP comm overlap (round-2
# machinery, opt-in until xGMI-measured — compare against the first run)
NXDA_SP_OVERLAP=1 $TR --nproc-per-node 8 bench.py --gpus 8 --steps 8 \
    --warmup 3 | tee results/llama2_7b_tp8_spoverlap.json

# scaling curve (weak): N = 1, 2, 4
for N in 1 2 4; do
  $TR --nproc-per-node $N bench.py --gpus $N --steps 6 --warmup 2 \
      | tee results/llama2_7b_tp${N}.json
done
