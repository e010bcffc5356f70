#!/bin/bash
# Final round-2 validation: --graph training now runs ALL in-loop eager
# sections (checkpoint + validation + sampling) via one drop+re-capture.
set -x
cd /root/repo
mkdir -p gpurun_out
{
python tools/make_synthetic_fasta.py 6000
cat > configs/data/synth.toml <<'TOML'
read_from = "./synthetic.fasta"
write_to = "./train_data"
num_samples = 6000
max_seq_len = 1024
prob_invert_seq_annotation = 0.5
fraction_valid_data = 0.05
num_sequences_per_file = 100000
sort_annotations = true
TOML
python generate_data.py --name synth
echo "=== graphed train: ckpt_every=8 valid_every=5 sample_every=10 ==="
timeout 500 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 22 --checkpoint_every 8 --graph --grad_accum_every 1 \
  --validate_every 5 --sample_every 10 --wandb_off --yes --new \
  --data_path ./train_data 2>&1 | \
  grep -E "loss|checkpoint|captured|valid_loss|\*{10}|starting" | head -60
echo "=== bench confirm ==="
timeout 400 python bench.py --steps 30 --warmup 8 2>&1 | tail -2
} > gpurun_out/r02_graph_gates.log 2>&1
tail -70 gpurun_out/r02_graph_gates.log
