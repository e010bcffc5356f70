#!/bin/bash
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
echo "=== EAGER accum1 ==="
timeout 300 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 12 --checkpoint_every 100000 --grad_accum_every 1 \
  --validate_every 100000 --sample_every 100000 --wandb_off --yes --new \
  --data_path ./train_data 2>&1 | grep -aE "^loss|starting" | head -14
echo "=== GRAPHED accum1 ==="
rm -rf ckpts
timeout 300 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 12 --checkpoint_every 100000 --grad_accum_every 1 \
  --graph --validate_every 100000 --sample_every 100000 --wandb_off --yes --new \
  --data_path ./train_data 2>&1 | grep -aE "^loss|captured|starting" | head -16
} > gpurun_out/r02_nan_ab.log 2>&1
tail -40 gpurun_out/r02_nan_ab.log
