#!/bin/bash
# Round-2 end-to-end training evidence: the round-1 recipe re-run on the
# round-2 kernels (8-wave attn bwd, tr-read fwd, fast-tanh GLU, colsum
# dbias) + the graphed-checkpoint path (--graph now saves).
set -x
cd /root/repo
mkdir -p gpurun_out
{
python tools/make_synthetic_fasta.py 20000
cat > configs/data/synth.toml <<'TOML'
read_from = "./synthetic.fasta"
write_to = "./train_data"
num_samples = 20000
max_seq_len = 1024
prob_invert_seq_annotation = 0.5
fraction_valid_data = 0.025
num_sequences_per_file = 100000
sort_annotations = true
TOML
python generate_data.py --name synth
echo "=== train 120 steps (eager, checkpoints) ==="
timeout 700 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 120 --checkpoint_every 100 \
  --validate_every 50 --sample_every 1000 --wandb_off --yes --new \
  --data_path ./train_data 2>&1 | grep -E "loss|checkpoint|params|sequence" | head -40
echo "=== resume ==="
timeout 400 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 10 --checkpoint_every 100 \
  --validate_every 1000 --sample_every 1000 --wandb_off \
  --data_path ./train_data 2>&1 | grep -E "loss|starting|params" | head -12
echo "=== graphed train WITH in-loop checkpoint (drop+recapture) ==="
timeout 500 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 25 --checkpoint_every 10 --graph --grad_accum_every 1 \
  --validate_every 1000 --sample_every 1000 --wandb_off \
  --data_path ./train_data 2>&1 | grep -E "loss|checkpoint|captured|starting" | head -30
echo "=== sample from trained ckpt (graphed decode) ==="
timeout 300 python sample.py --cached --graph --prime "# M" 2>&1 | tail -4
} > gpurun_out/r02_evidence.log 2>&1
tail -80 gpurun_out/r02_evidence.log
