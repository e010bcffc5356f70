#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
python tools/make_synthetic_fasta.py 2000
cat > configs/data/synth.toml <<'TOML'
read_from = "./synthetic.fasta"
write_to = "./train_data"
num_samples = 2000
max_seq_len = 1024
prob_invert_seq_annotation = 0.5
fraction_valid_data = 0.05
num_sequences_per_file = 100000
sort_annotations = true
TOML
python generate_data.py --name synth
timeout 300 python train.py --model_name small --mixed_precision \
  --batch_size 32 --max_steps 3 --checkpoint_every 1000 --graph \
  --validate_every 1000 --sample_every 1000 --wandb_off --yes --new \
  --data_path ./train_data 2>&1 | head -80
} > gpurun_out/r02_graph_repro.log 2>&1
tail -80 gpurun_out/r02_graph_repro.log
