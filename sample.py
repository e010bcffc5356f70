"""Sampling CLI (reference parity: /root/reference/sample.py).

Loads the lexically-last checkpoint, rebuilds the model FROM the stored
model_config (not the toml — reference: sample.py:46-47), primes with the
given string and emits a gumbel-max top-k sample.
"""

import click
import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.checkpoint import get_checkpoint_fns, numpy_to_tensors
from progen_amd.data import decode_tokens, encode_tokens
from progen_amd.utils import load_dotenv, sample, sample_fast

load_dotenv()  # reference: sample.py:1-2


@click.command()
@click.option('--seed', default=42)
@click.option('--checkpoint_path', default='./ckpts')
@click.option('--prime', default='')
@click.option('--fast', default=False, is_flag=True,
              help='length-growing decode with EOS early-exit (identical tokens)')
@click.option('--cached', default=False, is_flag=True,
              help='incremental decode with per-layer recurrent caches '
                   '(O(window) per token; identical semantics)')
@click.option('--graph', default=False, is_flag=True,
              help='with --cached on GPU: replay the per-token step as one '
                   'captured hipGraph (~2x the eager cached step)')
def main(seed, checkpoint_path, prime, fast, cached, graph):
    from progen_amd.tuning import enable_tuned_gemms
    enable_tuned_gemms()
    _, get_last_checkpoint, _ = get_checkpoint_fns(checkpoint_path)
    last_checkpoint = get_last_checkpoint()
    if last_checkpoint is None:
        raise SystemExit(f'no checkpoints found at {checkpoint_path}')

    params = numpy_to_tensors(last_checkpoint['params'])
    num_seqs = max(last_checkpoint['next_seq_index'], 0)

    model_kwargs = last_checkpoint['model_config']
    cfg = ProGenConfig.from_dict(model_kwargs)
    module = ProGenBase(cfg)
    module.load_state_dict({k: torch.as_tensor(v) for k, v in params.items()},
                           strict=False)
    device = torch.device('cuda') if torch.cuda.is_available() else torch.device('cpu')
    module = module.to(device)
    if device.type == 'cuda':
        module = module.to(torch.bfloat16)
        module.rotary_sin = module.rotary_sin.float()
        module.rotary_cos = module.rotary_cos.float()
    module.eval()

    seq_len = cfg.seq_len
    num_params = module.num_params()
    print(f'params: {num_params}')
    print(f'sequence length: {seq_len}')
    print(f'trained for {num_seqs} sequences')

    prime_tokens = encode_tokens(prime)
    prime_length = len(prime_tokens) + 1
    prime_tensor = torch.tensor(prime_tokens, dtype=torch.long)

    g = torch.Generator().manual_seed(seed)

    def fwd(seq):
        with torch.no_grad():
            return module(seq.to(device))[0].float().cpu()

    if cached:
        from progen_amd.decode import sample_cached
        sampled = sample_cached(module, prime_tensor, seq_len, top_k=25,
                                add_bos=True, generator=g, graph=graph)
    elif fast:
        sampled = sample_fast(fwd, prime_tensor, seq_len, top_k=25,
                              add_bos=True, generator=g,
                              window_size=cfg.window_size)
    else:
        sampled = sample(fwd, prime_tensor, seq_len, top_k=25, add_bos=True,
                         generator=g)
    sampled_str = decode_tokens(sampled[prime_length:].numpy())

    print('\n', prime, '\n', '*' * 40, '\n', sampled_str)


if __name__ == '__main__':
    main()
