"""Training CLI (reference parity: /root/reference/train.py).

Flag surface mirrors the reference (train.py:36-57). Differences, by
design (MI355X-native):
  - data parallelism: launch with torchrun (one process per GPU, RCCL
    over xGMI); the --data_parallel flag is accepted and implied by
    WORLD_SIZE>1 (the reference used jax.pmap, train.py:42 / utils.py:70);
  - --mixed_precision selects bf16 compute (MFMA-native) instead of the
    reference's jmp fp16 policy;
  - --grad_accum_mode {sum,apply_every} exposes the reference's
    apply_every optimizer quirk (train.py:117-121) behind a flag,
    defaulting to standard summed accumulation;
  - wandb is optional (not installed in the offline image): --wandb_off
    or a missing wandb package degrade to stdout logging.
"""

import os
import time
from pathlib import Path

import click
import numpy as np
import torch

try:
    import tomllib
except ModuleNotFoundError:
    import tomli as tomllib

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.checkpoint import get_checkpoint_fns, numpy_to_tensors, tensors_to_numpy
from progen_amd.data import decode_tokens, iterator_from_tfrecords_folder
from progen_amd.optim import ProGenAdamW
from progen_amd.parallel import DistributedTrainer, init_distributed
from progen_amd.utils import (compute_loss, confirm, exists, load_dotenv,
                              sample_fast, set_hardware_rng_)

load_dotenv()        # reference: train.py:1-2
set_hardware_rng_()  # reference: train.py:32 (no-op on ROCm, see utils)


def _wandb(wandb_off):
    if wandb_off:
        return None
    try:
        import wandb  # type: ignore
        return wandb
    except ImportError:
        return None


@click.command()
@click.option('--seed', default=42)
@click.option('--batch_size', default=4)
@click.option('--grad_accum_every', default=4)
@click.option('--learning_rate', default=2e-4)
@click.option('--weight_decay', default=1e-3)
@click.option('--data_parallel', default=False, is_flag=True)
@click.option('--max_grad_norm', default=0.5)
@click.option('--validate_every', default=100)
@click.option('--sample_every', default=500)
@click.option('--checkpoint_every', default=1000)
@click.option('--checkpoint_path', default='./ckpts')
@click.option('--checkpoint_keep_n', default=500)
@click.option('--config_path', default='./configs/model')
@click.option('--model_name', default='default')
@click.option('--prime_length', default=25)
@click.option('--seq_len', default=1024)
@click.option('--mixed_precision', default=False, is_flag=True)
@click.option('--grad_accum_mode', default='sum',
              type=click.Choice(['sum', 'apply_every']))
@click.option('--data_path', default='./train_data')
@click.option('--wandb_off', default=False, is_flag=True)
@click.option('--wandb_project_name', default='progen-training')
@click.option('--new', default=False, is_flag=True)
@click.option('--max_steps', default=0, help='stop after N effective batches (0 = full epoch)')
@click.option('--graph/--no-graph', default=False,
              help='hipGraph-capture the training step (single-GPU, '
                   'grad_accum_every=1, pure replay). Checkpoints, validation '
                   'AND sampling still run in-loop: the graph is dropped and '
                   're-captured around them. KNOWN OPEN ISSUE: multi-step '
                   'replay can corrupt gradients on this ROCm stack (the '
                   'AdamW non-finite skip guard contains it, and the loop '
                   'aborts on a NaN loss) — eager is the safe default; see '
                   'profiles/r02_graphed_nan_investigation.md')
@click.option('--yes', default=False, is_flag=True, help='skip the --new confirmation prompt')
def main(seed, batch_size, grad_accum_every, learning_rate, weight_decay,
         data_parallel, max_grad_norm, validate_every, sample_every,
         checkpoint_every, checkpoint_path, checkpoint_keep_n, config_path,
         model_name, prime_length, seq_len, mixed_precision, grad_accum_mode,
         data_path, wandb_off, wandb_project_name, new, max_steps, yes, graph):
    from progen_amd.tuning import enable_tuned_gemms
    enable_tuned_gemms()
    local_rank = init_distributed()
    world = int(os.environ.get('WORLD_SIZE', '1'))
    rank = int(os.environ.get('RANK', '0'))
    is_main = rank == 0
    device = torch.device('cuda', local_rank) if torch.cuda.is_available() \
        else torch.device('cpu')

    torch.manual_seed(seed)  # identical model init on every rank

    # checkpoints (reference: train.py:83-92)
    reset_checkpoint, get_last_checkpoint, save_checkpoint = \
        get_checkpoint_fns(checkpoint_path)
    if new and is_main:
        if not yes and not confirm('are you sure you want to clear all your '
                                   'checkpoints and restart training?'):
            raise SystemExit
        reset_checkpoint()

    last_checkpoint = get_last_checkpoint()

    # model config: checkpoint beats toml (reference: train.py:95-100)
    if not exists(last_checkpoint):
        cfg_file = Path(config_path) / f'{model_name}.toml'
        assert cfg_file.exists(), \
            f'path to your model config {cfg_file} does not exist'
        model_kwargs = tomllib.loads(cfg_file.read_text())
    else:
        model_kwargs = last_checkpoint['model_config']

    cfg = ProGenConfig.from_dict(model_kwargs)
    seq_len = cfg.seq_len

    module = ProGenBase(cfg)
    dtype = torch.bfloat16 if (mixed_precision and device.type == 'cuda') \
        else torch.float32
    module = module.to(device=device, dtype=dtype)
    # rotary tables stay fp32 for accuracy
    module.rotary_sin = module.rotary_sin.float()
    module.rotary_cos = module.rotary_cos.float()

    opt_kwargs = dict(lr=learning_rate, weight_decay=weight_decay,
                      max_grad_norm=max_grad_norm,
                      accum_mode=grad_accum_mode,
                      grad_accum_every=grad_accum_every)
    if world > 1 and os.environ.get('PROGEN_ZERO1') == '1':
        from progen_amd.parallel.zero1 import Zero1AdamW
        optim = Zero1AdamW(module, **opt_kwargs)
    else:
        optim = ProGenAdamW(module, **opt_kwargs)
    ddp = DistributedTrainer(optim.space)
    if world > 1:  # DP replicas must start bitwise-identical
        import torch.distributed as dist
        dist.broadcast(optim.space.flat, src=0)
        optim.resync_master()

    start_seq_index = 0
    if exists(last_checkpoint):
        module.load_state_dict({
            k: torch.as_tensor(v).to(device=device)
            for k, v in numpy_to_tensors(last_checkpoint['params']).items()
        }, strict=False)  # copies in place -> flat param buffer updated
        optim.resync_master()
        if exists(last_checkpoint.get('optim_state')):
            optim.load_state_dict(numpy_to_tensors(last_checkpoint['optim_state']))
        start_seq_index = last_checkpoint['next_seq_index']

    num_params = module.num_params()

    run_id = last_checkpoint.get('run_id') if exists(last_checkpoint) else None
    wandb = _wandb(wandb_off) if is_main else None
    if wandb is not None:
        kwargs = {'id': run_id, 'resume': 'allow'} if exists(run_id) else {}
        wandb.init(project=wandb_project_name, **kwargs)
        wandb.config.num_params = num_params
        run_id = wandb.run.id

    # data (reference: train.py:153-172); each rank reads the same global
    # batch and slices its shard (pmap-reshape parity, utils.py:89)
    global_batch = batch_size * world
    total_train_seqs, get_train_dataset = \
        iterator_from_tfrecords_folder(data_path, data_type='train')
    total_valid_seqs, get_valid_dataset = \
        iterator_from_tfrecords_folder(data_path, data_type='valid')
    assert total_train_seqs > 0, 'no protein sequences found for training'
    assert total_valid_seqs > 0, 'no protein sequences found for validation'

    train_dataset = get_train_dataset(seq_len=seq_len, batch_size=global_batch,
                                      skip=start_seq_index)
    valid_dataset = get_valid_dataset(seq_len=seq_len, batch_size=global_batch,
                                      loop=True)

    # hipGraph-captured step (progen_amd/runtime.py): replays the whole
    # fwd+bwd(+optimizer) sequence as one graph when shapes are static
    graph_ok = graph and device.type == 'cuda' and grad_accum_every == 1 \
        and grad_accum_mode == 'sum' and world == 1
    if graph and not graph_ok and is_main:
        # say WHY, loudly: --graph silently running eager hides a 15%+
        # perf loss (found the hard way: --grad_accum_every defaults to 4)
        why = []
        if device.type != 'cuda':
            why.append('no GPU')
        if grad_accum_every != 1:
            why.append(f'--grad_accum_every {grad_accum_every} != 1')
        if grad_accum_mode != 'sum':
            why.append(f"--grad_accum_mode {grad_accum_mode} != 'sum'")
        if world != 1:
            why.append(f'world {world} > 1 (graphed DP lives in bench.py '
                       f'behind PROGEN_GRAPH_DP=1)')
        print(f"--graph disabled ({', '.join(why)}); running eager")

    def make_graphed():
        from progen_amd.runtime import GraphedTrainStep
        try:
            g = GraphedTrainStep(module, optim, ddp, batch_size,
                                 seq_len, device)
            if is_main:
                print('hipGraph training step captured')
            return g
        except Exception as e:  # noqa: BLE001
            if is_main:
                import traceback
                traceback.print_exc()
                print(f'hipGraph capture failed ({e}); running eager')
            return None

    graphed = make_graphed() if graph_ok else None

    if is_main:
        print(f'params: {num_params}')
        print(f'sequence length: {seq_len}')
        print(f'num sequences: {total_train_seqs}')
        print(f'starting from sequence {start_seq_index}')

    def my_shard(batch_np, drop_ragged=True):
        t = torch.from_numpy(batch_np.astype(np.int64))
        if drop_ragged and world > 1 and t.shape[0] < global_batch:
            # tf.data drop_remainder parity: duplicating rows to fill the
            # ragged tail would double-count them in the averaged
            # gradient — end the epoch instead
            raise StopIteration
        lo = rank * batch_size
        shard = t[lo:lo + batch_size]
        if shard.shape[0] == 0:  # ragged validation batch: reuse row 0
            shard = t[:1]
        return shard.to(device)

    effective_batch_size = global_batch * grad_accum_every
    seq_index_ranges = range(start_seq_index, total_train_seqs,
                             effective_batch_size)

    step_times = []
    for i, seq_index in enumerate(seq_index_ranges):
        if max_steps and i >= max_steps:
            break
        t0 = time.perf_counter()
        loss = None
        for micro in range(grad_accum_every):
            try:
                data = my_shard(next(train_dataset))
            except StopIteration:
                break
            if graphed is not None:
                loss = graphed.run(data)
                continue
            # apply_every advances Adam moments EVERY micro-batch, so the
            # grads it consumes must already be all-reduced each micro —
            # the reference pmap reduces per micro-batch too (ADVICE r1:
            # no_sync here would diverge the DP replicas)
            last_micro = micro == grad_accum_every - 1
            if last_micro or grad_accum_mode == 'apply_every':
                loss = compute_loss(module, data)
                loss.backward()
                ddp.finish_backward()
            else:
                with ddp.no_sync():
                    loss = compute_loss(module, data)
                    loss.backward()
            optim.micro_step()
        if loss is None:
            break
        step_times.append(time.perf_counter() - t0)

        # NOTE: like the reference (train.py:192), the logged loss is the
        # last micro-batch's
        loss_val = ddp.all_reduce_scalar(loss).item()
        if graphed is not None and loss_val != loss_val:
            # tripwire for the open graphed-replay NaN issue
            # (profiles/r02_graphed_nan_investigation.md): the eager path
            # is trajectory-stable on the same data — fail loudly instead
            # of silently training on NaNs
            raise RuntimeError(
                f'loss went NaN at step {i} under --graph; re-run without '
                f'--graph (eager path). Known open issue: see '
                f'profiles/r02_graphed_nan_investigation.md')
        if is_main:
            toks_per_sec = effective_batch_size * seq_len / step_times[-1]
            print(f'loss: {loss_val}')
            if i % 10 == 0:
                print(f'tokens/sec (whole job): {toks_per_sec:.0f}')
            if wandb is not None:
                wandb.log({'loss': loss_val, 'tokens_per_sec': toks_per_sec})

        # checkpointing / validation / sampling are eager work, and a
        # live hipGraph forbids eager kernels between replays (pure-replay
        # rule) — so the graph is dropped once, all due eager sections
        # run, and ONE re-capture follows (--graph used to
        # skip checkpoints entirely, losing all progress on interruption)
        do_ckpt = i % checkpoint_every == 0 and (graphed is None or i > 0)
        do_valid = i % validate_every == 0 and (graphed is None or i > 0)
        do_sample = i % sample_every == 0 and (graphed is None or i > 0)
        was_graphed = graphed is not None
        if (do_ckpt or do_valid or do_sample) and was_graphed:
            torch.cuda.synchronize()
            graphed = None  # free replay state before eager/D2H work
        if do_ckpt:
            # ZeRO-1's state_dict all-gathers shards: collective call
            optim_sd = optim.state_dict() \
                if (is_main or getattr(optim, 'state_dict_is_collective',
                                       False)) else None
            if is_main:
                package = {
                    'next_seq_index': seq_index + effective_batch_size,
                    'params': tensors_to_numpy(
                        {k: v for k, v in module.state_dict().items()}),
                    'optim_state': tensors_to_numpy(optim_sd),
                    'model_config': model_kwargs,
                    'run_id': run_id,
                }
                save_checkpoint(package, checkpoint_keep_n)
                print(f"checkpoint to start at sequence index of "
                      f"{package['next_seq_index']}")

        if do_valid:
            valid_data = my_shard(next(valid_dataset))
            with torch.no_grad():
                vloss = compute_loss(module, valid_data)
            vloss_val = ddp.all_reduce_scalar(vloss).item()
            if is_main:
                print(f'valid_loss: {vloss_val}')
                if wandb is not None:
                    wandb.log({'valid_loss': vloss_val})

        if do_sample and is_main:
            valid_data = my_shard(next(valid_dataset))[0]
            prime = valid_data[:prime_length]
            prime_str = decode_tokens(prime.cpu().numpy())

            def fwd(seq):
                with torch.no_grad():
                    return module(seq.to(device))[0].float().cpu()

            # identical tokens to the reference sampler, O(prefix) forwards
            sampled = sample_fast(fwd, prime.cpu(), seq_len, top_k=25,
                                  window_size=cfg.window_size)
            sampled_str = decode_tokens(sampled[prime_length:].numpy())
            print(prime_str, '\n', '*' * 40, '\n', sampled_str)
            if wandb is not None:
                # same markup the reference renders via jinja2
                # (reference: train.py:28,222)
                html = (f"<i>{prime_str}</i><br/><br/>"
                        f'<div style="overflow-wrap: break-word;">'
                        f"{sampled_str}</div>")
                wandb.log({'samples': wandb.Html(html)})

        if was_graphed and graphed is None and graph_ok:
            graphed = make_graphed()

    # synchronized teardown: without the barrier a fast rank can exit and
    # close its gloo/RCCL connections while a slower rank is still inside
    # process-group destruction, which SIGABRTs the slower rank
    if world > 1:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
