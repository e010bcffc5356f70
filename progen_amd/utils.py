"""Training math and sampling (reference: progen_transformer/utils.py).

get_loss_fn / cross_entropy / sample / select_top_k with the reference's
semantics preserved; batching is native (torch batch dim) instead of
vmap, and data parallelism is explicit RCCL (progen_amd/parallel/) instead
of the reference's pmap (utils.py:70).
"""

from __future__ import annotations

import os
import shutil
from typing import Callable, Optional

import torch

from .ops import functional as OF
from .ops import reference as R


# -- fs helpers (reference: utils.py:23-38) ---------------------------------

def exists(v) -> bool:
    return v is not None


def load_dotenv(path: str = ".env") -> None:
    """Minimal .env loader (parity with the reference's dotenv usage,
    reference: train.py:1-2, .env:1-2 — there it carries XLA flags; here
    it can carry ROCm/RCCL knobs like NCCL_MIN_NCHANNELS)."""
    try:
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#") or "=" not in line:
                    continue
                k, _, v = line.partition("=")
                os.environ.setdefault(k.strip(), v.strip())
    except FileNotFoundError:
        pass


def set_hardware_rng_(*_args, **_kwargs) -> None:
    """API-parity stub. The reference monkey-patches jax.random with a
    key-ignoring hardware RNG for TPU throughput (reference:
    utils.py:139-158). The PyTorch-ROCm stack's default generator is
    already the fast on-device Philox path, so there is nothing to
    patch; kept so reference-style call sites keep working."""
    return None


def confirm(question: str) -> bool:
    while True:
        resp = input(f"{question} (y/n) ").lower()
        if resp in ("y", "n"):
            return resp == "y"


def clear_directory_(path) -> None:
    shutil.rmtree(str(path), ignore_errors=True)
    path.mkdir(exist_ok=True, parents=True)


def silentremove(filename) -> None:
    try:
        os.remove(filename)
    except OSError:
        pass


# -- loss --------------------------------------------------------------------

masked_mean = R.masked_mean
cross_entropy = OF.cross_entropy


def compute_loss(model: torch.nn.Module, data: torch.Tensor) -> torch.Tensor:
    """One training loss on a (B, seq_len+1) int batch.

    ids = data[:, :-1], labels = data[:, 1:]  (reference: utils.py:63);
    per-sequence masked CE then batch mean (reference: utils.py:45-59,67).
    """
    ids, labels = data[:, :-1], data[:, 1:]
    logits = model(ids)
    return cross_entropy(logits, labels)


def get_loss_fn(model: torch.nn.Module, data_parallel: bool = False) -> Callable:
    """Returns loss_fn(data) -> (loss, None); gradients are produced by
    loss.backward() (torch autograd) rather than returned — the explicit
    analog of the reference's value_and_grad (utils.py:61-93). Under data
    parallelism the gradient all-reduce is performed by the DDP wrapper
    (progen_amd/parallel/ddp.py), not here."""

    def loss_fn(data: torch.Tensor) -> torch.Tensor:
        return compute_loss(model, data)

    return loss_fn


# -- sampling (reference: utils.py:97-135) -----------------------------------

select_top_k = R.select_top_k
gumbel_noise = R.gumbel_noise


@torch.no_grad()
def sample(
    fn: Callable[[torch.Tensor], torch.Tensor],
    prime: torch.Tensor,
    length: int,
    top_k: Optional[int] = None,
    add_bos: bool = False,
    generator: Optional[torch.Generator] = None,
    device=None,
    reference_add_bos_quirk: bool = False,
) -> torch.Tensor:
    """Gumbel-max top-k autoregressive decoding, reference semantics
    (reference: utils.py:106-135):

      - the sequence is padded to full ``length`` and every step runs a
        full-length forward (no KV cache) — parity path;
      - top-k uses a strict `>` mask vs the k-th value and sets excluded
        logits to 0, not -inf (utils.py:97-100);
      - everything after the second pad/EOS token is zeroed (utils.py:132-133).

    ``reference_add_bos_quirk``: reproduce the reference's add_bos
    off-by-one bit-for-bit (utils.py:110-116): the BOS pad shifts the
    prime right but start_pos is NOT advanced, so the loop starts at
    the position now holding the LAST PRIME TOKEN, and the one-hot
    `seq += one_hot * sampled` ADDS the first sample onto it. Off by
    default (the fixed semantics keep the prime intact); on request the
    reference behavior is reproduced exactly (each quirk
    deviation needs an explicit test + flag — tests/test_sample.py).

    fn: (n,) int64 tensor -> (n, V) logits (e.g. a closure over
    TransformedProGen.apply or the bare module).
    """
    prime = torch.as_tensor(prime, device=device).long().flatten()
    start_pos = prime.shape[-1]
    pad = (0, length - start_pos) if not add_bos else (1, length - start_pos - 1)
    seq = torch.nn.functional.pad(prime, pad)
    if add_bos and not reference_add_bos_quirk:
        start_pos += 1

    for curr_pos in range(start_pos, length):
        logits = fn(seq)
        logits = logits[curr_pos - 1].float()

        noise = gumbel_noise(logits.shape, generator=generator,
                             device=logits.device)

        if top_k is not None:
            mask, logits = select_top_k(logits, top_k)
            noise = noise * mask

        sampled = (logits + noise).argmax(dim=-1)
        # reference uses `seq += one_hot * sampled` (utils.py:128-129):
        # an ADD, which only differs from assignment at the quirk's
        # first position (everywhere else the slot is 0)
        seq[curr_pos] = seq[curr_pos] + sampled

    # zero after 2nd pad token (the 1st learned pad acts as EOS)
    remove_after_eos = (seq == 0).long().cumsum(dim=-1) > 1
    seq = seq * (~remove_after_eos).long()
    return seq


@torch.no_grad()
def sample_fast(
    fn: Callable[[torch.Tensor], torch.Tensor],
    prime: torch.Tensor,
    length: int,
    top_k: Optional[int] = None,
    add_bos: bool = False,
    generator: Optional[torch.Generator] = None,
    device=None,
    window_size: int = 256,
    eos_early_exit: bool = True,
) -> torch.Tensor:
    """Length-growing variant of ``sample``: each step forwards only the
    prefix padded up to the next window multiple instead of the full
    ``length`` (the model is causal — token shift, windowed attention and
    the tril-masked SGU — so logits at position p-1 are unaffected by the
    zero tail, and the emitted tokens are IDENTICAL to ``sample``'s; see
    tests/test_sample.py::test_sample_fast_matches_reference).

    O(sum n_i^2) instead of O(L * seq_len^2) — for short outputs this is
    several-fold faster; it also stops at the EOS (second pad) instead of
    emitting to full length.
    """
    prime = torch.as_tensor(prime, device=device).long().flatten()
    start_pos = prime.shape[-1]
    pad = (0, length - start_pos) if not add_bos else (1, length - start_pos - 1)
    seq = torch.nn.functional.pad(prime, pad)
    if add_bos:
        start_pos += 1

    pads_seen = int((seq[:start_pos] == 0).sum())  # BOS counts (utils.py:132)
    for curr_pos in range(start_pos, length):
        n_fwd = min(-(-curr_pos // window_size) * window_size, length)
        logits = fn(seq[:n_fwd])
        logits = logits[curr_pos - 1].float()

        noise = gumbel_noise(logits.shape, generator=generator,
                             device=logits.device)
        if top_k is not None:
            mask, logits = select_top_k(logits, top_k)
            noise = noise * mask

        sampled = (logits + noise).argmax(dim=-1)
        seq[curr_pos] = sampled
        if eos_early_exit and sampled.item() == 0:
            pads_seen += 1
            if pads_seen >= 2:
                break

    remove_after_eos = (seq == 0).long().cumsum(dim=-1) > 1
    seq = seq * (~remove_after_eos).long()
    return seq
