"""Data-preparation CLI (reference parity: /root/reference/generate_data.py).

Two-stage pipeline: uniref50-style FASTA -> per-sequence gzip tmp files ->
GZIP TFRecord shards. The reference wraps the two stages in a Prefect
flow (generate_data.py:155-158); the DAG is linear, so here they are
plain functions run in order (Prefect is not in the offline image, and a
workflow engine adds nothing to a 2-node chain).

Semantics preserved:
  - length filter rlen <= max_seq_len, islice to num_samples
    (reference: generate_data.py:96-99);
  - annotation extraction: Tax=... regex from the description
    (generate_data.py:37), emitted as "[tax=X] # SEQ" with annotation/
    sequence order inverted with prob_invert_seq_annotation
    (generate_data.py:63-64), PLUS always the plain "# SEQ" string
    (generate_data.py:70-72) — i.e. an annotated record yields TWO
    training sequences;
  - random permutation, fraction_valid_data split, shards of
    num_sequences_per_file named {idx}.{count}.{type}.tfrecord.gz
    (generate_data.py:117-149);
  - gs:// output requires google-cloud-storage and raises offline.

The FASTA reader is a minimal streaming parser (the reference used
pyfaidx; an index is unnecessary for one sequential pass).
"""

import gzip
import re
from math import ceil
from pathlib import Path
from random import random

import click
import numpy as np

try:
    import tomllib
except ModuleNotFoundError:
    import tomli as tomllib

from progen_amd.data import with_tfrecord_writer
from progen_amd.utils import clear_directory_

TMP_DIR = Path('./.tmp')


def read_fasta(path):
    """Yield (description, sequence) streaming over a FASTA file."""
    desc, chunks = None, []
    with open(path) as f:
        for line in f:
            line = line.rstrip()
            if line.startswith('>'):
                if desc is not None:
                    yield desc, ''.join(chunks).upper()
                desc, chunks = line[1:], []
            elif line:
                chunks.append(line)
    if desc is not None:
        yield desc, ''.join(chunks).upper()


def get_annotations_from_description(config, description):
    """Tax=... regex (reference: generate_data.py:36-43)."""
    taxonomy_matches = re.findall(r'Tax=([a-zA-Z\s]*)\s[a-zA-Z\=]', description)
    annotations = {}
    if len(taxonomy_matches) > 0:
        annotations['tax'] = taxonomy_matches[0]
    return annotations


def row_to_sequence_strings(config, description, seq):
    """(reference: generate_data.py:45-74)"""
    sequences = []
    annotations = get_annotations_from_description(config, description)
    if len(annotations) > 0:
        keys = sorted(annotations.keys()) if config['sort_annotations'] \
            else list(np.random.permutation(list(annotations.keys())))
        annotation_str = ' '.join(f'[{k}={annotations[k]}]' for k in keys)
        pair = (annotation_str, seq)
        if random() <= config['prob_invert_seq_annotation']:
            pair = tuple(reversed(pair))
        sequences.append(' # '.join(pair).encode('utf-8'))
    sequences.append(f'# {seq}'.encode('utf-8'))
    return sequences


def fasta_to_tmp_files(config):
    """(reference: generate_data.py:87-105)"""
    clear_directory_(TMP_DIR)
    print('reading from fasta')
    count = 0
    written = 0
    for description, seq in read_fasta(config['read_from']):
        if len(seq) > config['max_seq_len']:
            continue
        if count >= config['num_samples']:
            break
        count += 1
        for s in row_to_sequence_strings(config, description, seq):
            with gzip.open(str(TMP_DIR / str(written)), 'wb') as f:
                f.write(s)
            written += 1
    print(f'wrote {written} tmp sequences from {count} fasta records')


def _write_shard(task):
    """One output TFRecord file (picklable worker for --workers > 1)."""
    out_path, index_list, tmp_names = task
    with with_tfrecord_writer(out_path) as write:
        for index in index_list:
            with gzip.open(tmp_names[index], 'rb') as f:
                write(f.read())
    return out_path


def files_to_tfrecords(config, workers: int = 1):
    """(reference: generate_data.py:107-153). Output shards are
    independent, so ``workers > 1`` writes them in parallel — the file
    set and every byte in it are identical to the serial run (each shard
    is still written sequentially by one process)."""
    filenames = sorted(TMP_DIR.glob('**/*'), key=lambda p: int(p.name))
    num_samples = len(filenames)
    num_valids = ceil(config['fraction_valid_data'] * num_samples)
    num_sequences_per_file = config['num_sequences_per_file']

    permuted = np.random.permutation(num_samples)
    valid_seqs, train_seqs = np.split(permuted, [num_valids])

    write_to = config['write_to']
    if write_to.startswith('gs://'):
        raise NotImplementedError(
            'gs:// output requires google-cloud-storage (offline image)')
    write_to_path = Path(write_to)
    clear_directory_(write_to_path)

    tasks = []
    for seq_type, seqs in (('train', train_seqs), ('valid', valid_seqs)):
        if seqs.shape[0] == 0:
            continue
        num_split = ceil(seqs.shape[0] / num_sequences_per_file)
        for file_index, indices in enumerate(np.array_split(seqs, num_split)):
            name = f'{file_index}.{len(indices)}.{seq_type}.tfrecord.gz'
            tasks.append((str(write_to_path / name), indices.tolist(),
                          filenames))
    if workers > 1 and len(tasks) > 1:
        from multiprocessing import Pool
        with Pool(min(workers, len(tasks))) as pool:
            for out in pool.imap_unordered(_write_shard, tasks):
                print(f'wrote {out}')
    else:
        for task in tasks:
            print(f'wrote {_write_shard(task)}')


@click.command()
@click.option('--data_dir', default='./configs/data')
@click.option('--name', default='default')
@click.option('--workers', default=1,
              help='parallel TFRecord shard writers (output is byte-'
                   'identical to the serial run)')
def main(data_dir, name, workers):
    config_path = Path(data_dir) / f'{name}.toml'
    assert config_path.exists(), f'config does not exist at {config_path}'
    config = tomllib.loads(config_path.read_text())
    fasta_to_tmp_files(config)
    files_to_tfrecords(config, workers=workers)


if __name__ == '__main__':
    main()
