"""Pretrained-LM embedding extraction helpers.

Capability parity: reference utils.py:257-390.  These wrap *frozen*
third-party language models (ESM / MSA-Transformer / ProtTrans); the
models themselves are external and loaded by the caller — this module
owns the id <-> string conversion and the representation extraction.
"""
import re

import torch

from .vocab import VOCAB
from .geometry.backend import exists, expand_dims_to


def ids_to_embed_input(x):
    """Nested lists of residue ids -> (label, string) pairs for the ESM
    batch converters."""
    assert isinstance(x, list), 'input must be a list'
    id2aa = VOCAB._int2char
    out = []
    for el in x:
        if isinstance(el, list):
            out.append(ids_to_embed_input(el))
        elif isinstance(el, int):
            out.append(id2aa[el])
        else:
            raise TypeError('type must be either list or character')
    if all(isinstance(c, str) for c in out):
        return (None, ''.join(out))
    return out


def ids_to_prottran_input(x):
    """Nested lists of residue ids -> space-separated strings for
    ProtTrans tokenizers (U/Z/O/B folded to X)."""
    assert isinstance(x, list), 'input must be a list'
    id2aa = VOCAB._int2char
    out = []
    for ids in x:
        chars = ' '.join(id2aa[i] for i in ids)
        chars = re.sub(r"[UZOB]", "X", chars)
        out.append(chars)
    return out


def get_prottran_embedd(seq, model, tokenizer, device=None):
    from transformers import pipeline
    fe = pipeline('feature-extraction', model=model, tokenizer=tokenizer,
                  device=(-1 if not exists(device) else device.index))
    max_seq_len = seq.shape[1]
    embedd_inputs = ids_to_prottran_input(seq.cpu().tolist())
    embedding = fe(embedd_inputs)
    embedding = torch.tensor(embedding, device=device)
    return embedding[:, 1:(max_seq_len + 1)]


def get_msa_embedd(msa, embedd_model, batch_converter, device=None):
    """MSA-Transformer layer-12 representations: (b, n_seqs, L, 768)."""
    REPR_LAYER_NUM = 12
    device = msa.device if device is None else device
    max_seq_len = msa.shape[-1]
    embedd_inputs = ids_to_embed_input(msa.cpu().tolist())
    _, _, msa_batch_tokens = batch_converter(embedd_inputs)
    with torch.no_grad():
        results = embedd_model(msa_batch_tokens.to(device),
                               repr_layers=[REPR_LAYER_NUM],
                               return_contacts=False)
    # position 0 is the start token
    return results["representations"][REPR_LAYER_NUM][..., 1:max_seq_len + 1, :]


def get_esm_embedd(seq, embedd_model, batch_converter, msa_data=None):
    """ESM-1b layer-33 representations: (b, 1, L, 1280)."""
    device = seq.device
    REPR_LAYER_NUM = 33
    max_seq_len = seq.shape[-1]
    embedd_inputs = ids_to_embed_input(seq.cpu().tolist())
    _, _, batch_tokens = batch_converter(embedd_inputs)
    with torch.no_grad():
        results = embedd_model(batch_tokens.to(device),
                               repr_layers=[REPR_LAYER_NUM],
                               return_contacts=False)
    return results["representations"][REPR_LAYER_NUM][..., 1:max_seq_len + 1, :].unsqueeze(dim=1)


def get_t5_embedd(seq, tokenizer, encoder, msa_data=None, device=None):
    """ProtT5-XL-U50 last-hidden-state representations: (b, 1, L, 1024)."""
    device = seq.device if device is None else device
    embedd_inputs = ids_to_prottran_input(seq.cpu().tolist())
    shift_left, shift_right = 0, -1
    ids = tokenizer.batch_encode_plus(embedd_inputs, add_special_tokens=True,
                                      padding=True, return_tensors="pt")
    with torch.no_grad():
        embedding = encoder(input_ids=ids['input_ids'].to(device),
                            attention_mask=ids['attention_mask'].to(device))
    token_reps = embedding.last_hidden_state[:, shift_left:shift_right].to(device)
    token_reps = expand_dims_to(token_reps, 4 - len(token_reps.shape))
    return token_reps.float()


def get_all_protein_ids(dataloader, verbose=False):
    """Collect the protein-entry ids from a sidechainnet-style
    dataloader (fixed version of the reference helper, which read an
    undefined global and never advanced its iterator correctly)."""
    ids = set()
    for batch in dataloader:
        for pid in getattr(batch, 'pids', []):
            max_len_10 = len(pid) < 10
            fragments = [len(x) <= 4 for x in pid.split("_")]
            if max_len_10 and all(fragments):
                ids.add(pid)
            elif verbose:
                print("skip:", pid)
    return ids
