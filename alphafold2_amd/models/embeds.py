"""Pretrained-LM embedding front-ends with a pluggable embedder interface.

Capability parity: reference embeds.py:10-103 (ProtTran / MSA-Transformer
/ ESM-1b wrappers).  The heavy pretrained models are *frozen externals*
loaded lazily (torch.hub / transformers need network); for offline and
test use, every wrapper accepts an injected `embedder` object exposing
the same (model, batch_converter) surface, and a `FakeEmbedder` is
provided so the plumbing is exercisable without weights.
"""
import torch
import torch.nn.functional as F
from torch import nn

from ..constants import (
    ESM_EMBED_DIM, ESM_MODEL_PATH, MSA_EMBED_DIM, MSA_MODEL_PATH,
    PROTTRAN_EMBED_DIM,
)
from ..embedd_utils import get_esm_embedd, get_msa_embedd, get_prottran_embedd
from ..geometry.backend import exists


class FakeEmbedder:
    """Offline stand-in for an ESM-style (model, batch_converter) pair:
    deterministic random-projection embeddings of the token ids.  Lets
    the wrapper plumbing (projection, padding, splitting) run in tests
    and offline environments."""

    def __init__(self, embed_dim, vocab_size=24, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.table = torch.randn(vocab_size, embed_dim, generator=g)
        self.embed_dim = embed_dim

    def __call__(self, tokens, repr_layers=None, return_contacts=False):
        reps = self.table.to(tokens.device)[tokens.clamp(min=0) % self.table.shape[0]]
        layer = repr_layers[0] if repr_layers else 0
        return {"representations": {layer: reps}}

    def batch_converter(self, inputs):
        # inputs: list of (label, str) or list of lists of those
        def encode(pair_or_list):
            if isinstance(pair_or_list, tuple):
                _, s = pair_or_list
                return [0] + [min(ord(c) % 24, 23) for c in s]
            return [encode(p) for p in pair_or_list]
        toks = [encode(el) for el in inputs]
        t = torch.tensor(toks) if not isinstance(toks[0][0], list) \
            else torch.tensor(toks)
        return None, None, t


class ProtTranEmbedWrapper(nn.Module):
    def __init__(self, *, alphafold2, model=None, tokenizer=None):
        super().__init__()
        self.alphafold2 = alphafold2
        self.project_embed = nn.Linear(PROTTRAN_EMBED_DIM, alphafold2.dim)
        if model is None or tokenizer is None:
            from transformers import AutoModel, AutoTokenizer
            tokenizer = AutoTokenizer.from_pretrained(
                'Rostlab/prot_bert', do_lower_case=False)
            model = AutoModel.from_pretrained('Rostlab/prot_bert')
        self.tokenizer = tokenizer
        self.model = model

    def forward(self, seq, msa, msa_mask=None, **kwargs):
        device = seq.device
        num_msa = msa.shape[1]
        msa_flat = msa.reshape(-1, msa.shape[-1])

        seq_embed = get_prottran_embedd(seq, self.model, self.tokenizer,
                                        device=device)
        msa_embed = get_prottran_embedd(msa_flat, self.model, self.tokenizer,
                                        device=device)
        seq_embed = self.project_embed(seq_embed)
        msa_embed = self.project_embed(msa_embed)
        msa_embed = msa_embed.reshape(-1, num_msa, *msa_embed.shape[1:])

        return self.alphafold2(seq, msa, seq_embed=seq_embed,
                               msa_embed=msa_embed, msa_mask=msa_mask,
                               **kwargs)


class MSAEmbedWrapper(nn.Module):
    def __init__(self, *, alphafold2, embedder=None):
        super().__init__()
        self.alphafold2 = alphafold2
        if embedder is None:
            model, alphabet = torch.hub.load(*MSA_MODEL_PATH)
            batch_converter = alphabet.get_batch_converter()
        else:
            model, batch_converter = embedder, embedder.batch_converter
        self.model = model
        self.batch_converter = batch_converter
        self.project_embed = nn.Linear(MSA_EMBED_DIM, alphafold2.dim) \
            if MSA_EMBED_DIM != alphafold2.dim else nn.Identity()

    def forward(self, seq, msa, msa_mask=None, **kwargs):
        assert seq.shape[-1] == msa.shape[-1], \
            'sequence and msa must have the same length for MSA-transformer embeddings'
        model, batch_converter = self.model, self.batch_converter
        device = seq.device

        seq_and_msa = torch.cat((seq.unsqueeze(1), msa), dim=1)

        if exists(msa_mask):
            # fully-padded MSA rows must not join the row-tied attention:
            # embed each batch element with only its real rows, re-pad after
            num_msa = msa_mask.any(dim=-1).sum(dim=-1).tolist()
            num_rows = seq_and_msa.shape[1]
            embeds = []
            for num, batch_el in zip(num_msa, seq_and_msa.unbind(dim=0)):
                batch_el = batch_el[None, :num]
                embed = get_msa_embedd(batch_el, model, batch_converter,
                                       device=device)
                embed = F.pad(embed, (0, 0, 0, 0, 0, num_rows - num), value=0.)
                embeds.append(embed)
            embeds = torch.cat(embeds, dim=0)
        else:
            embeds = get_msa_embedd(seq_and_msa, model, batch_converter,
                                    device=device)

        embeds = self.project_embed(embeds)
        seq_embed, msa_embed = embeds[:, 0], embeds[:, 1:]

        return self.alphafold2(seq, msa, seq_embed=seq_embed,
                               msa_embed=msa_embed, msa_mask=msa_mask,
                               **kwargs)


class ESMEmbedWrapper(nn.Module):
    def __init__(self, *, alphafold2, embedder=None):
        super().__init__()
        self.alphafold2 = alphafold2
        if embedder is None:
            model, alphabet = torch.hub.load(*ESM_MODEL_PATH)
            batch_converter = alphabet.get_batch_converter()
        else:
            model, batch_converter = embedder, embedder.batch_converter
        self.model = model
        self.batch_converter = batch_converter
        self.project_embed = nn.Linear(ESM_EMBED_DIM, alphafold2.dim) \
            if ESM_EMBED_DIM != alphafold2.dim else nn.Identity()

    def forward(self, seq, msa=None, **kwargs):
        model, batch_converter = self.model, self.batch_converter

        seq_embeds = get_esm_embedd(seq, model, batch_converter)
        seq_embeds = self.project_embed(seq_embeds)[:, 0]

        if msa is not None:
            num_msa = msa.shape[1]
            flat_msa = msa.reshape(-1, msa.shape[-1])
            msa_embeds = get_esm_embedd(flat_msa, model, batch_converter)
            msa_embeds = msa_embeds.reshape(
                -1, num_msa, *msa_embeds.shape[2:])
            msa_embeds = self.project_embed(msa_embeds)
        else:
            msa_embeds = None

        return self.alphafold2(seq, msa, seq_embed=seq_embeds,
                               msa_embed=msa_embeds, **kwargs)
