"""BERT-style masked-language-model objective over MSA tokens.

Capability parity: reference mlm.py:11-92.  `noise()` corrupts a
(b, m, n) MSA during training (15% of maskable positions: mask token,
keep-same, or random-replace); `forward()` computes the CE loss over the
corrupted positions from the trunk's MSA embeddings.
"""
import math

import torch
import torch.nn.functional as F
from torch import nn

from . import constants


def get_mask_subset_with_prob(mask, prob):
    """Uniformly sample a subset of the allowed positions per row.

    Row i receives ceil(prob * allowed_i) picks (capped by
    ceil(prob * seq_len), the reference budget).  Rank-based selection:
    draw one uniform score per position, push disallowed positions to
    the bottom, and keep every position whose within-row rank falls
    inside the row's quota — no scatter round-trip, fully batched.
    """
    seq_len = mask.shape[-1]
    budget = math.ceil(prob * seq_len)
    allowed = mask.sum(dim=-1, keepdim=True)
    quota = torch.clamp((allowed * prob).ceil().long(), max=budget)

    scores = torch.rand(mask.shape, device=mask.device)
    scores = scores.masked_fill(~mask, -1.0)
    # rank[r, j] = how many positions in row r score higher than j
    order = scores.argsort(dim=-1, descending=True)
    rank = torch.empty_like(order)
    rank.scatter_(-1, order,
                  torch.arange(seq_len, device=mask.device)
                  .expand_as(order))
    return (rank < quota) & mask


class MLM(nn.Module):
    def __init__(self, dim, num_tokens, mask_id, mask_prob=0.15,
                 random_replace_token_prob=0.1, keep_token_same_prob=0.1,
                 exclude_token_ids=(0,)):
        super().__init__()
        self.to_logits = nn.Linear(dim, num_tokens)
        self.mask_id = mask_id
        self.mask_prob = mask_prob
        self.exclude_token_ids = exclude_token_ids
        self.keep_token_same_prob = keep_token_same_prob
        self.random_replace_token_prob = random_replace_token_prob

    def noise(self, seq, mask):
        """Corrupt (b, m, n) MSA tokens; returns (noised, replaced_mask)."""
        num_msa = seq.shape[1]
        seq = seq.reshape(-1, seq.shape[-1])
        mask = mask.reshape(-1, mask.shape[-1])

        excluded_tokens_mask = mask
        for token_id in self.exclude_token_ids:
            excluded_tokens_mask = excluded_tokens_mask & (seq != token_id)

        mlm_mask = get_mask_subset_with_prob(excluded_tokens_mask, self.mask_prob)

        # of the selected positions: (1 - keep_same) get the mask token
        seq = seq.masked_fill(mlm_mask, self.mask_id)

        random_replace_mask = get_mask_subset_with_prob(
            mlm_mask, (1 - self.keep_token_same_prob) * self.random_replace_token_prob)
        random_tokens = torch.randint(1, constants.NUM_AMINO_ACIDS, seq.shape,
                                      device=seq.device)
        for token_id in self.exclude_token_ids:
            random_replace_mask = random_replace_mask & (random_tokens != token_id)

        noised_seq = torch.where(random_replace_mask, random_tokens, seq)
        noised_seq = noised_seq.reshape(-1, num_msa, noised_seq.shape[-1])
        mlm_mask = mlm_mask.reshape(-1, num_msa, mlm_mask.shape[-1])
        return noised_seq, mlm_mask

    def forward(self, seq_embed, original_seq, mask):
        logits = self.to_logits(seq_embed)
        # static-shape masked mean (no boolean gather: a gather would
        # force a device sync and break hipGraph step capture)
        V = logits.shape[-1]
        per_tok = F.cross_entropy(
            logits.reshape(-1, V).float(), original_seq.reshape(-1),
            reduction='none')
        fmask = mask.reshape(-1).to(per_tok.dtype)
        return (per_tok * fmask).sum() / fmask.sum().clamp(min=1)
