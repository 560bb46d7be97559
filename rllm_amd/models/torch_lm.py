"""TinyTorchLM — a small pure-PyTorch causal LM for the CPU plumbing path
(BASELINE.json config 1: whole-pipeline training with no GPU, the role the
tinker backend plays in the reference). Fp32, eager torch ops only; NOT
part of the MI355X compute path."""

from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F


class TinyTorchLM(nn.Module):
    def __init__(self, vocab_size: int = 512, hidden: int = 64, layers: int = 2,
                 heads: int = 4, max_pos: int = 2048, seed: int = 0):
        super().__init__()
        torch.manual_seed(seed)
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.heads = heads
        self.embed = nn.Embedding(vocab_size, hidden)
        self.pos = nn.Embedding(max_pos, hidden)
        layer = nn.TransformerEncoderLayer(
            d_model=hidden, nhead=heads, dim_feedforward=hidden * 4,
            batch_first=True, norm_first=True, dropout=0.0)
        self.blocks = nn.TransformerEncoder(layer, num_layers=layers)
        self.norm = nn.LayerNorm(hidden)
        self.lm_head = nn.Linear(hidden, vocab_size, bias=False)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        """input_ids [B, S] -> logits [B, S, V] (causal)."""
        B, S = input_ids.shape
        x = self.embed(input_ids) + self.pos(torch.arange(S, device=input_ids.device))
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=input_ids.device), 1)
        x = self.blocks(x, mask=mask)
        return self.lm_head(self.norm(x))

    @torch.no_grad()
    def generate(self, prompt_ids: list[int], max_tokens: int, temperature: float = 1.0,
                 eos_token_id: int | None = None, generator: torch.Generator | None = None):
        """Returns (token_ids, logprobs, finish_reason)."""
        ids = list(prompt_ids)
        out_ids: list[int] = []
        logprobs: list[float] = []
        finish = "length"
        for _ in range(max_tokens):
            x = torch.tensor([ids], dtype=torch.long)
            logits = self.forward(x)[0, -1]
            if temperature <= 0:
                tok = int(logits.argmax())
                lp = float(F.log_softmax(logits, -1)[tok])
            else:
                z = logits / temperature
                probs = F.softmax(z, -1)
                tok = int(torch.multinomial(probs, 1, generator=generator))
                lp = float(F.log_softmax(z, -1)[tok])
            out_ids.append(tok)
            logprobs.append(lp)
            ids.append(tok)
            if eos_token_id is not None and tok == eos_token_id:
                finish = "stop"
                break
        return out_ids, logprobs, finish

    def logprobs_of(self, tokens: list[int], prompt_len: int) -> torch.Tensor:
        """Differentiable logprobs of tokens[prompt_len:] given the prefix."""
        ids = torch.tensor([tokens], dtype=torch.long)
        logits = self.forward(ids)[0]  # [S, V]
        logp = F.log_softmax(logits, -1)
        rows = torch.arange(prompt_len - 1, len(tokens) - 1)
        tgt = ids[0, rows + 1]
        return logp[rows, tgt]
