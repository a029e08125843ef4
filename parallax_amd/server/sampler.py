"""Batched sampler: temperature / top-k / top-p / min-p / penalties.

Parity with the reference's Sampler + SamplingBatchInfo (sampling/sampler.py);
fresh torch design. Penalty bookkeeping uses the head-stage request state."""

from __future__ import annotations

from typing import List, Optional

import torch

from .. import ops
from .request import InitialRequest


class Sampler:
    def __init__(self, device: torch.device, seed: Optional[int] = None):
        self.device = device
        self.generator = None
        # id -> token string table; enables json_schema constrained decoding
        # (set via Engine.set_grammar_vocab)
        self.grammar_vocab: Optional[List[str]] = None
        if seed is not None:
            self.generator = torch.Generator(
                device=device if device.type == "cuda" else "cpu"
            ).manual_seed(seed)

    def _apply_grammar_masks(
        self, logits: torch.Tensor, reqs: List[InitialRequest]
    ) -> torch.Tensor:
        """Mask logits rows of requests with a json_schema to the token set
        the schema FSM allows next (EOS once the schema is satisfied)."""
        if self.grammar_vocab is None:
            return logits
        rows = [
            i for i, r in enumerate(reqs) if r.sampling_params.json_schema
        ]
        if not rows:
            return logits
        from .constrained import GrammarMatcher

        logits = logits.clone()
        for i in rows:
            r = reqs[i]
            if r.grammar is None:
                eos = list(r.eos_token_ids) + list(
                    r.sampling_params.stop_token_ids
                )
                r.grammar = GrammarMatcher(
                    r.sampling_params.json_schema, self.grammar_vocab, eos
                )
            r.grammar.catch_up(r.output_token_ids)
            ids = r.grammar.allowed_ids()
            if not ids:
                continue  # no EOS registered: leave unconstrained
            idx = torch.tensor(ids, dtype=torch.long, device=logits.device)
            row = torch.full_like(logits[i], float("-inf"))
            row[idx] = logits[i, idx]
            logits[i] = row
        return logits

    def sample(self, logits: torch.Tensor, reqs: List[InitialRequest]) -> List[int]:
        """logits: [B, vocab] fp32, row i belongs to reqs[i]. Returns token ids."""
        return [t for t, _ in self.sample_with_logprobs(logits, reqs)]

    def sample_device(
        self,
        logits: torch.Tensor,
        reqs: List[InitialRequest],
        want_logprobs: Optional[bool] = None,
    ):
        """Device-resident sampling: returns (tokens int64 [B] on device,
        logprobs fp32 [B] on device or None). No host synchronization — the
        PP token broadcast sends these tensors directly over RCCL."""
        B = logits.shape[0]
        assert B == len(reqs)
        sp = [r.sampling_params for r in reqs]
        logits = self._apply_grammar_masks(logits, reqs)
        if any(s.logit_bias for s in sp):
            logits = logits.clone()
            V = logits.shape[-1]
            for i, s_i in enumerate(sp):
                if s_i.logit_bias:
                    for tid, b in s_i.logit_bias.items():
                        if 0 <= tid < V:
                            logits[i, tid] += b
        need_penalties = any(
            s.repetition_penalty != 1.0 or s.presence_penalty != 0.0
            or s.frequency_penalty != 0.0
            for s in sp
        )
        if need_penalties:
            logits = ops.apply_penalties(
                logits,
                [r.output_token_ids for r in reqs],
                [r.prompt_token_ids for r in reqs],
                torch.tensor([s.repetition_penalty for s in sp], device=logits.device),
                torch.tensor([s.presence_penalty for s in sp], device=logits.device),
                torch.tensor([s.frequency_penalty for s in sp], device=logits.device),
            )
        tokens = ops.sample_tokens(
            logits,
            [s.temperature for s in sp],
            [s.top_p for s in sp],
            [s.top_k for s in sp],
            [s.min_p for s in sp],
            generator=self.generator,
        )
        # per-request seed (OpenAI `seed`): rows re-sample with their own
        # generator keyed by (seed, position), so the draw is reproducible
        # regardless of batch composition
        for i, s_i in enumerate(sp):
            if s_i.seed is not None and s_i.temperature > 0.0:
                from ..ops import reference as ref_ops

                g = torch.Generator(device=logits.device)
                g.manual_seed((int(s_i.seed) * 1000003
                               + len(reqs[i].output_token_ids))
                              & 0x7FFFFFFFFFFFFFFF)
                tokens[i] = ref_ops.sample_tokens(
                    logits[i:i + 1], [s_i.temperature], [s_i.top_p],
                    [s_i.top_k], [s_i.min_p], g,
                )[0]
        if want_logprobs is None:
            want_logprobs = any(s.logprobs for s in sp)
        if not want_logprobs:
            return tokens, None
        lp = torch.log_softmax(logits.float(), dim=-1).gather(
            1, tokens.view(-1, 1).to(logits.device)
        ).squeeze(1)
        return tokens, lp

    def sample_with_logprobs(
        self, logits: torch.Tensor, reqs: List[InitialRequest]
    ) -> List[tuple]:
        """[(token_id, logprob-or-None)] — the logprob is log_softmax of the
        post-penalty, pre-temperature logits at the sampled token, computed
        only for requests with sampling_params.logprobs (reference wire fields
        token_prob / return_probs)."""
        sp = [r.sampling_params for r in reqs]
        tokens, lp = self.sample_device(logits, reqs)
        tok_list = tokens.tolist()
        if lp is None:
            return [(t, None) for t in tok_list]
        lp_list = lp.tolist()
        return [
            (t, lp_list[i] if sp[i].logprobs else None)
            for i, t in enumerate(tok_list)
        ]
