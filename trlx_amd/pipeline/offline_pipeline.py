"""Offline pipelines: dialogue tokenization, prompt dataset, ILQL/SFT stores.

Parity target: reference trlx/pipeline/offline_pipeline.py — DialogMessage
(22), tokenize_dialogue (38-87: interleaved prompt/output tokenization with
left/right truncation and BOS/EOS fix-ups), DialogStore (90-115),
PromptPipeline (118-188), ilql collates + ILQL(Seq2Seq)RolloutStorage
(191-289, drop_last under distributed).
"""

from dataclasses import dataclass
from typing import Any, Dict, Iterable, List, Tuple, Union

import torch
import torch.distributed as dist
from torch.nn.utils.rnn import pad_sequence
from torch.utils.data import DataLoader

from ..data.ilql_types import (
    ILQLBatch,
    ILQLElement,
    ILQLSeq2SeqBatch,
    ILQLSeq2SeqElement,
)
from . import BasePipeline, BaseRolloutStore, dp_sampler, register_datapipeline


@dataclass
class DialogMessage:
    """One message of a dialogue; ``is_output`` marks model outputs."""

    is_output: bool
    tokens: Tuple[int, ...]


def tokenize_dialogue(dialogue: Union[str, Iterable[str]], tokenizer, max_length=2048) -> List[DialogMessage]:
    """Tokenize an interleaved (prompt_1, output_1, prompt_2, ...) dialogue.

    Semantics match reference offline_pipeline.py:38-87: a bare string becomes
    (BOS, string); EOS is appended to the final output; truncation respects
    ``tokenizer.truncation_side`` (left truncation flips, truncates, flips
    back); empty messages are dropped; if the first remaining message is an
    output, a BOS message is prepended (evicting one token if at max_length).
    """
    if isinstance(dialogue, str):
        bos_token = tokenizer.bos_token or tokenizer.eos_token
        dialogue = [bos_token, dialogue]
    elif isinstance(dialogue, Iterable):
        dialogue = list(dialogue)
        if len(dialogue) % 2 != 0:
            raise ValueError("Dialogue must have an even number of phrases, alternating prompt and output")

    if not dialogue[-1].endswith(tokenizer.eos_token):
        dialogue[-1] = dialogue[-1] + tokenizer.eos_token

    tokenized = [
        DialogMessage(
            is_output=i % 2 == 1,
            tokens=tuple(tokenizer(dialogue[i], add_special_tokens=False).input_ids),
        )
        for i in range(len(dialogue))
    ]

    left = tokenizer.truncation_side == "left"
    if left:
        tokenized = [DialogMessage(m.is_output, m.tokens[::-1]) for m in tokenized[::-1]]

    lengths = [len(t.tokens) for t in tokenized]
    cumsum = [sum(lengths[:i]) for i in range(len(lengths))]
    truncated = [
        DialogMessage(t.is_output, t.tokens[: max(max_length - cl, 0)]) for t, cl in zip(tokenized, cumsum)
    ]

    if left:
        truncated = [DialogMessage(m.is_output, m.tokens[::-1]) for m in truncated[::-1]]

    out = [t for t in truncated if len(t.tokens) > 0]

    if out[0].is_output:
        if sum(len(m.tokens) for m in out) == max_length:
            if left:
                out[0] = DialogMessage(out[0].is_output, out[0].tokens[1:])
            else:
                out[-1] = DialogMessage(out[-1].is_output, out[-1].tokens[:-1])
        out.insert(0, DialogMessage(False, (tokenizer.bos_token_id,)))
    return out


class DialogStore(BaseRolloutStore):
    """SFT store: dialogue tokens with -100-masked prompt labels
    (reference offline_pipeline.py:90-115)."""

    def __init__(self, dialogs: List[List[DialogMessage]], tokenizer):
        super().__init__()
        self.tokenizer = tokenizer
        attention_masks = [torch.ones(sum(len(m.tokens) for m in d), dtype=torch.long) for d in dialogs]
        input_ids = [torch.tensor([t for m in d for t in m.tokens], dtype=torch.long) for d in dialogs]
        labels = [
            torch.tensor([t if m.is_output else -100 for m in d for t in m.tokens], dtype=torch.long)
            for d in dialogs
        ]
        self.history = [
            dict(input_ids=i, attention_mask=a, labels=l)
            for i, a, l in zip(input_ids, attention_masks, labels)
        ]

    def create_loader(self, batch_size: int, shuffle=False) -> DataLoader:
        pad_id = self.tokenizer.pad_token_id or 0

        def collate_fn(elems: Iterable[dict]):
            return dict(
                input_ids=pad_sequence([e["input_ids"] for e in elems], batch_first=True, padding_value=pad_id),
                attention_mask=pad_sequence(
                    [e["attention_mask"] for e in elems], batch_first=True, padding_value=0
                ),
                labels=pad_sequence([e["labels"] for e in elems], batch_first=True, padding_value=-100),
            )

        sampler = dp_sampler(self, shuffle=shuffle)
        return DataLoader(self, batch_size=batch_size, collate_fn=collate_fn,
                          shuffle=shuffle and sampler is None, sampler=sampler)


@register_datapipeline
class PromptPipeline(BasePipeline):
    """Tokenized prompts + metadata passthrough for generation
    (reference offline_pipeline.py:118-188)."""

    def __init__(self, prompts: Union[List[Dict[str, Any]], List[str]], max_prompt_length: int,
                 tokenizer, add_special_tokens: bool = False):
        super().__init__()
        if prompts and isinstance(prompts[0], dict):
            metadata = [dict(x) for x in prompts]
            prompts = [x.pop("prompt") for x in metadata]
        else:
            metadata = [{}] * len(prompts)

        model_inputs = tokenizer(
            list(prompts), truncation=True, padding=False, max_length=max_prompt_length,
            add_special_tokens=add_special_tokens,
        )
        self.tokenizer = tokenizer
        self.prompts = [
            {"input_ids": tokens, "attention_mask": mask, **md}
            for tokens, mask, md in zip(model_inputs["input_ids"], model_inputs["attention_mask"], metadata)
        ]

    def __getitem__(self, ix: int):
        return self.prompts[ix]

    def __len__(self) -> int:
        return len(self.prompts)

    def create_loader(self, batch_size: int, shuffle=False, sampler=None, drop_last=False) -> DataLoader:
        tokenizer = self.tokenizer

        def collate_fn(xs):
            # left-pad (tokenizer.padding_side is "left" for causal RL)
            ids = [torch.tensor(x["input_ids"], dtype=torch.long) for x in xs]
            pad_id = tokenizer.pad_token_id if tokenizer.pad_token_id is not None else 0
            left = getattr(tokenizer, "padding_side", "left") == "left"
            if left:
                input_ids = pad_sequence([i.flip(0) for i in ids], batch_first=True, padding_value=pad_id).flip(1)
            else:
                input_ids = pad_sequence(ids, batch_first=True, padding_value=pad_id)
            attention_mask = (input_ids != pad_id).long()
            # all-pad-token prompts edge case: trust original masks when equal length
            out = {"input_ids": input_ids, "attention_mask": attention_mask}
            for key in xs[0]:
                if key not in ("input_ids", "attention_mask"):
                    out[key] = [x[key] for x in xs]
            return out

        if sampler is None:
            sampler = dp_sampler(self, shuffle=shuffle)
            if sampler is not None:
                shuffle = False
        return DataLoader(self, batch_size=batch_size, collate_fn=collate_fn, shuffle=shuffle,
                          sampler=sampler, num_workers=0, drop_last=drop_last)


def ilql_collate_fn(elems: Iterable[ILQLElement]) -> ILQLBatch:
    return ILQLBatch(
        pad_sequence([x.input_ids for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.attention_mask for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.rewards for x in elems], batch_first=True, padding_value=0.0),
        pad_sequence([x.states_ixs for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.actions_ixs for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.dones for x in elems], batch_first=True, padding_value=0),
    )


class ILQLRolloutStorage(BaseRolloutStore):
    """Rollout storage for ILQL training (reference offline_pipeline.py:202-237)."""

    def __init__(self, input_ids, attention_mask, rewards, states_ixs, actions_ixs, dones):
        super().__init__()
        self.input_ids = input_ids
        self.attention_mask = attention_mask
        self.rewards = rewards
        self.states_ixs = states_ixs
        self.actions_ixs = actions_ixs
        self.dones = dones

    def __getitem__(self, ix: int) -> ILQLElement:
        return ILQLElement(
            self.input_ids[ix], self.attention_mask[ix], self.rewards[ix],
            self.states_ixs[ix], self.actions_ixs[ix], self.dones[ix],
        )

    def __len__(self) -> int:
        return len(self.input_ids)

    def create_loader(self, batch_size: int):
        sampler = dp_sampler(self, shuffle=True)
        return DataLoader(
            self, batch_size=batch_size, shuffle=sampler is None, sampler=sampler,
            collate_fn=ilql_collate_fn, drop_last=dist.is_initialized(),
        )


def ilql_seq2seq_collate_fn(elems: Iterable[ILQLSeq2SeqElement]) -> ILQLSeq2SeqBatch:
    return ILQLSeq2SeqBatch(
        pad_sequence([x.input_ids for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.attention_mask for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.decoder_input_ids for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.rewards for x in elems], batch_first=True, padding_value=0.0),
        pad_sequence([x.states_ixs for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.actions_ixs for x in elems], batch_first=True, padding_value=0),
        pad_sequence([x.dones for x in elems], batch_first=True, padding_value=0),
    )


class ILQLSeq2SeqRolloutStorage(BaseRolloutStore):
    """Rollout storage for seq2seq ILQL (reference offline_pipeline.py:252-289)."""

    def __init__(self, input_ids, attention_mask, decoder_input_ids, rewards, states_ixs, actions_ixs, dones):
        super().__init__()
        self.input_ids = input_ids
        self.attention_mask = attention_mask
        self.decoder_input_ids = decoder_input_ids
        self.rewards = rewards
        self.states_ixs = states_ixs
        self.actions_ixs = actions_ixs
        self.dones = dones

    def __getitem__(self, ix: int) -> ILQLSeq2SeqElement:
        return ILQLSeq2SeqElement(
            self.input_ids[ix], self.attention_mask[ix], self.decoder_input_ids[ix],
            self.rewards[ix], self.states_ixs[ix], self.actions_ixs[ix], self.dones[ix],
        )

    def __len__(self) -> int:
        return len(self.input_ids)

    def create_loader(self, batch_size: int):
        sampler = dp_sampler(self, shuffle=True)
        return DataLoader(
            self, batch_size=batch_size, shuffle=sampler is None, sampler=sampler,
            collate_fn=ilql_seq2seq_collate_fn, drop_last=dist.is_initialized(),
        )
