"""Pipeline/store tests (parity: reference tests/test_pipelines.py +
test_minibatch.py)."""

import torch

from trlx_amd.data.ilql_types import ILQLBatch, flatten_dataclass, unflatten_dataclass
from trlx_amd.data.ppo_types import PPORLElement
from trlx_amd.pipeline import MiniBatchIterator
from trlx_amd.pipeline.offline_pipeline import DialogStore, PromptPipeline, tokenize_dialogue
from trlx_amd.pipeline.ppo_pipeline import PPORolloutStorage, ppo_collate_fn
from trlx_amd.utils.tokenizer import ByteTokenizer


def test_tokenize_dialogue_basic(byte_tokenizer):
    tok = byte_tokenizer
    msgs = tokenize_dialogue("hello", tok, max_length=32)
    # bare string -> (BOS prompt, output + eos)
    assert not msgs[0].is_output
    assert msgs[-1].is_output
    text = tok.decode([t for m in msgs for t in m.tokens], skip_special_tokens=False)
    assert "hello" in text
    assert text.endswith(tok.eos_token)


def test_tokenize_dialogue_interleaved_and_truncation(byte_tokenizer):
    tok = byte_tokenizer
    msgs = tokenize_dialogue(["q1", "a1", "q2", "a2"], tok, max_length=8)
    total = sum(len(m.tokens) for m in msgs)
    assert total <= 8
    # right truncation keeps the head
    tok.truncation_side = "right"
    msgs_r = tokenize_dialogue(["abcdef", "xyz"], tok, max_length=4)
    flat = [t for m in msgs_r for t in m.tokens]
    assert len(flat) <= 4
    # left truncation keeps the tail (must still end with eos)
    tok.truncation_side = "left"
    msgs_l = tokenize_dialogue(["abcdef", "xyz"], tok, max_length=4)
    flat_l = [t for m in msgs_l for t in m.tokens]
    assert len(flat_l) <= 5  # possible BOS insertion
    assert flat_l[-1] == tok.eos_token_id


def test_tokenize_dialogue_output_first_gets_bos(byte_tokenizer):
    tok = byte_tokenizer
    tok.truncation_side = "left"
    # prompt fully truncated away -> output-first -> BOS must be inserted
    # ("out" + eos = 4 tokens fill max_length exactly, evicting one token)
    msgs = tokenize_dialogue(["p" * 30, "out"], tok, max_length=4)
    assert not msgs[0].is_output
    assert msgs[0].tokens == (tok.bos_token_id,)
    assert sum(len(m.tokens) for m in msgs) <= 4 + 1


def test_prompt_pipeline_left_pad(byte_tokenizer):
    pipe = PromptPipeline(["ab", "abcdef"], max_prompt_length=10, tokenizer=byte_tokenizer)
    loader = pipe.create_loader(2)
    batch = next(iter(loader))
    ids, mask = batch["input_ids"], batch["attention_mask"]
    assert ids.shape == (2, 6)
    # row 0 left-padded
    assert (ids[0, :4] == byte_tokenizer.pad_token_id).all()
    assert mask[0].tolist() == [0, 0, 0, 0, 1, 1]
    assert mask[1].tolist() == [1] * 6


def test_prompt_pipeline_metadata_passthrough(byte_tokenizer):
    prompts = [{"prompt": "ab", "label": "x"}, {"prompt": "cd", "label": "y"}]
    pipe = PromptPipeline(prompts, 10, byte_tokenizer)
    batch = next(iter(pipe.create_loader(2)))
    assert batch["label"] == ["x", "y"]


def test_ppo_collate_padding():
    elems = [
        PPORLElement(torch.tensor([5, 6]), torch.tensor([7, 8, 9]),
                     torch.tensor([0.1, 0.2, 0.3]), torch.tensor([1.0, 2.0, 3.0]),
                     torch.tensor([0.0, 0.0, 1.0])),
        PPORLElement(torch.tensor([4]), torch.tensor([1, 2]),
                     torch.tensor([0.5, 0.6]), torch.tensor([5.0, 6.0]),
                     torch.tensor([0.0, 0.5])),
    ]
    batch = ppo_collate_fn("left", 99, elems)
    # queries left-padded
    assert batch.query_tensors.tolist() == [[5, 6], [99, 4]]
    # responses right-padded
    assert batch.response_tensors.tolist() == [[7, 8, 9], [1, 2, 99]]
    assert batch.logprobs.shape == (2, 3)
    assert batch.rewards[1].tolist() == [0.0, 0.5, 0.0]


def test_ppo_storage_and_loader():
    store = PPORolloutStorage(pad_token_id=0, padding_side="left")
    store.clear_history()
    elems = [
        PPORLElement(torch.tensor([1]), torch.tensor([2, 3]), torch.tensor([0.0, 0.0]),
                     torch.tensor([0.0, 0.0]), torch.tensor([0.0, 1.0]))
        for _ in range(6)
    ]
    store.push(elems)
    assert len(store) == 6
    loader = store.create_loader(2, shuffle=False)
    batches = list(loader)
    assert len(batches) == 3
    assert batches[0].query_tensors.shape == (2, 1)


def test_minibatch_iterator_dict_and_dataclass():
    data = [dict(a=torch.arange(4), b=torch.arange(4) * 2)]
    it = MiniBatchIterator(iter(data).__iter__() and data, mb_size=2, num_mb=2)
    mbs = next(iter(it))
    assert len(mbs) == 2
    assert mbs[0]["a"].tolist() == [0, 1]
    assert mbs[1]["a"].tolist() == [2, 3]

    batch = ILQLBatch(
        input_ids=torch.zeros(4, 3), attention_mask=torch.ones(4, 3), rewards=torch.zeros(4, 2),
        states_ixs=torch.zeros(4, 3), actions_ixs=torch.zeros(4, 2), dones=torch.ones(4, 3),
    )
    it = MiniBatchIterator([batch], mb_size=2, num_mb=2)
    mbs = next(iter(it))
    assert isinstance(mbs[0], ILQLBatch)
    assert mbs[0].input_ids.shape == (2, 3)


def test_minibatch_iterator_uneven_tail():
    data = [dict(a=torch.arange(3))]
    it = MiniBatchIterator(data, mb_size=2, num_mb=2)
    mbs = next(iter(it))
    assert len(mbs) == 2
    assert len(mbs[1]["a"]) == 1  # short tail kept with a warning


def test_flatten_unflatten_dataclass():
    batch = ILQLBatch(
        input_ids=torch.ones(1, 2), attention_mask=torch.ones(1, 2), rewards=torch.zeros(1, 1),
        states_ixs=torch.zeros(1, 2), actions_ixs=torch.zeros(1, 1), dones=torch.ones(1, 2),
    )
    flat = flatten_dataclass(ILQLBatch)(batch)
    assert len(flat) == 6
    rebuilt = unflatten_dataclass(ILQLBatch)(flat)
    assert torch.equal(rebuilt.input_ids, batch.input_ids)
    assert torch.equal(rebuilt.dones, batch.dones)


def test_dialog_store_labels_mask_prompt(byte_tokenizer):
    dialogs = [tokenize_dialogue(["ab", "cd"], byte_tokenizer, 16)]
    store = DialogStore(dialogs, byte_tokenizer)
    batch = next(iter(store.create_loader(1)))
    labels = batch["labels"][0]
    ids = batch["input_ids"][0]
    # prompt positions are -100, output positions match input ids
    assert (labels[:2] == -100).all()
    assert (labels[labels != -100] == ids[labels != -100]).all()


def test_rollout_json_export(tmp_path):
    """Rollout JSON export for Algorithm Distillation
    (reference ppo_pipeline.py:71-89)."""
    import json

    store = PPORolloutStorage(pad_token_id=0, padding_side="left")
    store.clear_history()
    store.push([
        PPORLElement(torch.tensor([1, 2]), torch.tensor([3, 4]), torch.zeros(2),
                     torch.zeros(2), torch.zeros(2))
    ])
    store.export_history(str(tmp_path))
    files = list(tmp_path.glob("epoch-*.json"))
    assert len(files) == 1
    data = json.loads(files[0].read_text())
    assert data[0]["query_tensor"] == [1, 2]
    assert data[0]["response_tensor"] == [3, 4]
    assert "logprobs" not in data[0]  # only_text=True


def test_running_moments_matches_batch_statistics():
    """RunningMoments over chunks converges to the full-population statistics
    (reference tests/test_utils.py test_running_moments)."""
    import torch
    from trlx_amd.utils.modeling import RunningMoments

    torch.manual_seed(0)
    rm = RunningMoments()
    all_chunks = []
    for _ in range(10):
        xs = torch.randn(100) * 3 + 1.5
        all_chunks.append(xs)
        b_mean, b_std = rm.update(xs)
        # per-batch return values are the batch's own stats
        assert abs(b_mean - xs.mean().item()) < 1e-4
        assert abs(b_std - xs.std().item()) < 1e-2
    full = torch.cat(all_chunks)
    assert abs(rm.mean - full.mean().item()) < 1e-4
    assert abs(rm.std - full.std().item()) < 1e-3
