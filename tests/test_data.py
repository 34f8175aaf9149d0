"""Data-layer tests: indexed dataset round trip, causal dataset, blendable,
zero padding packing."""
import os

import numpy as np
import pytest
import torch

from paddlenlp_amd.data.causal_dataset import (
    build_train_valid_test_datasets,
    get_train_valid_test_split_,
)
from paddlenlp_amd.data.indexed_dataset import (
    MMapIndexedDataset,
    MMapIndexedDatasetBuilder,
)
from paddlenlp_amd.datasets.zero_padding_dataset import (
    ZeroPaddingMapDataset,
    generate_startend_row_indices,
)


def make_corpus(tmp_path, n_docs=50, doc_len=100, vocab=1000, seed=0):
    rng = np.random.default_rng(seed)
    prefix = str(tmp_path / "corpus")
    builder = MMapIndexedDatasetBuilder(prefix, dtype=np.uint16)
    docs = []
    for _ in range(n_docs):
        doc = rng.integers(0, vocab, rng.integers(doc_len // 2, doc_len)).astype(np.uint16)
        builder.add_item(doc)
        builder.end_document()
        docs.append(doc)
    builder.finalize()
    return prefix, docs


def test_indexed_dataset_roundtrip(tmp_path):
    prefix, docs = make_corpus(tmp_path)
    ds = MMapIndexedDataset(prefix)
    assert len(ds) == len(docs)
    for i in (0, 7, len(docs) - 1):
        assert np.array_equal(ds[i], docs[i])
    # partial reads
    assert np.array_equal(ds.get(3, offset=5, length=10), docs[3][5:15])


def test_split_string():
    idx = get_train_valid_test_split_("949,50,1", 1000)
    assert idx[0] == 0 and idx[-1] == 1000
    assert idx[1] == 949


def test_causal_dataset(tmp_path):
    prefix, _ = make_corpus(tmp_path)
    train, valid, test = build_train_valid_test_datasets(
        prefix, "90,5,5", [64, 8, 8], seq_length=64, seed=0,
    )
    assert train is not None and len(train) >= 64
    ex = train[0]
    assert ex["input_ids"].shape == (64,)
    assert ex["labels"].shape == (64,)
    # labels are input_ids shifted by one in the underlying stream
    ex2 = train[1]
    assert not np.array_equal(ex["input_ids"], ex2["input_ids"])
    # deterministic across constructions (cached indexes)
    train_b, _, _ = build_train_valid_test_datasets(
        prefix, "90,5,5", [64, 8, 8], seq_length=64, seed=0,
    )
    assert np.array_equal(train[5]["input_ids"], train_b[5]["input_ids"])


def test_blendable(tmp_path):
    p1, _ = make_corpus(tmp_path, seed=1)
    os.rename(p1 + ".bin", str(tmp_path / "c2.bin"))
    os.rename(p1 + ".idx", str(tmp_path / "c2.idx"))
    p2, _ = make_corpus(tmp_path, seed=2)
    train, _, _ = build_train_valid_test_datasets(
        ["0.7", p2, "0.3", str(tmp_path / "c2")], "100,0,0", [40, 0, 0],
        seq_length=32, seed=0,
    )
    assert len(train) == 40
    counts = np.bincount(train.dataset_index, minlength=2)
    assert counts[0] == 28 and counts[1] == 12  # largest-remainder 70/30


def test_zero_padding_packing():
    samples = [
        {"input_ids": list(range(10)), "labels": list(range(10))},
        {"input_ids": list(range(20)), "labels": list(range(20))},
        {"input_ids": list(range(15)), "labels": list(range(15))},
    ]
    ds = ZeroPaddingMapDataset(samples, max_length=32)
    assert len(ds) == 2  # 10+20 fits, 15 in second pack
    ex = ds[0]
    assert ex["input_ids"].shape == (32,)
    idx = ex["attn_mask_startend_row_indices"]
    assert idx.shape == (1, 32, 1)
    # first sample's keys visible until row 10, second until 30, pad none
    assert idx[0, 0, 0] == 10 and idx[0, 10, 0] == 30 and idx[0, 30, 0] == 30
    assert (ex["labels"][30:] == -100).all()


def test_flashmask_semantics_match_block_diagonal():
    """Packed FlashMask == running samples separately."""
    from paddlenlp_amd.ops import reference

    torch.manual_seed(0)
    D, H = 8, 2
    s1, s2 = 6, 10
    q = torch.randn(1, 16, H, D)
    k = torch.randn(1, 16, H, D)
    v = torch.randn(1, 16, H, D)
    idx = torch.from_numpy(generate_startend_row_indices([s1, 16], 16))[None] \
        if False else torch.tensor(
            [[[ [s1] ]*s1 + [[16]]*(16-s1) ]], dtype=torch.int32).reshape(1, 1, 16, 1)
    out = reference.flash_attention(q, k, v, causal=True, startend_row_indices=idx)
    ref1 = reference.flash_attention(q[:, :s1], k[:, :s1], v[:, :s1], causal=True)
    ref2 = reference.flash_attention(q[:, s1:], k[:, s1:], v[:, s1:], causal=True)
    assert torch.allclose(out[:, :s1], ref1, atol=1e-5)
    assert torch.allclose(out[:, s1:], ref2, atol=1e-5)


def test_collate_stack_pad_tuple_dict():
    import numpy as np

    from paddlenlp_amd.data import Dict, Pad, Stack, Tuple

    stack = Stack(dtype="int64")
    out = stack([[1, 2], [3, 4]])
    assert out.shape == (2, 2) and out.dtype == np.int64

    pad = Pad(pad_val=9, ret_length=True)
    batch, lens = pad([[1, 2, 3], [4]])
    assert batch.tolist() == [[1, 2, 3], [4, 9, 9]]
    assert lens.tolist() == [3, 1]
    left = Pad(pad_val=0, pad_right=False)
    assert left([[1, 2], [3]]).tolist() == [[1, 2], [0, 3]]

    tup = Tuple(Stack(), Pad(pad_val=0))
    a, b = tup([([1, 2], [5]), ([3, 4], [6, 7])])
    assert a.tolist() == [[1, 2], [3, 4]]
    assert b.tolist() == [[5, 0], [6, 7]]

    dic = Dict({"x": Stack(), "y": Pad(pad_val=-1)})
    a, b = dic([{"x": [1], "y": [2, 3]}, {"x": [4], "y": [5]}])
    assert a.tolist() == [[1], [4]]
    assert b.tolist() == [[2, 3], [5, -1]]


def test_vocab_and_greedy_tokenizer():
    from paddlenlp_amd.data import JiebaLikeTokenizer, Vocab

    vocab = Vocab.build_vocab(
        [["hello", "world"], ["hello", "there"]],
        unk_token="[UNK]", pad_token="[PAD]")
    assert vocab.to_indices("hello") != vocab.to_indices("[UNK]")
    assert vocab.to_indices("missing") == vocab.to_indices("[UNK]")
    assert vocab.to_tokens(vocab.to_indices(["hello", "world"])) == ["hello", "world"]
    assert "hello" in vocab and len(vocab) >= 4

    cn_vocab = Vocab(token_to_idx={"深度": 0, "学习": 1, "深": 2, "度": 3,
                                   "[UNK]": 4}, unk_token="[UNK]")
    tok = JiebaLikeTokenizer(cn_vocab)
    assert tok.cut("深度学习") == ["深度", "学习"]  # longest match wins
    assert tok.cut("深度x") == ["深度", "x"]
    assert tok.encode("深度学习") == [0, 1]


def test_mlm_collator_masks_and_labels():
    import torch

    from paddlenlp_amd.data import DataCollatorForLanguageModeling

    class Tok:
        pad_token_id = 0
        cls_token_id = 1
        sep_token_id = 2
        mask_token_id = 3

        def __len__(self):
            return 50

    torch.manual_seed(0)
    feats = [{"input_ids": [1] + list(range(10, 28)) + [2]}
             for _ in range(4)]
    coll = DataCollatorForLanguageModeling(tokenizer=Tok(), mlm=True,
                                           mlm_probability=0.5)
    batch = coll(feats)
    ids, labels = batch["input_ids"], batch["labels"]
    # specials never masked
    assert (labels[:, 0] == -100).all() and (labels[:, -1] == -100).all()
    assert (ids[:, 0] == 1).all() and (ids[:, -1] == 2).all()
    # some positions masked, labels hold the originals there
    sel = labels != -100
    assert sel.any()
    orig = torch.tensor([f["input_ids"] for f in feats])
    assert (labels[sel] == orig[sel]).all()
    # masked positions mostly [MASK]
    assert (ids[sel] == 3).float().mean() > 0.5


def test_whole_word_mask_collator():
    import torch

    from paddlenlp_amd.data import DataCollatorForWholeWordMask

    class Tok:
        pad_token_id = 0
        cls_token_id = None
        sep_token_id = None
        mask_token_id = 3

        VOCAB = {10: "play", 11: "##ing", 12: "ball", 13: "##s", 14: "go"}

        def __len__(self):
            return 20

        def convert_ids_to_tokens(self, ids):
            return [self.VOCAB.get(i, f"tok{i}") for i in ids]

    torch.manual_seed(1)
    feats = [{"input_ids": [10, 11, 12, 13, 14]}]
    coll = DataCollatorForWholeWordMask(tokenizer=Tok(), mlm_probability=0.99)
    batch = coll(feats)
    labels = batch["labels"][0]
    # whole words masked together: 10/11 share fate, 12/13 share fate
    assert (labels[0] == -100) == (labels[1] == -100)
    assert (labels[2] == -100) == (labels[3] == -100)
    assert (labels != -100).any()


def test_token_classification_collator_pads_labels():
    from paddlenlp_amd.data import DataCollatorForTokenClassification

    feats = [{"input_ids": [5, 6, 7], "labels": [1, 2, 3]},
             {"input_ids": [8], "labels": [0]}]
    batch = DataCollatorForTokenClassification()(feats)
    assert batch["input_ids"].shape == (2, 3)
    assert batch["labels"][1].tolist() == [0, -100, -100]


def test_distributed_batch_sampler_consumed_samples_resume():
    """Resume skips exactly the consumed samples on the first epoch
    (reference consumed_samples replay, trainer.py:916-923)."""
    from paddlenlp_amd.data.sampler import DistributedBatchSampler

    ds = list(range(20))
    fresh = DistributedBatchSampler(ds, batch_size=2, num_replicas=2,
                                    rank=0, shuffle=False, drop_last=False,
                                    seed=0)
    all_batches = list(iter(fresh))
    resumed = DistributedBatchSampler(ds, batch_size=2, num_replicas=2,
                                      rank=0, shuffle=False, drop_last=False,
                                      seed=0, consumed_samples=8)
    rest = list(iter(resumed))
    # 8 consumed over 2 replicas = 4 per rank = 2 batches of 2 skipped
    assert rest == all_batches[2:], (rest, all_batches)
    # second epoch: reset -> full pass again
    resumed.consumed_samples = 0
    assert list(iter(resumed)) == all_batches
