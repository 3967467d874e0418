import torch

from polyrl_amd.core.seqlen import (balance_batch_indices, fixed_micro_batches,
                                    get_seqlen_balanced_partitions,
                                    prepare_dynamic_batch,
                                    restore_dynamic_batch)
from polyrl_amd.protocol import TensorBatch


def test_equal_size_partitions_are_balanced():
    seqlens = [100, 1, 1, 100, 50, 50, 25, 75]
    parts = get_seqlen_balanced_partitions(seqlens, 4, equal_size=True)
    assert sorted(i for p in parts for i in p) == list(range(8))
    assert all(len(p) == 2 for p in parts)
    sums = [sum(seqlens[i] for i in p) for p in parts]
    assert max(sums) - min(sums) <= 50


def test_unequal_partitions_cover_all():
    seqlens = [10, 20, 30, 40, 50]
    parts = get_seqlen_balanced_partitions(seqlens, 2, equal_size=False)
    assert sorted(i for p in parts for i in p) == list(range(5))
    sums = [sum(seqlens[i] for i in p) for p in parts]
    assert max(sums) <= 90


def test_balance_batch_indices_permutation():
    mask = torch.zeros(8, 16)
    for i, L in enumerate([16, 1, 2, 15, 8, 8, 4, 12]):
        mask[i, :L] = 1
    idx = balance_batch_indices(mask, 4)
    assert sorted(idx.tolist()) == list(range(8))


def make_batch(seqlens, L=32):
    n = len(seqlens)
    am = torch.zeros(n, L)
    for i, s in enumerate(seqlens):
        am[i, :s] = 1
    return TensorBatch.from_dict(tensors={
        "attention_mask": am,
        "val": torch.arange(n).float(),
    })


def test_dynamic_batch_budget_respected_and_restores():
    seqlens = [30, 10, 25, 5, 20, 15, 8, 12]
    b = make_batch(seqlens)
    micro, parts = prepare_dynamic_batch(b, max_token_len=40)
    for mb, p in zip(micro, parts):
        assert mb["attention_mask"].sum().item() <= 40
    # restore ordering
    vals = torch.cat([mb["val"] for mb in micro])
    restored = restore_dynamic_batch(vals, parts)
    assert torch.equal(restored, b["val"])


def test_fixed_micro_batches():
    b = make_batch([4] * 10)
    micro, parts = fixed_micro_batches(b, 4)
    assert [len(m) for m in micro] == [4, 4, 2]
