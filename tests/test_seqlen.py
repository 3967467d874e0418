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


def test_dense_partitions_fill_budget():
    """Dense packing (default): micros ~full + one small tail, vs balanced
    k equally-underfull micros; both respect the budget and partition."""
    import torch

    from polyrl_amd.core.seqlen import prepare_dynamic_batch
    from polyrl_amd.protocol import TensorBatch
    torch.manual_seed(0)
    B, L = 64, 512
    lens = torch.randint(300, 512, (B,))
    am = torch.zeros(B, L, dtype=torch.long)
    for i in range(B):
        am[i, :lens[i]] = 1
    batch = TensorBatch(tensors={"attention_mask": am,
                                 "input_ids": torch.zeros(B, L, dtype=torch.long)})
    budget = 8192
    micro_d, parts_d = prepare_dynamic_batch(batch, budget, packing="dense")
    micro_b, parts_b = prepare_dynamic_batch(batch, budget,
                                             packing="balanced")
    for parts in (parts_d, parts_b):
        got = sorted(i for p in parts for i in p)
        assert got == list(range(B))
        for p in parts:
            assert sum(int(lens[i]) for i in p) <= budget
    # dense: all but (at most) one micro filled to >= 75% of the budget --
    # the fixed-M padding threshold
    fills = sorted((sum(int(lens[i]) for i in p) for p in parts_d),
                   reverse=True)
    assert all(f >= 0.75 * budget for f in fills[:-1]), fills
    # and dense never uses more micros than balanced
    assert len(parts_d) <= len(parts_b) + 1
