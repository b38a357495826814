"""Single-process distributed helper semantics."""

import torch

from yet_another_mobilenet_series_amd.parallel import distributed as udist
from yet_another_mobilenet_series_amd.utils.common import AverageMeter, accuracy


def test_rank_helpers_without_init():
    assert udist.get_rank() == 0
    assert udist.get_world_size() == 1
    assert udist.is_master()


def test_all_reduce_identity_world1():
    t = torch.arange(6, dtype=torch.float32)
    out = udist.dist_all_reduce_tensor(t.clone(), op="sum")
    assert torch.equal(out, t)
    out = udist.dist_all_reduce_tensor(t.clone(), op="mean")
    assert torch.equal(out, t)


def test_master_only_decorator():
    calls = []

    @udist.master_only
    def f(v):
        calls.append(v)

    f(1)
    assert calls == [1]  # world 1: rank 0 is master


def test_average_meter():
    m = AverageMeter("x")
    m.update(1.0, 2)
    m.update(3.0, 2)
    assert m.avg == 2.0
    assert m.val == 3.0


def test_accuracy_topk():
    logits = torch.tensor([[0.1, 0.9, 0.0], [0.8, 0.1, 0.1]])
    y = torch.tensor([1, 2])
    top1, top2 = accuracy(logits, y, (1, 2))
    assert top1.item() == 50.0
    assert top2.item() == 50.0
