import torch
import torch.nn as nn

from yet_another_mobilenet_series_amd.models import build_model
from yet_another_mobilenet_series_amd.utils import checkpoint as ckpt
from yet_another_mobilenet_series_amd.utils.optim import ExponentialMovingAverage


def test_save_load_round_trip(tmp_path):
    m = build_model("mobilenet_v3_small", width_mult=0.75, dropout=0.0)
    opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
    ema = ExponentialMovingAverage(m, 0.9)
    x = torch.randn(2, 3, 64, 64)
    m(x).sum().backward()
    opt.step()
    ema.update()
    path = str(tmp_path / "ck.pt")
    ckpt.save_checkpoint(path, m, opt, ema, epoch=3, global_step=42,
                         meta={"best_top1": 1.5})

    m2 = build_model("mobilenet_v3_small", width_mult=0.75, dropout=0.0)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9)
    ema2 = ExponentialMovingAverage(m2, 0.9)
    state = ckpt.load_checkpoint(path, m2, opt2, ema2)
    assert state["epoch"] == 3
    assert state["global_step"] == 42
    for (k, a), (_, b) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert torch.equal(a, b), k
    for k in ema.shadow:
        assert torch.equal(ema.shadow[k], ema2.shadow[k])
    # momentum buffers restored
    s1 = opt.state_dict()["state"]
    s2 = opt2.state_dict()["state"]
    assert all(torch.equal(s1[k]["momentum_buffer"], s2[k]["momentum_buffer"])
               for k in s1)


def test_remap_reshapes_conv_vs_linear_head():
    m = build_model("mobilenet_v1", dropout=0.0)
    sd = m.state_dict()
    # simulate a reference checkpoint storing the classifier as 1x1 conv
    foreign = {k: v.clone() for k, v in sd.items()}
    w = foreign["classifier.fc.weight"]
    foreign["classifier.fc.weight"] = w.reshape(w.shape[0], w.shape[1], 1, 1)
    out = ckpt.remap_state_dict(foreign, m)
    assert out["classifier.fc.weight"].shape == w.shape
    m.load_state_dict(out)


def test_pretrained_loads_weights_only(tmp_path):
    m = build_model("mobilenet_v2", dropout=0.0)
    path = str(tmp_path / "pre.pt")
    ckpt.save_checkpoint(path, m, epoch=9)
    m2 = build_model("mobilenet_v2", dropout=0.0)
    ckpt.load_pretrained(path, m2)
    for (k, a), (_, b) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert torch.equal(a, b), k
