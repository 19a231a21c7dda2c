"""Side-cars: meters, accuracy, metrics writer, pool CPU fallback,
no_sync context."""

import json
import os

import torch

from bdbnn_amd.utils import AverageMeter, ProgressMeter, accuracy
from bdbnn_amd.utils.metrics import MetricsWriter, trace_range
from bdbnn_amd.ops.pool import FusedMaxPool2d
from bdbnn_amd.parallel import BucketedDataParallel
from bdbnn_amd.models import cifar10 as cm


def test_average_meter():
    m = AverageMeter("x", ":.2f")
    m.update(1.0, 2)
    m.update(3.0, 2)
    assert m.avg == 2.0 and m.count == 4
    assert "x" in str(m)


def test_accuracy_topk():
    out = torch.tensor([[0.1, 0.9, 0.0], [0.8, 0.1, 0.1]])
    target = torch.tensor([1, 2])
    acc1, acc3 = accuracy(out, target, topk=(1, 3))
    assert acc1.item() == 50.0
    assert acc3.item() == 100.0


def test_metrics_writer_jsonl(tmp_path):
    w = MetricsWriter(str(tmp_path))
    w.add_scalar("loss", 1.5, 0)
    w.add_scalar("loss", 1.2, 1)
    w.close()
    lines = open(os.path.join(str(tmp_path), "metrics.jsonl")).readlines()
    recs = [json.loads(l) for l in lines]
    assert recs[0]["tag"] == "loss" and recs[1]["value"] == 1.2


def test_trace_range_noop_on_cpu():
    with trace_range("fwd"):
        pass


def test_fused_maxpool_cpu_fallback():
    mp = FusedMaxPool2d(3, 2, 1)
    x = torch.randn(2, 16, 9, 9, requires_grad=True)
    out = mp(x)
    ref = torch.nn.functional.max_pool2d(x.detach(), 3, 2, 1)
    assert torch.equal(out, ref)
    out.sum().backward()
    assert x.grad is not None


def test_no_sync_context():
    model = BucketedDataParallel(cm.resnet20())
    assert model.require_backward_grad_sync
    with model.no_sync():
        assert not model.require_backward_grad_sync
    assert model.require_backward_grad_sync


def test_progress_meter_formats(caplog):
    import logging
    logger = logging.getLogger("t")
    m = AverageMeter("loss", ":.2f")
    m.update(1.0)
    pm = ProgressMeter(100, [m], logger, prefix="E[0]")
    with caplog.at_level(logging.INFO, logger="t"):
        pm.display(5)
    assert "loss" in caplog.text
