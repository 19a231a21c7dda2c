"""Training side-cars (ref:utils/utils.py): EDE schedule, checkpointing,
meters, top-k accuracy.  Same public API; checkpoint format is the
compatibility contract (checkpoint.pth.tar + model_best.pth.tar copy,
ref:utils/utils.py:21-25)."""

import math
import os
import shutil

import torch

# EDE t-range: t sweeps 1e-2 -> 1e1 log-linearly over training
# (ref:utils/utils.py:6-14); k = max(1/t, 1).
T_MIN, T_MAX = 1e-2, 1e1


def cpt_tk(epoch, tot_epochs):
    """Per-epoch (t, k) for the EDE backward k*t*(1-tanh^2(t*x))."""
    frac = epoch / tot_epochs
    t = 10.0 ** (math.log10(T_MIN) + (math.log10(T_MAX) - math.log10(T_MIN)) * frac)
    k = max(1.0 / t, 1.0)
    return torch.tensor([t], dtype=torch.float32), torch.tensor([k], dtype=torch.float32)


def find_weight_tensor_by_name(model, name_in):
    for name, param in model.named_parameters():
        if name == name_in:
            return param
    return None


def save_checkpoint(state, is_best, save_path):
    """Write <save_path>/checkpoint.pth.tar; copy to model_best.pth.tar if best.

    Atomic: writes to a temp file then renames, so a crash mid-save never
    corrupts the resume point (an upgrade over the reference).
    """
    os.makedirs(save_path, exist_ok=True)
    filename = os.path.join(save_path, "checkpoint.pth.tar")
    tmp = filename + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, filename)
    if is_best:
        shutil.copyfile(filename, os.path.join(save_path, "model_best.pth.tar"))


class AverageMeter:
    """Running value/average meter (ref:utils/utils.py:27-51)."""

    def __init__(self, name, fmt=":f"):
        self.name = name
        self.fmt = fmt
        self.reset()

    def reset(self):
        self.val = 0.0
        self.avg = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val, n=1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / max(self.count, 1)

    def set_totals(self, total, count, last=None):
        """Load the meter from an externally accumulated (sum, count) —
        used by the device-side meter accumulation path, which syncs to
        host only at print_freq/epoch boundaries."""
        self.sum = float(total)
        self.count = int(count)
        self.avg = self.sum / max(self.count, 1)
        self.val = self.avg if last is None else float(last)

    def get_avg(self):
        return self.avg

    def all_reduce(self, device=None):
        """Cross-rank (sum, count) all-reduce so every rank logs global
        averages — absent in the reference (SURVEY.md C4 gap)."""
        import torch.distributed as dist
        if not (dist.is_available() and dist.is_initialized()):
            return
        dev = device if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
        t = torch.tensor([self.sum, float(self.count)], dtype=torch.float64, device=dev)
        dist.all_reduce(t)
        self.sum, self.count = t[0].item(), int(t[1].item())
        self.avg = self.sum / max(self.count, 1)

    def __str__(self):
        fmtstr = "{name} {val" + self.fmt + "} ({avg" + self.fmt + "})"
        return fmtstr.format(**self.__dict__)


class ProgressMeter:
    def __init__(self, num_batches, meters, logger, prefix=""):
        digits = len(str(num_batches))
        self.fmt = "[{:" + str(digits) + "d}/" + str(num_batches) + "]"
        self.meters = meters
        self.prefix = prefix
        self.logger = logger

    def display(self, batch):
        entries = [self.prefix + self.fmt.format(batch)]
        entries += [str(m) for m in self.meters]
        self.logger.info("\t".join(entries))


def correct_counts(output, target, topk=(1, 5)):
    """Raw top-k correct counts as ONE device tensor [len(topk)] — no
    host sync; the trainer accumulates these device-side and reads them
    back only at print_freq/epoch end (the reference syncs 4x per step,
    ref:train.py:518-524)."""
    with torch.no_grad():
        maxk = max(topk)
        _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)
        correct = pred.eq(target.view(-1, 1))
        return torch.stack(
            [correct[:, :k].sum(dtype=torch.float32) for k in topk])


def accuracy(output, target, topk=(1,)):
    """Top-k accuracy in percent; returns a list of 1-element tensors
    (call-site contract: acc1[0], ref:train.py:518-523)."""
    with torch.no_grad():
        maxk = max(topk)
        batch = target.size(0)
        _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)
        correct = pred.eq(target.view(-1, 1))
        res = []
        for k in topk:
            ck = correct[:, :k].sum(dtype=torch.float32)
            res.append(ck.mul(100.0 / batch).reshape(1))
        return res
