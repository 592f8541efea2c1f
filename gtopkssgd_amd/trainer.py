"""DLTrainer — the training loop.

Reference parity: `upstream:dl_trainer.py` class DLTrainer (SURVEY.md
§2.1 L5, §3.2): model/dataset factory, ``train(num_of_iters)`` with
``nsteps_update`` gradient accumulation, ``update_model()``,
``test(epoch)`` (top-1/top-5 or perplexity), ``adjust_learning_rate``
(epoch-milestone ÷10 schedule), per-phase io/fwd/bwd/update timers and
samples/sec logging.

MI355X-native notes: compute runs under bf16 autocast (master weights
fp32 — the compression/update pipeline is fp32 end-to-end), images go
through channels_last for MIOpen, and the update is the
DistributedOptimizer's fused flat step.
"""
from __future__ import annotations

import time
from typing import Optional

import torch
import torch.nn as nn

from . import models as model_zoo
from .checkpoint import save_checkpoint
from .data import make_dataset, make_loader
from .models import repackage_hidden
from .settings import logger
from .utils import Timer

_LR_MILESTONES = {
    "cifar10": [80, 120],
    "imagenet": [30, 60, 80],
    "mnist": [],
    "ptb": [20, 30],
    "an4": [],
    "an4wav": [],
}


class DLTrainer:
    def __init__(self, rank: int = 0, size: int = 1, dnn: str = "resnet20",
                 dataset: str = "cifar10", batch_size: int = 32,
                 nsteps_update: int = 1, lr: float = 0.1,
                 device: Optional[str] = None, amp: bool = True,
                 num_workers: int = 2, n_samples: Optional[int] = None,
                 seed: int = 42, norm_clip: Optional[float] = None,
                 momentum: float = 0.9, weight_decay: float = 5e-4,
                 model_kwargs: Optional[dict] = None):
        self.rank = rank
        self.size = size
        self.dnn = dnn
        self.dataset = dataset
        self.batch_size = batch_size
        self.nsteps_update = nsteps_update
        self.base_lr = lr
        self.norm_clip = norm_clip
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.amp = amp and self.device.type == "cuda"

        torch.manual_seed(seed + rank)
        self.model_kwargs = dict(model_kwargs or {})
        self.net = model_zoo.create_net(dnn, dataset, **self.model_kwargs)
        self.is_lm = dnn == "lstm"
        self.is_ctc = dnn == "lstman4"
        self.is_image = not (self.is_lm or self.is_ctc)
        self.channels_last = (self.is_image and self.device.type == "cuda")
        self.net.to(self.device)
        if self.channels_last:
            self.net.to(memory_format=torch.channels_last)

        if self.is_ctc:
            self.criterion = nn.CTCLoss(blank=0, zero_infinity=True)
        else:
            self.criterion = nn.CrossEntropyLoss()

        self.optimizer = torch.optim.SGD(
            self.net.parameters(), lr=lr, momentum=momentum,
            weight_decay=weight_decay)

        self.data_prepare(num_workers, n_samples, seed)
        self.hidden = None
        self.train_iter = 0
        self.train_epoch = 0
        self.timer = Timer()
        self.avg_loss = 0.0
        self._loss_count = 0

    # ------------------------------------------------------------------
    def data_prepare(self, num_workers, n_samples, seed):
        data_kwargs = {}
        if self.is_lm:
            data_kwargs["vocab_size"] = getattr(self.net, "vocab_size", 10000)
        self.trainset = make_dataset(self.dataset, n_samples, seed,
                                     **data_kwargs)
        self.trainloader = make_loader(self.trainset, self.batch_size,
                                       self.rank, self.size,
                                       num_workers=num_workers, seed=seed)
        self.testset = make_dataset(self.dataset,
                                    max(64, (n_samples or 1024) // 10),
                                    seed + 1, **data_kwargs)
        self.testloader = make_loader(self.testset, self.batch_size,
                                      self.rank, self.size,
                                      num_workers=0, drop_last=False,
                                      seed=seed)
        self._train_it = iter(self.trainloader)

    def _next_batch(self):
        try:
            return next(self._train_it)
        except StopIteration:
            self.train_epoch += 1
            if hasattr(self.trainloader.sampler, "set_epoch"):
                self.trainloader.sampler.set_epoch(self.train_epoch)
            self._train_it = iter(self.trainloader)
            return next(self._train_it)

    # ------------------------------------------------------------------
    def _forward_loss(self, batch):
        if self.is_lm:
            tokens, targets = batch
            tokens = tokens.t().contiguous().to(self.device)   # [seq, B]
            targets = targets.t().contiguous().to(self.device)
            if self.hidden is None or \
                    self.hidden[0].size(1) != tokens.size(1):
                self.hidden = self.net.init_hidden(tokens.size(1))
            self.hidden = repackage_hidden(self.hidden)
            logits, self.hidden = self.net(tokens, self.hidden)
            return self.criterion(logits.reshape(-1, logits.size(-1)),
                                  targets.reshape(-1))
        if self.is_ctc:
            x, ys = batch
            x = x.to(self.device)
            logits = self.net(x)  # [T, B, L]
            log_probs = logits.log_softmax(2)
            t, b, _ = log_probs.shape
            input_lengths = torch.full((b,), t, dtype=torch.long)
            target_lengths = torch.tensor([len(y) for y in ys],
                                          dtype=torch.long)
            targets = torch.cat([torch.as_tensor(y) for y in ys])
            return self.criterion(log_probs, targets, input_lengths,
                                  target_lengths)
        x, y = batch
        x = x.to(self.device, non_blocking=True)
        if self.channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        y = y.to(self.device, non_blocking=True)
        out = self.net(x)
        return self.criterion(out, y)

    def train(self, num_of_iters: int = 1, dist_optimizer=None):
        """Run fwd/bwd for num_of_iters micro-steps, accumulating
        gradients (grads land in the flat buffer via .grad views)."""
        self.net.train()
        total_loss = 0.0
        for s in range(num_of_iters):
            self.timer.start("io")
            batch = self._next_batch()
            self.timer.stop("io")
            self.timer.start("fwd")
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                                enabled=self.amp):
                loss = self._forward_loss(batch)
            self.timer.stop("fwd")
            self.timer.start("bwd")
            if dist_optimizer is not None and \
                    hasattr(dist_optimizer, "set_accumulation_boundary"):
                dist_optimizer.set_accumulation_boundary(
                    s == num_of_iters - 1)
            (loss / self.nsteps_update).backward()
            self.timer.stop("bwd")
            total_loss += loss.item()
            self.train_iter += 1
        self.avg_loss += total_loss / num_of_iters
        self._loss_count += 1
        return total_loss / num_of_iters

    def update_model(self, dist_optimizer=None):
        """Apply the (possibly distributed) optimizer step."""
        opt = dist_optimizer if dist_optimizer is not None else self.optimizer
        self.timer.start("update")
        if self.norm_clip is not None:
            if hasattr(opt, "clip_grad_norm_"):
                opt.clip_grad_norm_(self.norm_clip)  # one flat-norm kernel
            else:
                torch.nn.utils.clip_grad_norm_(self.net.parameters(),
                                               self.norm_clip)
        opt.step()
        opt.zero_grad()
        self.timer.stop("update")

    # ------------------------------------------------------------------
    def adjust_learning_rate(self, epoch: int, opt=None):
        """Epoch-milestone schedule: ÷10 at each milestone."""
        opt = opt if opt is not None else self.optimizer
        milestones = _LR_MILESTONES.get(self.dataset, [])
        lr = self.base_lr
        for m in milestones:
            if epoch >= m:
                lr *= 0.1
        for group in opt.param_groups:
            group["lr"] = lr
        return lr

    # ------------------------------------------------------------------
    def _sync_metrics(self, metrics: dict) -> dict:
        """C6: average scalar metrics across ranks when distributed.

        The scalar tensor lives on the backend's device: RCCL (backend
        "nccl") can only reduce GPU-resident tensors, gloo reduces CPU
        tensors — all-reducing a CPU tensor under NCCL raises.
        """
        import torch.distributed as dist
        if not (dist.is_available() and dist.is_initialized()) or \
                dist.get_world_size() == 1:
            return metrics
        backend = dist.get_backend()
        dev = self.device if backend == "nccl" else torch.device("cpu")
        keys = sorted(metrics)
        t = torch.tensor([float(metrics[k]) for k in keys], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        t /= dist.get_world_size()
        return {k: float(v) for k, v in zip(keys, t.cpu())}

    @torch.no_grad()
    def test(self, epoch: int = 0):
        """Top-1/top-5 for classification, perplexity for the LM,
        greedy-decode sanity for CTC.  Returns a metrics dict."""
        self.net.eval()
        if self.is_lm:
            total_loss, count = 0.0, 0
            hidden = None
            for tokens, targets in self.testloader:
                tokens = tokens.t().contiguous().to(self.device)
                targets = targets.t().contiguous().to(self.device)
                if hidden is None or hidden[0].size(1) != tokens.size(1):
                    hidden = self.net.init_hidden(tokens.size(1))
                hidden = repackage_hidden(hidden)
                logits, hidden = self.net(tokens, hidden)
                total_loss += self.criterion(
                    logits.reshape(-1, logits.size(-1)),
                    targets.reshape(-1)).item()
                count += 1
            ppl = float(torch.exp(torch.tensor(total_loss / max(1, count))))
            logger.info("[epoch %d] test perplexity %.2f", epoch, ppl)
            return self._sync_metrics({"perplexity": ppl})
        if self.is_ctc:
            from .models import GreedyDecoder
            decoder = GreedyDecoder()
            total_loss, count = 0.0, 0
            wer_sum, cer_sum, n_utt = 0.0, 0.0, 0
            for x, ys in self.testloader:
                x = x.to(self.device)
                logits = self.net(x)  # [T, B, L]
                log_probs = logits.log_softmax(2)
                t, b, _ = log_probs.shape
                input_lengths = torch.full((b,), t, dtype=torch.long)
                target_lengths = torch.tensor([len(y) for y in ys],
                                              dtype=torch.long)
                targets = torch.cat([torch.as_tensor(y) for y in ys])
                total_loss += self.criterion(
                    log_probs, targets, input_lengths,
                    target_lengths).item()
                count += 1
                for hyp, y in zip(decoder.decode(logits), ys):
                    ref = decoder.labels_to_string(y)
                    wer_sum += decoder.wer(hyp, ref)
                    cer_sum += decoder.cer(hyp, ref)
                    n_utt += 1
            avg = total_loss / max(1, count)
            wer = wer_sum / max(1, n_utt)
            cer = cer_sum / max(1, n_utt)
            logger.info("[epoch %d] test ctc-loss %.4f wer %.4f cer %.4f",
                        epoch, avg, wer, cer)
            return self._sync_metrics({"ctc_loss": avg, "wer": wer,
                                       "cer": cer})
        top1, top5, total = 0, 0, 0
        for x, y in self.testloader:
            x = x.to(self.device)
            y = y.to(self.device)
            out = self.net(x)
            _, pred = out.topk(min(5, out.size(1)), dim=1)
            correct = pred.eq(y.view(-1, 1))
            top1 += int(correct[:, :1].sum())
            top5 += int(correct.sum())
            total += y.numel()
        acc1, acc5 = top1 / max(1, total), top5 / max(1, total)
        logger.info("[epoch %d] test top1 %.4f top5 %.4f", epoch, acc1, acc5)
        return self._sync_metrics({"top1": acc1, "top5": acc5})

    # ------------------------------------------------------------------
    def save(self, prefix: str, run_tag: str, epoch: int, optimizer=None):
        """Call from every rank: rank 0 writes model+optimizer, every
        rank writes its own EF-residual sidecar."""
        return save_checkpoint(prefix, run_tag, epoch, self.net,
                               optimizer, rank=self.rank)

    def get_acc(self):
        """Reference parity: running average training loss since the
        last reset (`upstream:dl_trainer.py` get_acc)."""
        avg = self.avg_loss / max(1, self._loss_count)
        self.avg_loss = 0.0
        self._loss_count = 0
        return avg

    def speed_summary(self, iters: int, wall: float) -> str:
        sps = iters * self.batch_size / wall
        return (f"Speed: {sps:.1f} samples/sec ({self.timer.summary()})")
