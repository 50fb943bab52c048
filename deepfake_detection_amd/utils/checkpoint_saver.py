"""CheckpointSaver — ranked `.pth.tar` retention, best-copy, backup mirror,
mid-epoch recovery files.

Parity: reference dfd/timm/utils.py:36-149 — save-state dict layout
{epoch, arch, state_dict, optimizer, args, version:2 [, amp, state_dict_ema,
metric]}, `checkpoint-{epoch}.pth.tar` naming, top-`max_history` ranked
retention, `model_best.pth.tar`, `_bak` dir mirror, `recovery-{epoch}-
{batch}.pth.tar` with previous-file cleanup, `find_recovery`.
"""

import glob
import logging
import operator
import os
import shutil

import torch

from .model import get_state_dict

_logger = logging.getLogger(__name__)


class CheckpointSaver:
    def __init__(self, checkpoint_prefix="checkpoint", recovery_prefix="recovery",
                 checkpoint_dir="", recovery_dir="", backup_dir="",
                 decreasing=False, max_history=10):
        # state
        self.checkpoint_files = []  # (filename, metric) sorted best -> worst
        self.best_epoch = None
        self.best_metric = None
        self.curr_recovery_file = ""
        self.last_recovery_file = ""

        # config
        self.checkpoint_dir = checkpoint_dir
        self.recovery_dir = recovery_dir
        self.backup_dir = backup_dir
        self.save_prefix = checkpoint_prefix
        self.recovery_prefix = recovery_prefix
        self.extension = ".pth.tar"
        self.decreasing = decreasing  # lower metric is better (e.g. loss)
        self.cmp = operator.lt if decreasing else operator.gt
        self.max_history = max_history
        assert self.max_history >= 1

    def save_checkpoint(self, model, optimizer, args, epoch, model_ema=None,
                        metric=None, use_amp=False, amp_state=None):
        assert epoch >= 0
        worst_file = self.checkpoint_files[-1] if self.checkpoint_files else None
        if len(self.checkpoint_files) < self.max_history or metric is None \
                or self.cmp(metric, worst_file[1]):
            if len(self.checkpoint_files) >= self.max_history:
                self._cleanup_checkpoints(1)

            filename = "-".join([self.save_prefix, str(epoch)]) + self.extension
            save_path = os.path.join(self.checkpoint_dir, filename)
            self._save(save_path, model, optimizer, args, epoch, model_ema, metric,
                       use_amp, amp_state)
            if self.backup_dir:
                try:
                    shutil.copyfile(save_path, os.path.join(self.backup_dir, filename))
                except OSError as e:
                    _logger.error("Backup copy failed: %s", e)
            self.checkpoint_files.append((save_path, metric))
            self.checkpoint_files = sorted(
                self.checkpoint_files, key=lambda x: x[1] if x[1] is not None else float("inf"),
                reverse=not self.decreasing)

            checkpoints_str = "Current checkpoints:\n"
            for c in self.checkpoint_files:
                checkpoints_str += " {}\n".format(c)
            _logger.info(checkpoints_str)

            if metric is not None and (self.best_metric is None or self.cmp(metric, self.best_metric)):
                self.best_epoch = epoch
                self.best_metric = metric
                shutil.copyfile(
                    save_path, os.path.join(self.checkpoint_dir, "model_best" + self.extension))

        return (None, None) if self.best_metric is None else (self.best_metric, self.best_epoch)

    def _save(self, save_path, model, optimizer, args, epoch, model_ema=None,
              metric=None, use_amp=False, amp_state=None):
        save_state = {
            "epoch": epoch,
            "arch": getattr(args, "model", "unknown"),
            "state_dict": get_state_dict(model),
            "optimizer": optimizer.state_dict(),
            "args": args,
            "version": 2,  # version 2: epoch saved at END of epoch (resume +1)
        }
        if use_amp and amp_state is not None:
            # bf16 training needs no loss scaler; slot kept for layout parity
            save_state["amp"] = amp_state
        if model_ema is not None:
            save_state["state_dict_ema"] = get_state_dict(model_ema)
        if metric is not None:
            save_state["metric"] = metric
        torch.save(save_state, save_path)

    def _cleanup_checkpoints(self, trim=0):
        trim = min(len(self.checkpoint_files), trim)
        delete_index = self.max_history - trim
        if delete_index <= 0 or len(self.checkpoint_files) <= delete_index:
            return
        to_delete = self.checkpoint_files[delete_index:]
        for d in to_delete:
            try:
                _logger.debug("Cleaning checkpoint: %s", d)
                os.remove(d[0])
            except OSError as e:
                _logger.error("Exception (%s) while deleting checkpoint", e)
        self.checkpoint_files = self.checkpoint_files[:delete_index]

    def save_recovery(self, model, optimizer, args, epoch, model_ema=None,
                      use_amp=False, amp_state=None, batch_idx=0):
        assert epoch >= 0
        filename = "-".join([self.recovery_prefix, str(epoch), str(batch_idx)]) + self.extension
        save_path = os.path.join(self.recovery_dir, filename)
        self._save(save_path, model, optimizer, args, epoch, model_ema,
                   use_amp=use_amp, amp_state=amp_state)
        if os.path.exists(self.last_recovery_file):
            try:
                _logger.debug("Cleaning recovery: %s", self.last_recovery_file)
                os.remove(self.last_recovery_file)
            except OSError as e:
                _logger.error("Exception (%s) while removing %s", e, self.last_recovery_file)
        self.last_recovery_file = self.curr_recovery_file
        self.curr_recovery_file = save_path

    def find_recovery(self):
        recovery_path = os.path.join(self.recovery_dir, self.recovery_prefix)
        files = glob.glob(recovery_path + "*" + self.extension)
        files = sorted(files)
        return files[0] if len(files) else ""
