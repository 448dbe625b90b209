"""Experiment orchestration (reference: ``experiment_builder.py``).

Same capabilities: epoch loop over train iterations; per-epoch validation;
best-val tracking; checkpoint every epoch (``train_model_{epoch}`` +
``train_model_latest``) with auto-resume; CSV/JSON statistics; planned pause
(``total_epochs_before_pause`` -> clean exit); final top-N-checkpoint
ensemble evaluation on the test set.

Distributed additions (no reference equivalent): rank 0 owns checkpoints,
CSV/JSON and progress printing; validation/test metrics are all-reduced
across ranks so every rank agrees on best-val bookkeeping.
"""

from __future__ import annotations

import os
import sys
import time
from collections import defaultdict
from typing import Dict, List, Optional

import numpy as np
import torch

from .storage import build_experiment_folder, save_statistics, save_to_json


class ExperimentBuilder:
    def __init__(self, args, data, model, device, dist_ctx=None):
        self.args = args
        self.data = data
        self.model = model
        self.device = device
        self.dist = dist_ctx
        self.rank = dist_ctx.rank if dist_ctx else 0
        self.world_size = dist_ctx.world_size if dist_ctx else 1

        self.saved_models_filepath, self.logs_filepath, self.visuals_filepath = \
            build_experiment_folder(args.experiment_name,
                                    root=getattr(args, "experiment_root", "."))

        self.total_losses: Dict[str, List[float]] = defaultdict(list)
        self.state = {
            "current_iter": 0,
            "best_val_acc": 0.0,
            "best_val_iter": 0,
            "best_epoch": 0,
            "per_epoch_statistics": defaultdict(list),
        }
        self.start_epoch = 0
        self.augment_flag = "omniglot" in args.dataset_name.lower()

        self._maybe_resume()

    # ------------------------------------------------------------------
    def _latest_ckpt_exists(self) -> bool:
        return os.path.isfile(os.path.join(self.saved_models_filepath,
                                           "train_model_latest"))

    def _maybe_resume(self) -> None:
        cont = self.args.continue_from_epoch
        if cont in ("from_scratch", -2, "-2"):
            return
        if cont == "latest":
            if not self._latest_ckpt_exists():
                return
            idx = "latest"
        else:
            idx = int(cont)
        state = self.model.load_model(self.saved_models_filepath, "train_model", idx)
        for k in ("current_iter", "best_val_acc", "best_val_iter", "best_epoch"):
            if k in state:
                self.state[k] = state[k]
        if "per_epoch_statistics" in state:
            self.state["per_epoch_statistics"] = defaultdict(
                list, state["per_epoch_statistics"])
        self.start_epoch = int(self.state["current_iter"] // self.args.total_iter_per_epoch)
        if hasattr(self.data, "continue_from_iter"):
            self.data.continue_from_iter(self.state["current_iter"])
        self._print(f"resumed from iter {self.state['current_iter']} "
                    f"(epoch {self.start_epoch})")

    def _print(self, *a) -> None:
        if self.rank == 0:
            print("[experiment]", *a, flush=True)

    # ------------------------------------------------------------------
    def train_iteration(self, train_sample, epoch_idx: int,
                        current_iter: int) -> Dict[str, float]:
        losses, _ = self.model.run_train_iter(data_batch=train_sample, epoch=epoch_idx)
        for k, v in losses.items():
            if isinstance(v, (int, float)):
                self.total_losses[f"train_{k}"].append(float(v))
        return losses

    def evaluation_iteration(self, val_sample) -> Dict[str, float]:
        losses, _ = self.model.run_validation_iter(data_batch=val_sample)
        return losses

    # ------------------------------------------------------------------
    def _dist_mean_std(self, vals: List[float]) -> tuple:
        """World-size-invariant mean/std: all ranks contribute their local
        per-batch values via an all-reduced (count, sum, sum-of-squares)
        triple, so the logged statistics describe the full evaluation set
        regardless of GPU count."""
        n = float(len(vals))
        s = float(np.sum(vals)) if vals else 0.0
        q = float(np.sum(np.square(vals))) if vals else 0.0
        if self.dist is not None and self.world_size > 1:
            n, s, q = self.dist.all_reduce_sum_vector([n, s, q])
        if n == 0:
            return 0.0, 0.0
        mean = s / n
        var = max(0.0, q / n - mean * mean)
        return mean, float(np.sqrt(var))

    def _epoch_summary(self, val_losses: List[Dict[str, float]],
                       epoch: int, epoch_time: float) -> Dict[str, float]:
        summary: Dict[str, float] = {"epoch": epoch, "epoch_run_time": epoch_time}
        for key in ("train_loss", "train_accuracy"):
            vals = self.total_losses.get(key, [])
            if vals:
                m, s = self._dist_mean_std(vals)
                summary[f"{key}_mean"] = m
                summary[f"{key}_std"] = s
        for key in ("loss", "accuracy"):
            vals = [l[key] for l in val_losses if key in l]
            if vals:
                m, s = self._dist_mean_std(vals)
                summary[f"val_{key}_mean"] = m
                summary[f"val_{key}_std"] = s
        self.total_losses = defaultdict(list)
        return summary

    def save_models(self, epoch: int) -> None:
        if self.rank != 0:
            return
        for name in (f"train_model_{epoch}", "train_model_latest"):
            self.model.save_model(
                os.path.join(self.saved_models_filepath, name), state=self.state)

    def pack_and_save_metrics(self, summary: Dict[str, float], first: bool) -> None:
        # per_epoch_statistics is maintained on EVERY rank (the final
        # ensemble test ranks epochs from it on all ranks and participates
        # symmetrically in the collectives); only the file writes are
        # rank-0-gated.
        stats = self.state["per_epoch_statistics"]
        for k, v in summary.items():
            stats[k].append(v)
        if self.rank != 0:
            return
        header = sorted(summary.keys())
        csv_path = os.path.join(self.logs_filepath, "summary_statistics.csv")
        if first or not os.path.isfile(csv_path):
            # header whenever the file is missing, not only at epoch 0 —
            # resuming into a fresh logs dir must not yield headerless rows
            save_statistics(self.logs_filepath, header, create=True)
        save_statistics(self.logs_filepath, [summary.get(k, "") for k in header])
        save_to_json(os.path.join(self.logs_filepath, "summary_statistics.json"),
                     dict(stats))

    # ------------------------------------------------------------------
    def run_experiment(self) -> None:
        if getattr(self.args, "evaluate_on_test_set_only", False):
            # reference flag: skip training, evaluate the stored best
            # checkpoints on the test set (experiment_builder.py:302-371)
            self.evaluate_test_set_using_the_best_models(
                top_n_models=getattr(self.args, "max_models_to_save", 5))
            return
        total_iters = self.args.total_epochs * self.args.total_iter_per_epoch
        iters_per_epoch = self.args.total_iter_per_epoch
        epochs_done_this_run = 0
        while self.state["current_iter"] < total_iters:
            epoch = self.state["current_iter"] // iters_per_epoch
            remaining = min(iters_per_epoch - (self.state["current_iter"] % iters_per_epoch),
                            total_iters - self.state["current_iter"])
            t0 = time.time()
            pbar = None
            if self.rank == 0:
                try:
                    from tqdm import tqdm
                    pbar = tqdm(total=remaining, desc=f"epoch {epoch}", leave=False)
                except ImportError:
                    pass
            for train_sample in self.data.get_train_batches(
                    total_batches=remaining, augment_images=self.augment_flag):
                epoch_idx = self.state["current_iter"] // iters_per_epoch
                losses = self.train_iteration(train_sample, epoch_idx,
                                              self.state["current_iter"])
                self.state["current_iter"] += 1
                if pbar is not None:
                    pbar.set_postfix(loss=f"{losses['loss']:.4f}",
                                     acc=f"{losses['accuracy']:.4f}")
                    pbar.update(1)
            if pbar is not None:
                pbar.close()

            if self.state["current_iter"] % iters_per_epoch == 0:
                val_losses = []
                n_val_batches = max(1, self.args.num_evaluation_tasks // self.args.batch_size)
                for val_sample in self.data.get_val_batches(total_batches=n_val_batches):
                    val_losses.append(self.evaluation_iteration(val_sample))
                summary = self._epoch_summary(val_losses, epoch, time.time() - t0)
                if summary.get("val_accuracy_mean", 0.0) > self.state["best_val_acc"]:
                    self.state["best_val_acc"] = summary.get("val_accuracy_mean", 0.0)
                    self.state["best_val_iter"] = self.state["current_iter"]
                    self.state["best_epoch"] = epoch
                self.pack_and_save_metrics(summary, first=(epoch == 0))
                self.save_models(epoch)
                self._print(f"epoch {epoch}: " + ", ".join(
                    f"{k}={v:.4f}" for k, v in summary.items() if isinstance(v, float)))
                epochs_done_this_run += 1
                if epochs_done_this_run >= self.args.total_epochs_before_pause and \
                        self.state["current_iter"] < total_iters:
                    self._print("pause requested by total_epochs_before_pause; exiting")
                    if self.dist is not None:
                        self.dist.barrier()
                    sys.exit(0)

        self.evaluate_test_set_using_the_best_models(
            top_n_models=getattr(self.args, "max_models_to_save", 5))

    # ------------------------------------------------------------------
    def evaluate_test_set_using_the_best_models(self, top_n_models: int = 5) -> Optional[Dict]:
        """Ensemble the top-N best-val checkpoints over the test stream
        (reference: ``experiment_builder.py:247-300``): per model, softmax
        target predictions per task; average across models; argmax; report
        accuracy mean/std."""
        if self.dist is not None and self.world_size > 1:
            # rank 0 writes the checkpoints every rank is about to load
            self.dist.barrier()
        stats = self.state["per_epoch_statistics"]
        if not stats.get("val_accuracy_mean"):
            # consistent across ranks: per_epoch_statistics is maintained
            # on every rank, so either all ranks skip or none do
            self._print("no per-epoch stats; skipping ensemble test")
            return None
        val_acc = np.array(stats["val_accuracy_mean"], dtype=np.float64)
        epochs = np.array(stats["epoch"], dtype=np.int64)
        order = np.argsort(val_acc)[::-1][:top_n_models]
        top_epochs = epochs[order]
        self._print(f"ensemble test over checkpoints of epochs {top_epochs.tolist()}")

        n_test_batches = max(1, self.args.num_evaluation_tasks // self.args.batch_size)
        per_model_preds = []   # list over models of [num_tasks, N*T, ways]
        labels_ref = []
        for e in top_epochs:
            self.model.load_model(self.saved_models_filepath, "train_model", int(e))
            preds_accum = []
            labels_accum = []
            for test_sample in self.data.get_test_batches(total_batches=n_test_batches):
                losses, logits = self.model.run_validation_iter(data_batch=test_sample)
                preds_accum.append(torch.softmax(logits.float(), dim=-1).cpu())
                y_t = test_sample[3]
                labels_accum.append(y_t.reshape(y_t.shape[0], -1).cpu())
            per_model_preds.append(torch.cat(preds_accum, dim=0))
            if not labels_ref:
                labels_ref = labels_accum
        labels = torch.cat(labels_ref, dim=0)
        ensemble = torch.stack(per_model_preds, dim=0).mean(dim=0)
        pred_labels = ensemble.argmax(dim=-1)
        per_task_acc = (pred_labels == labels).float().mean(dim=1)
        # world-size-invariant mean/std over ALL ranks' task shards
        n = float(per_task_acc.numel())
        s = float(per_task_acc.sum())
        q = float((per_task_acc * per_task_acc).sum())
        if self.dist is not None and self.world_size > 1:
            n, s, q = self.dist.all_reduce_sum_vector([n, s, q])
        acc_mean = s / max(n, 1.0)
        acc_std = float(np.sqrt(max(0.0, q / max(n, 1.0) - acc_mean * acc_mean)))
        result = {"test_accuracy_mean": acc_mean, "test_accuracy_std": acc_std}
        if self.rank == 0:
            save_statistics(self.logs_filepath,
                            ["test_accuracy_mean", "test_accuracy_std"],
                            filename="test_summary.csv", create=True)
            save_statistics(self.logs_filepath, [acc_mean, acc_std],
                            filename="test_summary.csv")
        self._print(f"test ensemble accuracy: {acc_mean:.4f} ± {acc_std:.4f}")
        return result
