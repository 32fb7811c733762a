"""Checkpoint-polling evaluator — re-expression of /root/reference/src/nn_eval.py.

Behavior kept: restore latest checkpoint (nn_eval.py:70-78), skip when the
step hasn't advanced (:84-88), single full-validation-set batch
accuracy+loss (:95-100), scraper-compatible result line (:102), optional
TensorBoard scalars (:107-110), poll loop every eval_interval_secs
(:136-140).
"""

from __future__ import annotations

import logging
import time

import torch

from ..models import build_model
from .supervisor import Supervisor

log = logging.getLogger("dmnist.eval")


def do_eval(model, images, labels, device, dtype, batch_size: int = 10000):
    """Full-set eval in chunks (the reference feeds all 10k in one batch;
    we chunk to bound activation memory, identical result)."""
    model.eval()
    total = images.shape[0]
    correct_sum = 0.0
    loss_sum = 0.0
    with torch.no_grad():
        for s in range(0, total, batch_size):
            e = min(s + batch_size, total)
            x = images[s:e].to(device=device, dtype=dtype)
            y = labels[s:e].to(device=device)
            logits = model(x, train=False)
            loss, correct = model.loss_and_accuracy(logits, y)
            n = e - s
            correct_sum += float(correct) * n  # loss_and_accuracy returns mean acc
            loss_sum += float(loss) * n
    return correct_sum / total, loss_sum / total


def evaluate(dataset, flags, writer=None):
    """Poll loop. Returns list of (step, precision, loss)."""
    device = torch.device(flags.device) if flags.device not in ("auto", "") else (
        torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu"))
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    start_time = time.time()
    last_step = -1
    results = []
    n_evals = 0
    while True:
        restored = Supervisor.restore(flags.checkpoint_dir)
        if restored is not None:
            step, payload = restored
            if step != last_step:
                model = build_model(payload.get("model", flags.model),
                                    seed=payload.get("seed", 66478),
                                    compute_dtype=dtype).to(device)
                with torch.no_grad():
                    for name in model.PARAM_ORDER:
                        getattr(model, name).copy_(
                            payload["model_state"][name].to(device))
                # first-class GPU eval: FlatParams builds the bf16 shadows
                # AND the pre-transposed shadow_T copies, so evaluation
                # runs the SAME transposed-GEMM kernel path as training
                # (round-1 built shadows ad hoc without shadows_T and fell
                # onto the non-transposed GEMMs)
                from ..parallel import FlatParams
                FlatParams(model, device=device, compute_dtype=dtype)
                print('Succesfully loaded model from %s at step=%s.' %
                      (Supervisor.latest_checkpoint(flags.checkpoint_dir)[1], step),
                      flush=True)
                images = torch.as_tensor(dataset.images)
                labels = torch.as_tensor(dataset.labels)
                acc, loss = do_eval(model, images, labels, device, dtype)
                # exact reference line (nn_eval.py:102) — scraped by
                # benchmark.py extract_times_losses_precision
                print('Num examples: %d  Precision @ 1: %f Loss: %f Time: %f' %
                      (dataset.num_examples, acc, loss,
                       time.time() - start_time), flush=True)
                if writer is not None:
                    writer.add_scalar("Validation Accuracy", acc, step)
                    writer.add_scalar("Validation Loss", loss, step)
                results.append((step, acc, loss))
                last_step = step
                n_evals += 1
        if flags.run_once or (flags.max_evals and n_evals >= flags.max_evals):
            break
        time.sleep(flags.eval_interval_secs)
    return results
