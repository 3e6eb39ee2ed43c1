"""Evaluator protocol + inference_on_dataset loop.

Reference behavior: libai/evaluation/evaluator.py:37-278 and utils.py:24-53
(last-batch padding across dp ranks, gather-to-rank0, throughput log).
"""

import datetime
import logging
import time
from collections import OrderedDict
from contextlib import ExitStack

import torch

from ..data.structures import Instance
from ..utils import distributed as du

__all__ = ["DatasetEvaluator", "DatasetEvaluators", "inference_on_dataset",
           "flatten_results_dict"]


class DatasetEvaluator:
    def reset(self):
        pass

    def process(self, inputs, outputs):
        raise NotImplementedError

    def evaluate(self):
        raise NotImplementedError


class DatasetEvaluators(DatasetEvaluator):
    def __init__(self, evaluators):
        self._evaluators = evaluators

    def reset(self):
        for e in self._evaluators:
            e.reset()

    def process(self, inputs, outputs):
        for e in self._evaluators:
            e.process(inputs, outputs)

    def evaluate(self):
        results = OrderedDict()
        for e in self._evaluators:
            r = e.evaluate()
            if du.is_main_process() and r is not None:
                for k, v in r.items():
                    assert k not in results, f"duplicate eval key {k}"
                    results[k] = v
        return results


def flatten_results_dict(results):
    r = {}
    for k, v in results.items():
        if isinstance(v, dict):
            for kk, vv in flatten_results_dict(v).items():
                r[k + "/" + kk] = vv
        else:
            r[k] = v
    return r


def inference_on_dataset(model, data_loader, evaluator, eval_iter=None,
                         get_batch=None, pipeline_scheduler=None):
    """Eval loop: model(**batch) per batch; evaluator.process on the writer
    rank after DP gather (reference: evaluator.py:119-278).

    Under pipeline parallelism pass ``pipeline_scheduler``: every stage
    participates in the P2P forward (PipelineScheduler.run_eval), outputs
    exist on the LAST stage, and the evaluator runs on that stage's
    (dp0, tp0) rank; the final results dict is broadcast world-wide
    (reference runs test() on pipelined graphs the same way,
    evaluator.py:119-278)."""
    logger = logging.getLogger(__name__)
    if evaluator is None:
        return {}
    evaluator.reset()
    dutil = du.get_dist_util()
    pp = dutil.pipeline_parallel_size
    on_last_stage = pp == 1 or dutil.pipeline_parallel_rank == pp - 1
    # global rank that runs the evaluator: rank 0 for pp==1 (main process),
    # else the last stage's (dp0, tp0) rank (stage-major rank layout)
    writer_rank = 0 if pp == 1 else (
        (pp - 1) * dutil.data_parallel_size * dutil.tensor_parallel_size)
    is_writer = (du.is_main_process() if pp == 1 else
                 (on_last_stage and dutil.data_parallel_rank == 0
                  and dutil.tensor_parallel_rank == 0))
    total = len(data_loader) if hasattr(data_loader, "__len__") else None
    if eval_iter is not None and total is not None:
        total = min(total, eval_iter)
    device = du.get_device()
    num_warmup = min(5, (total or 10) - 1)
    start_time = time.perf_counter()
    total_compute_time = 0.0

    with ExitStack() as stack:
        if isinstance(model, torch.nn.Module):
            stack.enter_context(inference_context(model))
        stack.enter_context(torch.no_grad())
        for idx, inputs in enumerate(data_loader):
            if eval_iter is not None and idx >= eval_iter:
                break
            if idx == num_warmup:
                start_time = time.perf_counter()
                total_compute_time = 0.0
            if isinstance(inputs, Instance):
                data = inputs.to_dict()
            else:
                data = inputs
            data = {
                k: (v.to(device, non_blocking=True) if torch.is_tensor(v) else v)
                for k, v in data.items()
            }
            t0 = time.perf_counter()
            if pipeline_scheduler is not None:
                outputs = pipeline_scheduler.run_eval(data)
            else:
                outputs = model(**data)
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            total_compute_time += time.perf_counter() - t0

            if not on_last_stage:
                continue  # outputs live on the last stage only
            # gather DP-sharded outputs + labels across the stage's DP group
            gathered_out = {
                k: du.tensor_to_rank0(v) if torch.is_tensor(v) else v
                for k, v in outputs.items()
            }
            gathered_in = {
                k: du.tensor_to_rank0(v) if torch.is_tensor(v) else v
                for k, v in data.items()
            }
            if is_writer:
                evaluator.process(gathered_in, gathered_out)

    total_time = time.perf_counter() - start_time
    if total:
        logger.info(
            f"Total inference time: {datetime.timedelta(seconds=total_time)} "
            f"({total_compute_time:.3f}s compute)"
        )
    results = evaluator.evaluate() if is_writer else None
    results = du.broadcast_py_object(results, src=writer_rank)
    return results if results is not None else {}


class inference_context:
    def __init__(self, model):
        self.model = model

    def __enter__(self):
        self.training_mode = self.model.training
        self.model.eval()

    def __exit__(self, *args):
        self.model.train(self.training_mode)
