"""Canonical training defaults (reference: configs/common/train.py:8-150)."""

from libai_amd.config import ConfigDict

train = ConfigDict(
    output_dir="./output",
    train_micro_batch_size=4,
    global_batch_size=None,
    num_accumulation_steps=None,
    train_iter=10000,
    train_epoch=0,
    consumed_train_samples=0,
    seed=1234,
    log_period=20,
    # mixed precision: bf16 params + fp32 master weights (no loss scaler)
    amp=dict(enabled=False),
    activation_checkpoint=dict(enabled=False),
    zero_optimization=dict(enabled=False, stage=1),
    checkpointer=dict(period=5000, max_to_keep=100),
    evaluation=dict(
        enabled=False,
        evaluator=None,
        eval_period=5000,
        eval_iter=100,
    ),
    load_weight="",
    resume=False,
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=None,
        custom_pipeline_stage_id=None,
    ),
    rdma_enabled=False,
    scheduler=None,
    # torch.profiler window: dict(start_iter=10, end_iter=13[, with_stack])
    # -> chrome traces in <output_dir>/profiler/ (None = off)
    profiler=None,
    # experimental fp8 e4m3 FORWARD GEMMs (bwd stays bf16); ~1.5-2x on the
    # Linear1D matmuls (profiles/gemm_roofline.md)
    fp8=dict(enabled=False),
)
